"""Version of the anovos_amd package (reference: src/main/anovos/version.py:1)."""

__version__ = "0.2.0"
