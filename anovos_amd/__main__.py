"""python -m anovos_amd <config_path> <run_type> (reference parity:
``anovos/__main__.py`` :3-5)."""

import sys

from anovos_amd import workflow

if __name__ == "__main__":
    config_path = sys.argv[1] if len(sys.argv) > 1 else "config/configs.yaml"
    run_type = sys.argv[2] if len(sys.argv) > 2 else "local"
    workflow.run(config_path, run_type)
