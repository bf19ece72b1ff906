"""Packaging for anovos_amd (reference ships setup.py; the HIP extension
still builds IN-TREE via `python __graft_entry__.py build` / `make build`
so the .so travels with the source checkout)."""

from setuptools import find_packages, setup

setup(
    name="anovos-amd",
    version="0.2.0",
    description="MI355X-native columnar feature-engineering engine (Anovos capabilities, no Spark)",
    packages=find_packages(include=["anovos_amd", "anovos_amd.*"]),
    package_data={
        "anovos_amd.ops.hip": ["*.hip", "*.so", "build.py"],
        "anovos_amd.feature_recommender": ["data/*.csv", "data/README.md"],
        "anovos_amd.data_report": ["data/*.csv", "data/README.md"],
    },
    python_requires=">=3.10",
    install_requires=["numpy", "pandas", "pyarrow", "plotly", "scipy", "scikit-learn", "pyyaml", "sympy"],
)
