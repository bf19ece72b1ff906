"""Adapt a workflow YAML for containerized execution (reference parity:
``local/rewrite_configuration.py`` :1-154 — same CLI contract: argv[1]
is the config; writes ``config.yaml.tmp`` with every input path moved
under /data and every output path under /output, writes the host-side
common input root to ``data_directory.tmp``, and prints the rewritten
keys).

MI355X-native notes: the container is the ROCm image (see
local/Dockerfile); GPU access comes from --device=/dev/kfd,/dev/dri
passthrough in run_workload.sh, not from the config."""

from __future__ import annotations

import os
import pathlib
import sys

import yaml

DATA_MOUNT = pathlib.Path("/data")
OUTPUT_MOUNT = pathlib.Path("/output")
OUT_CONFIG = "config.yaml.tmp"
OUT_DATAROOT = "data_directory.tmp"

# features that need services a one-off container doesn't have
UNSUPPORTED = {"write_feast_features"}

# keys whose values are INPUT paths, wherever they appear in the tree
INPUT_PATH_KEYS = {"metricDict_path", "dataDict_path", "source_path", "model_path"}
# (block, key) pairs whose values are OUTPUT paths
OUTPUT_PATH_BLOCKS = [
    ("write_intermediate", "file_path"),
    ("write_main", "file_path"),
    ("write_stats", "file_path"),
    ("report_preprocessing", "master_path"),
    ("report_generation", "master_path"),
    ("report_generation", "final_report_path"),
]


def _walk(node, fn, crumbs=()):
    """Depth-first visit of every dict in the config tree."""
    if isinstance(node, dict):
        fn(node, crumbs)
        for k, v in node.items():
            _walk(v, fn, crumbs + (k,))


def collect_input_paths(cfg):
    paths = []

    def visit(d, crumbs):
        for k, v in d.items():
            if k in UNSUPPORTED:
                raise ValueError(f"{k} is not supported in Docker execution mode.")
            if k == "read_dataset" and isinstance(v, dict) and "file_path" in v:
                paths.append(str(v["file_path"]))
            elif k in INPUT_PATH_KEYS and isinstance(v, str) and v != "NA":
                paths.append(v)

    _walk(cfg, visit)
    for p in paths:
        if p.startswith(("dbfs:", "s3:", "wasbs:", "abfss:")):
            raise ValueError(f"Only local paths are supported: {p}")
    return paths


def rewrite(cfg, data_root):
    def to_data(p):
        rel = pathlib.Path(p).absolute().relative_to(data_root)
        return str(DATA_MOUNT / rel)

    changed = []

    def visit(d, crumbs):
        for k, v in d.items():
            if k == "read_dataset" and isinstance(v, dict) and "file_path" in v:
                new = to_data(v["file_path"])
                changed.append((crumbs + (k, "file_path"), v["file_path"], new))
                v["file_path"] = new
            elif k in INPUT_PATH_KEYS and isinstance(v, str) and v != "NA":
                new = to_data(v)
                changed.append((crumbs + (k,), v, new))
                d[k] = new

    _walk(cfg, visit)

    def out_rewrite(block, key):
        if isinstance(block, dict) and key in block:
            new = str(OUTPUT_MOUNT / pathlib.Path(block[key]))
            changed.append(((key,), block[key], new))
            block[key] = new

    for blk, key in OUTPUT_PATH_BLOCKS:
        out_rewrite(cfg.get(blk), key)
    basic = cfg.get("anovos_basic_report")
    if isinstance(basic, dict):
        out_rewrite(basic.get("report_args"), "output_path")
    return changed


def main(argv):
    config_file = argv[1]
    with open(config_file) as f:
        cfg = yaml.safe_load(f)

    inputs = collect_input_paths(cfg)
    if not inputs:
        raise ValueError("configuration references no local input datasets")
    data_root = pathlib.Path(
        os.path.commonpath([str(pathlib.Path(p).absolute()) for p in inputs])
    ).absolute()

    changed = rewrite(cfg, data_root)

    with open(OUT_CONFIG, "w") as f:
        yaml.safe_dump(cfg, f, sort_keys=False)
    with open(OUT_DATAROOT, "w") as f:
        f.write(str(data_root))

    print("Adapted configuration for execution inside an anovos-amd-worker container:")
    for crumbs, old, new in changed:
        print(f"{'.'.join(crumbs)}: {old} -> {new}")


if __name__ == "__main__":
    main(sys.argv)
