"""local/rewrite_configuration.py contract (reference parity:
local/rewrite_configuration.py + run_workload.sh): inputs re-rooted to
/data by common parent, outputs to /output, tmp artifacts written,
unsupported features rejected."""

import os
import subprocess
import sys

import pytest
import yaml

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SCRIPT = os.path.join(REPO, "local", "rewrite_configuration.py")


def _run(tmp_path, cfg):
    p = tmp_path / "cfg.yaml"
    with open(p, "w") as f:
        yaml.safe_dump(cfg, f, sort_keys=False)
    return subprocess.run([sys.executable, SCRIPT, str(p)], cwd=tmp_path,
                          capture_output=True, text=True)


def test_rewrite_paths(tmp_path):
    (tmp_path / "data" / "ds1" / "csv").mkdir(parents=True)
    (tmp_path / "data" / "ds2").mkdir(parents=True)
    cfg = {
        "input_dataset": {"read_dataset": {"file_path": str(tmp_path / "data/ds1/csv"),
                                            "file_type": "csv"}},
        "drift_detector": {"drift_statistics": {"configs": {"source_path": str(tmp_path / "data/ds2"),
                                                             "method_type": "all"}}},
        "report_preprocessing": {"master_path": "report_stats"},
        "report_generation": {"master_path": "report_stats", "final_report_path": "report_stats"},
        "write_intermediate": {"file_path": "intermediate", "file_type": "csv"},
    }
    r = _run(tmp_path, cfg)
    assert r.returncode == 0, r.stderr
    new = yaml.safe_load(open(tmp_path / "config.yaml.tmp"))
    assert new["input_dataset"]["read_dataset"]["file_path"] == "/data/ds1/csv"
    assert new["drift_detector"]["drift_statistics"]["configs"]["source_path"] == "/data/ds2"
    assert new["report_preprocessing"]["master_path"] == "/output/report_stats"
    assert new["report_generation"]["final_report_path"] == "/output/report_stats"
    assert new["write_intermediate"]["file_path"] == "/output/intermediate"
    root = open(tmp_path / "data_directory.tmp").read()
    assert root == str(tmp_path / "data")
    assert "->" in r.stdout  # diff printed


def test_rejects_remote_paths(tmp_path):
    cfg = {"input_dataset": {"read_dataset": {"file_path": "s3://bucket/x", "file_type": "csv"}}}
    r = _run(tmp_path, cfg)
    assert r.returncode != 0
    assert "Only local paths" in (r.stderr + r.stdout)


def test_rejects_unsupported_feature(tmp_path):
    cfg = {
        "input_dataset": {"read_dataset": {"file_path": str(tmp_path), "file_type": "csv"}},
        "feature_store": {"write_feast_features": {"x": 1}},
    }
    r = _run(tmp_path, cfg)
    assert r.returncode != 0
    assert "not supported in Docker execution mode" in (r.stderr + r.stdout)
