"""End-to-end workflow test: YAML config → full pipeline → HTML report
(reference parity: the full-demo CI workflow running configs.yaml;
here a compact income-like synthetic dataset on CPU)."""

import os

import numpy as np
import pandas as pd
import pytest
import yaml

from anovos_amd import workflow


@pytest.fixture
def income_csv(tmp_path):
    rng = np.random.default_rng(7)
    n = 3000
    pdf = pd.DataFrame(
        {
            "ifa": [f"id_{i}" for i in range(n)],
            "age": rng.integers(17, 90, n),
            "workclass": rng.choice(["Private", "Self-emp", "Gov", "Other"], n),
            "fnlwgt": rng.integers(10000, 1000000, n),
            "education": rng.choice(["HS-grad", "Bachelors", "Masters", "Doctorate", "Some-college"], n),
            "hours-per-week": rng.integers(1, 99, n),
            "capital-gain": np.where(rng.random(n) < 0.9, 0, rng.integers(1, 99999, n)),
            "income": rng.choice(["<=50K", ">50K"], n, p=[0.76, 0.24]),
        }
    )
    p = tmp_path / "income.csv"
    pdf.to_csv(p, index=False)
    return str(p)


def make_config(income_csv, tmp_path):
    out = str(tmp_path / "out")
    rep = str(tmp_path / "report")
    return {
        "input_dataset": {
            "read_dataset": {"file_path": income_csv, "file_type": "csv",
                             "file_configs": {"header": True, "inferSchema": True}},
            "delete_column": ["fnlwgt"],
        },
        "stats_generator": {
            "metric": ["global_summary", "measures_of_counts", "measures_of_centralTendency",
                       "measures_of_cardinality", "measures_of_dispersion", "measures_of_percentiles",
                       "measures_of_shape"],
            "metric_args": {"list_of_cols": "all", "drop_cols": ["ifa"]},
        },
        "quality_checker": {
            "duplicate_detection": {"list_of_cols": "all", "drop_cols": ["ifa"], "treatment": True},
            "nullColumns_detection": {"list_of_cols": "all", "drop_cols": ["ifa", "income"],
                                       "treatment": True, "treatment_method": "MMM"},
            "biasedness_detection": {"list_of_cols": "all", "drop_cols": ["ifa", "income"], "treatment": False},
        },
        "association_evaluator": {
            "IV_calculation": {"list_of_cols": "all", "drop_cols": ["ifa"],
                               "label_col": "income", "event_label": ">50K"},
        },
        "transformers": {
            "numerical_mathops": {
                "attribute_binning": {"list_of_cols": ["age", "hours-per-week"],
                                       "method_type": "equal_range", "bin_size": 10,
                                       "output_mode": "append"},
            },
            "numerical_rescaling": {
                "z_standardization": {"list_of_cols": ["capital-gain"], "output_mode": "append"},
            },
        },
        "report_preprocessing": {
            "master_path": rep,
            "charts_to_objects": {"list_of_cols": "all", "drop_cols": ["ifa"],
                                   "label_col": "income", "event_label": ">50K",
                                   "bin_method": "equal_frequency", "bin_size": 10,
                                   "source_path": str(tmp_path / "inter")},
        },
        "report_generation": {
            "master_path": rep,
            "final_report_path": rep,
            "label_col": "income",
            "event_label": ">50K",
        },
        "write_main": {"file_path": out, "file_type": "parquet"},
        "write_stats": {"file_path": out, "file_type": "csv",
                        "file_configs": {"header": True, "mode": "overwrite"}},
    }


def test_workflow_end_to_end(income_csv, tmp_path):
    cfg = make_config(income_csv, tmp_path)
    cfg_path = tmp_path / "cfg.yaml"
    with open(cfg_path, "w") as f:
        yaml.safe_dump(cfg, f, sort_keys=False)  # stage order = YAML order
    df = workflow.run(str(cfg_path))
    assert df is not None
    assert "fnlwgt" not in df.columns
    assert "age_binned" in df.columns
    assert "capital-gain_scaled" in df.columns
    rep = tmp_path / "report"
    assert (rep / "global_summary.csv").exists()
    assert (rep / "measures_of_counts.csv").exists()
    assert (rep / "nullColumns_detection.csv").exists()
    assert (rep / "IV_calculation.csv").exists()
    assert (rep / "ml_anovos_report.html").exists()
    html = open(rep / "ml_anovos_report.html").read()
    assert "Attribute Associations" in html
    # round-2 content-parity elements (docs/REPORT_PARITY.md)
    assert "Key Report Highlights" in html        # exec-summary narrative
    assert "Data Diagnosis" in html               # ✔/✘ matrix
    assert "Label Distribution" in html           # label pie
    assert "Metric Dictionary" in html            # wiki tab
    assert "Row-Level Checks" in html and "Column-Level Checks" in html
    assert "Global Summary" in html
    # main dataset persisted
    assert any("final_dataset" in d for d in os.listdir(tmp_path / "out"))


def test_workflow_basic_report_short_circuit(income_csv, tmp_path):
    cfg = {
        "input_dataset": {
            "read_dataset": {"file_path": income_csv, "file_type": "csv",
                             "file_configs": {"header": True, "inferSchema": True}},
        },
        "anovos_basic_report": {
            "basic_report": True,
            "report_args": {"id_col": "ifa", "label_col": "income", "event_label": ">50K",
                            "output_path": str(tmp_path / "basic")},
        },
        "stats_generator": {"metric": ["global_summary"], "metric_args": {}},
    }
    cfg_path = tmp_path / "cfg_basic.yaml"
    with open(cfg_path, "w") as f:
        yaml.safe_dump(cfg, f)
    workflow.run(str(cfg_path))
    assert (tmp_path / "basic" / "basic_report.html").exists()


def test_workflow_time_series_config(tmp_path, monkeypatch):
    """configs_time_series.yaml shape: ts auto-detect + stability over
    snapshots (compact synthetic variant)."""
    import subprocess
    import sys

    monkeypatch.chdir(tmp_path)
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    sys.path.insert(0, os.path.join(repo, "tools"))
    import make_income_data as mid

    df = mid.add_ts_cols(mid.make(3000))
    os.makedirs("data/income_dataset/csv", exist_ok=True)
    df.to_csv("data/income_dataset/csv/part-00000.csv", index=False)
    for k in range(12):
        snap = mid.make(800, seed=100 + k)
        snap["age"] = snap["age"] * (1 + 0.01 * k)
        d = f"data/income_dataset/snapshot{k + 1:02d}/csv"
        os.makedirs(d, exist_ok=True)
        snap.to_csv(os.path.join(d, "part-00000.csv"), index=False)
    from anovos_amd import workflow

    workflow.run(os.path.join(repo, "config", "configs_time_series.yaml"))
    assert os.path.exists("report_stats/stability_index.csv")
    assert os.path.exists("report_stats/ts_cols_stats.csv")
    si = __import__("pandas").read_csv("report_stats/stability_index.csv")
    assert "stability_index" in si.columns and len(si) >= 3
    assert os.path.exists("report_stats/ml_anovos_report.html")


def test_workflow_geospatial_config(tmp_path, monkeypatch):
    """configs_geospatial.yaml shape: autodetect + geo transformations."""
    import sys

    monkeypatch.chdir(tmp_path)
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    sys.path.insert(0, os.path.join(repo, "tools"))
    import make_income_data as mid

    df = mid.add_geo_cols(mid.make(3000))
    os.makedirs("data/income_dataset/csv", exist_ok=True)
    df.to_csv("data/income_dataset/csv/part-00000.csv", index=False)
    from anovos_amd import workflow

    out = workflow.run(os.path.join(repo, "config", "configs_geospatial.yaml"))
    assert "lat_long_in_usa" in out.columns
    assert "lat_long_geohash" in out.columns
    assert "radius_of_gyration" in out.columns
    assert os.path.exists("report_stats/Overall_Summary_1_latitude_longitude.csv")


def test_workflow_inmemory_pipeline_matches_materialized(income_csv, tmp_path, monkeypatch):
    """ANOVOS_AMD_INMEMORY_PIPELINE=1 skips the per-stage save/reread
    barrier; the final frame and stats must be identical to the
    materialized run (the barrier is optional, not semantic)."""
    import pandas as pd

    cfg = make_config(income_csv, tmp_path)
    cfg_path = tmp_path / "cfg2.yaml"
    with open(cfg_path, "w") as f:
        yaml.safe_dump(cfg, f, sort_keys=False)
    df_mat = workflow.run(str(cfg_path))
    iv_mat = pd.read_csv(tmp_path / "report" / "IV_calculation.csv")

    monkeypatch.setenv("ANOVOS_AMD_INMEMORY_PIPELINE", "1")
    # rerun into fresh dirs
    cfg2 = make_config(income_csv, tmp_path / "mem")
    (tmp_path / "mem").mkdir(exist_ok=True)
    cfg2_path = tmp_path / "cfg3.yaml"
    with open(cfg2_path, "w") as f:
        yaml.safe_dump(cfg2, f, sort_keys=False)
    df_mem = workflow.run(str(cfg2_path))
    iv_mem = pd.read_csv(tmp_path / "mem" / "report" / "IV_calculation.csv")

    assert sorted(df_mat.columns) == sorted(df_mem.columns)
    m = iv_mat.merge(iv_mem, on="attribute", suffixes=("_a", "_b"))
    assert (abs(m["iv_a"] - m["iv_b"]) < 1e-9).all()
    for c in df_mat.columns:
        a, b = df_mat.col(c), df_mem.col(c)
        if a.kind == "numerical":
            import torch as _t

            assert float(_t.nan_to_num(a.data.double() - b.data.double()).abs().max()) < 1e-6, c


def test_bench_json_contract():
    """The driver's bench contract: one JSON line on stdout with the
    required keys, value == whole-job aggregate, ms_per_step > 0."""
    import json
    import os
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run([sys.executable, os.path.join(repo, "bench.py"),
                        "--steps", "2", "--warmup", "1", "--rows", "200000"],
                       capture_output=True, text=True, timeout=600, cwd=repo)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    j = json.loads(line)
    for k in ("metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
              "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config"):
        assert k in j, k
    assert j["n_gpus"] == 1 and j["steps"] == 2 and j["warmup"] == 1
    assert j["higher_is_better"] is True and j["scaling"] == "weak"
    assert j["data"] == "synthetic"
    assert j["value"] > 0 and j["ms_per_step"] > 0
    # whole-job aggregate: rows_per_gpu * n_gpus / (ms_per_step/1000)
    expect = j["config"]["rows_per_gpu"] * j["n_gpus"] / (j["ms_per_step"] / 1000.0)
    assert j["value"] == pytest.approx(expect, rel=1e-6)
