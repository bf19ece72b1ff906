"""GPU end-to-end: the full income YAML pipeline on cuda:0, HIP kernels
loaded (fails loudly if the extension is missing on a GPU box)."""

import os
import sys

import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="needs MI355X")


@requires_gpu
def test_income_workflow_on_gpu(tmp_path, monkeypatch):
    from anovos_amd.ops import backend

    assert backend.require_hip() is not None, "HIP extension must load on a GPU box"

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    monkeypatch.chdir(tmp_path)
    sys.path.insert(0, os.path.join(repo, "tools"))
    import make_income_data

    df = make_income_data.make(20000)
    os.makedirs("data/income_dataset/csv", exist_ok=True)
    os.makedirs("data/income_dataset/source/csv", exist_ok=True)
    df.to_csv("data/income_dataset/csv/part-00000.csv", index=False)
    src = make_income_data.make(20000, seed=12)
    src["age"] = src["age"] * 1.05
    src.to_csv("data/income_dataset/source/csv/part-00000.csv", index=False)

    from anovos_amd import workflow

    out = workflow.run(os.path.join(repo, "config", "configs.yaml"), device="cuda:0")
    assert out is not None
    assert out.device.type == "cuda"
    assert os.path.exists("report_stats/ml_anovos_report.html")
    assert os.path.getsize("report_stats/ml_anovos_report.html") > 1_000_000
