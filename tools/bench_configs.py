#!/usr/bin/env python3
"""Wall-clock benchmark of the YAML workflow configs on the local device
(configs_time_series / configs_geospatial / the
income full pipeline, measured on MI355X, numbers recorded in
docs/BENCHMARKS.md).

Generates the synthetic datasets the config expects (sized via --rows),
runs `workflow.run(<config>)` once end to end, prints a JSON line with
the wall time. All stages, including report generation, are inside the
timed region — this is the reference's own metric of record (per-stage
wall-clock, workflow.py:242).
"""

from __future__ import annotations

import argparse
import json
import os
import subprocess
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "tools"))


def prepare(config: str, rows: int):
    import make_income_data as mid

    gen = [sys.executable, os.path.join(REPO, "tools", "make_income_data.py"),
           "--rows", str(rows), "--out", "data/income_dataset"]
    if "time_series" in config:
        gen += ["--ts", "--snapshots", "12"]
        subprocess.run(gen, check=True, capture_output=True)
    elif "geospatial" in config:
        gen += ["--geo"]
        subprocess.run(gen, check=True, capture_output=True)
    elif "sales" in config:
        import make_demo_data as mdd

        mdd.make_sales("data/sales_dataset", rows=rows)
        import pandas as pd

        pd.DataFrame({"Metric": ["mean"], "Definition": ["mean"]}).to_csv("data/metric_dictionary.csv", index=False)
    else:
        subprocess.run(gen, check=True, capture_output=True)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("config")
    ap.add_argument("--rows", type=int, default=2_000_000)
    ap.add_argument("--workdir", default="")
    a = ap.parse_args()
    # /tmp, NOT gpurun_out: generated datasets would blow the 64 MiB
    # copy-back budget of the GPU harness
    wd = a.workdir or os.path.join("/tmp", "cfgbench_" + os.path.basename(a.config).replace(".yaml", ""))
    os.makedirs(wd, exist_ok=True)
    cfg_path = os.path.join(REPO, "config", a.config) if not os.path.isabs(a.config) else a.config
    os.chdir(wd)
    prepare(a.config, a.rows)

    import torch

    from anovos_amd import workflow

    t0 = time.perf_counter()
    workflow.run(cfg_path)
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t1 = time.perf_counter()
    print(json.dumps({
        "config": a.config,
        "rows": a.rows,
        "wall_s": round(t1 - t0, 3),
        "device": "cuda" if torch.cuda.is_available() else "cpu",
    }))


if __name__ == "__main__":
    main()
