"""Every examples/ script must run standalone (the reference ships the
equivalent set as examples/notebooks/*.ipynb; these are their runnable
per-module counterparts)."""

import glob
import os
import subprocess
import sys

import pytest

EX = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "examples")
SCRIPTS = sorted(
    os.path.basename(p)
    for p in glob.glob(os.path.join(EX, "*.py"))
    if not os.path.basename(p).startswith("_") and os.path.basename(p) != "quickstart.py"
)


@pytest.mark.parametrize("script", SCRIPTS)
def test_example_runs(script):
    r = subprocess.run([sys.executable, script], cwd=EX, capture_output=True, text=True,
                       timeout=240)
    assert r.returncode == 0, f"{script}\n{r.stderr[-2000:]}"
