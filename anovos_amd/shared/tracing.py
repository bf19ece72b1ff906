"""Lightweight op tracing (aux subsystem — SURVEY.md §5: the reference
only had wall-clock loguru lines; here every public analyzer/transformer
entry emits an nvtx/roctx range so `rocprofv3 --kernel-trace` timelines
attribute kernels to engine ops, plus an optional wall-clock registry).

Zero overhead when no profiler is attached (torch nvtx push/pop are
cheap no-ops without a collector); disable entirely with
ANOVOS_AMD_TRACE=0."""

from __future__ import annotations

import functools
import os
import time
from collections import defaultdict
from typing import Dict, List

import torch

_ENABLED = os.environ.get("ANOVOS_AMD_TRACE", "1") != "0"
_WALL = defaultdict(float)
_CALLS = defaultdict(int)
_COLLECT_WALL = os.environ.get("ANOVOS_AMD_TRACE_WALL", "0") == "1"


def traced(fn):
    """Decorator: wrap an engine op in an nvtx/roctx range named after it."""

    @functools.wraps(fn)
    def wrapper(*args, **kwargs):
        if not _ENABLED:
            return fn(*args, **kwargs)
        name = fn.__module__.rsplit(".", 1)[-1] + "." + fn.__name__
        torch.cuda.nvtx.range_push(name)
        t0 = time.perf_counter() if _COLLECT_WALL else 0.0
        try:
            return fn(*args, **kwargs)
        finally:
            torch.cuda.nvtx.range_pop()
            if _COLLECT_WALL:
                _WALL[name] += time.perf_counter() - t0
                _CALLS[name] += 1

    return wrapper


def wall_report() -> Dict[str, List]:
    """Per-op cumulative wall clock (only populated with
    ANOVOS_AMD_TRACE_WALL=1)."""
    return {k: [round(v * 1000, 3), _CALLS[k]] for k, v in sorted(_WALL.items(), key=lambda kv: -kv[1])}


def reset():
    _WALL.clear()
    _CALLS.clear()
