"""Distributed runtime: one process per GPU, RCCL over xGMI.

Replaces the reference's Spark shuffle/collect machinery (SURVEY.md §2.10
collective inventory). Design per BASELINE.json: stat payloads are tiny
(KBs) so collectives are batched across columns into single fused
all-reduces; histogram/sketch merges are all-reduce sums; dictionary
merges are all-gather of host objects. Collectives may run on a side HIP
stream overlapped with the next column batch (see ops/stats.py).

Backend: "nccl" (RCCL on ROCm) when CUDA/HIP devices are visible, else
"gloo" (CPU tests, world_size>1 works in the CI container).
"""

from __future__ import annotations

import datetime
import os
from typing import Any, List, Optional

import torch
import torch.distributed as td

_initialized = False


def is_dist() -> bool:
    return td.is_available() and td.is_initialized()


def init_from_env(timeout_s: int = 600) -> None:
    """Initialize torch.distributed from torchrun env vars if present."""
    global _initialized
    if is_dist() or "WORLD_SIZE" not in os.environ:
        return
    ws = int(os.environ.get("WORLD_SIZE", "1"))
    if ws <= 1:
        return
    backend = os.environ.get("ANOVOS_AMD_DIST_BACKEND") or ("nccl" if torch.cuda.is_available() else "gloo")
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29517")
    td.init_process_group(backend=backend, timeout=datetime.timedelta(seconds=timeout_s))
    if torch.cuda.is_available():
        # ranks may oversubscribe one device in tests (gloo backend)
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")) % max(torch.cuda.device_count(), 1))
    _initialized = True


def rank() -> int:
    return td.get_rank() if is_dist() else 0


def world_size() -> int:
    return td.get_world_size() if is_dist() else 1


def barrier() -> None:
    if is_dist():
        td.barrier()


def all_reduce_scalar(x, op: str = "sum"):
    """All-reduce a python scalar; returns python number."""
    if not is_dist():
        return x
    t = torch.tensor([float(x)], dtype=torch.float64)
    if torch.cuda.is_available():
        t = t.cuda()
    td.all_reduce(t, op=_op(op))
    v = t.item()
    return int(v) if isinstance(x, int) and op in ("sum", "min", "max") else v


def all_reduce_(t: torch.Tensor, op: str = "sum") -> torch.Tensor:
    """In-place all-reduce of a tensor (fused stat vectors, histograms,
    HLL registers with op='max', min/max vectors)."""
    if is_dist():
        td.all_reduce(t, op=_op(op))
    return t


def all_gather_object(obj: Any) -> List[Any]:
    """All-gather arbitrary host objects (dictionary merges, tiny stats)."""
    if not is_dist():
        return [obj]
    out = [None] * world_size()
    td.all_gather_object(out, obj)
    return out


def broadcast_object(obj: Any, src: int = 0) -> Any:
    if not is_dist():
        return obj
    box = [obj if rank() == src else None]
    td.broadcast_object_list(box, src=src)
    return box[0]


def _op(op: str):
    return {"sum": td.ReduceOp.SUM, "min": td.ReduceOp.MIN, "max": td.ReduceOp.MAX}[op]


class SideStream:
    """A side HIP stream used to overlap collectives with the next column
    batch (SURVEY.md §2.10 overlap rule). No-op on CPU."""

    def __init__(self):
        self.stream: Optional[torch.cuda.Stream] = torch.cuda.Stream() if torch.cuda.is_available() else None

    def __enter__(self):
        if self.stream is not None:
            self.stream.wait_stream(torch.cuda.current_stream())
            self._ctx = torch.cuda.stream(self.stream)
            self._ctx.__enter__()
        return self

    def __exit__(self, *a):
        if self.stream is not None:
            self._ctx.__exit__(*a)

    def join(self):
        if self.stream is not None:
            torch.cuda.current_stream().wait_stream(self.stream)
