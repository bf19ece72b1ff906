"""Distributed runtime: one process per GPU, RCCL over xGMI.

Replaces the reference's Spark shuffle/collect machinery (SURVEY.md §2.10
collective inventory). Design per BASELINE.json: stat payloads are tiny
(KBs) so collectives are batched across columns into single fused
all-reduces; histogram/sketch merges are all-reduce sums; variable-length
value/count merges go through the tensorized ``all_gather_tensor`` (a
length exchange + one padded all-gather — device-resident under RCCL,
no host pickling). ``all_gather_object`` remains only for cold paths
(string dictionaries, model artifacts). Collectives may run on a side
HIP stream overlapped with the next column batch (see ops/stats.py).

Backend: "nccl" (RCCL on ROCm) when CUDA/HIP devices are visible, else
"gloo" (CPU tests, world_size>1 works in the CI container). Every
wrapper stages tensors onto the backend's collective device — cuda for
RCCL, host for gloo gathers (gloo only supports CUDA tensors for
broadcast/all_reduce) — and returns them on the caller's device.
"""

from __future__ import annotations

import datetime
import os
from typing import Any, List, Optional, Sequence

import torch
import torch.distributed as td

_initialized = False


def is_dist() -> bool:
    return td.is_available() and td.is_initialized()


def backend() -> str:
    """Active process-group backend name ("nccl" = RCCL on ROCm, "gloo"),
    or "" when not distributed."""
    return str(td.get_backend()) if is_dist() else ""


def init_from_env(timeout_s: int = 600) -> None:
    """Initialize torch.distributed from torchrun env vars if present."""
    global _initialized
    if is_dist() or "WORLD_SIZE" not in os.environ:
        return
    ws = int(os.environ.get("WORLD_SIZE", "1"))
    if ws <= 1:
        return
    be = os.environ.get("ANOVOS_AMD_DIST_BACKEND") or ("nccl" if torch.cuda.is_available() else "gloo")
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29517")
    if torch.cuda.is_available():
        # set the device BEFORE init so RCCL binds each rank's communicator
        # to its own GPU (ranks may oversubscribe one device under gloo)
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")) % max(torch.cuda.device_count(), 1))
    td.init_process_group(backend=be, timeout=datetime.timedelta(seconds=timeout_s))
    _initialized = True


def init_single_rank(backend_name: str) -> None:
    """Explicit world_size=1 init (used by the GPU self-communicator RCCL
    test: every collective wrapper executes the real nccl/RCCL code path
    with a 1-rank communicator)."""
    if is_dist():
        return
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29531")
    if backend_name == "nccl":
        torch.cuda.set_device(0)
    td.init_process_group(backend=backend_name, rank=0, world_size=1,
                          timeout=datetime.timedelta(seconds=120))


def rank() -> int:
    return td.get_rank() if is_dist() else 0


def world_size() -> int:
    return td.get_world_size() if is_dist() else 1


def barrier() -> None:
    if is_dist():
        if backend() == "nccl":
            td.barrier(device_ids=[torch.cuda.current_device()])
        else:
            td.barrier()


def _coll_device() -> torch.device:
    """Device collectives must run on: RCCL needs device tensors; gloo
    gathers need host tensors."""
    if backend() == "nccl":
        return torch.device("cuda", torch.cuda.current_device())
    return torch.device("cpu")


def _stage(t: torch.Tensor) -> torch.Tensor:
    """Move a tensor to the collective device (no-op when already there)."""
    dev = _coll_device()
    return t if t.device == dev else t.to(dev)


def all_reduce_scalar(x, op: str = "sum"):
    """All-reduce a python scalar; returns python number. Device-resident
    under RCCL (fill_ with an immediate — no H2D copy), host under gloo."""
    if not is_dist():
        return x
    buf = _scalar_buf()
    buf.fill_(float(x))
    td.all_reduce(buf, op=_op(op))
    v = buf.item()
    return int(v) if isinstance(x, int) and op in ("sum", "min", "max") else v


_SCALAR_BUFS = {}


def _scalar_buf() -> torch.Tensor:
    dev = _coll_device()
    buf = _SCALAR_BUFS.get(dev)
    if buf is None:
        buf = torch.zeros(1, dtype=torch.float64, device=dev)
        _SCALAR_BUFS[dev] = buf
    return buf


def all_reduce_scalars(xs: Sequence[float], op: str = "sum") -> List[float]:
    """Batched scalar all-reduce: ONE collective for a list of values
    (replaces per-column all_reduce_scalar loops on CPU fallback paths)."""
    if not is_dist():
        return list(xs)
    t = torch.tensor(list(xs), dtype=torch.float64, device=_coll_device())
    td.all_reduce(t, op=_op(op))
    return t.tolist()


def all_reduce_(t: torch.Tensor, op: str = "sum") -> torch.Tensor:
    """In-place all-reduce of a tensor (fused stat vectors, histograms,
    HLL registers with op='max', min/max vectors). Under RCCL a host
    tensor is staged to device and copied back; gloo reduces CUDA
    tensors natively (supported op), so device tensors stay put."""
    if not is_dist():
        return t
    if backend() == "nccl" and not t.is_cuda:
        d = t.cuda()
        td.all_reduce(d, op=_op(op))
        t.copy_(d)
        return t
    td.all_reduce(t, op=_op(op))
    return t


def all_gather_tensor(t: torch.Tensor) -> List[torch.Tensor]:
    """All-gather a variable-length tensor (flattened): a fixed-size
    length exchange + one padded all-gather. Device-resident under RCCL —
    the tensorized replacement for all_gather_object on hot merge paths
    (value-count merges, exact-quantile gathers, row-hash dedup).
    Returns per-rank tensors on the caller's original device."""
    if not is_dist():
        return [t]
    orig_dev = t.device
    orig_shape = None
    if t.dim() != 1:
        orig_shape = tuple(t.shape[1:])
        t = t.reshape(-1)
    t = _stage(t.contiguous())
    ws = world_size()
    dev = t.device
    n = torch.tensor([t.numel()], dtype=torch.int64, device=dev)
    lens = [torch.zeros(1, dtype=torch.int64, device=dev) for _ in range(ws)]
    td.all_gather(lens, n)
    lens = [int(l.item()) for l in lens]
    m = max(lens)
    if m == 0:
        outs = [torch.empty(0, dtype=t.dtype, device=orig_dev) for _ in range(ws)]
    else:
        pad = torch.zeros(m, dtype=t.dtype, device=dev)
        pad[: t.numel()] = t
        bufs = [torch.empty(m, dtype=t.dtype, device=dev) for _ in range(ws)]
        td.all_gather(bufs, pad)
        outs = [b[:l].to(orig_dev) for b, l in zip(bufs, lens)]
    if orig_shape is not None:
        inner = 1
        for s in orig_shape:
            inner *= s
        outs = [o.reshape(-1, *orig_shape) if inner else o for o in outs]
    return outs


def all_to_all_tensor(send_parts: List[torch.Tensor]) -> List[torch.Tensor]:
    """Variable-length all-to-all: send_parts[q] goes to rank q; returns
    the parts received from every rank. RCCL uses the native collective
    (device-resident, one length exchange + one all_to_all); gloo (CPU
    tests) emulates via an object all-gather. Used by hash-partitioned
    merges (exact dedup verdicts, exact continuous modes) where each
    value must cross the fabric exactly ONCE instead of being gathered
    to every rank."""
    if not is_dist():
        return [p for p in send_parts]
    ws = world_size()
    r = rank()
    if backend() == "nccl":
        dev = torch.device("cuda", torch.cuda.current_device())
        parts = [p.to(dev).contiguous() for p in send_parts]
        counts = torch.tensor([p.numel() for p in parts], dtype=torch.int64, device=dev)
        mat = [torch.zeros_like(counts) for _ in range(ws)]
        td.all_gather(mat, counts)
        recv = [torch.empty(int(mat[q][r]), dtype=parts[0].dtype, device=dev) for q in range(ws)]
        td.all_to_all(recv, parts)
        return recv
    import numpy as _np

    gathered = all_gather_object([p.cpu().numpy() for p in send_parts])
    return [torch.from_numpy(_np.ascontiguousarray(gathered[q][r])) for q in range(ws)]


def all_gather_object(obj: Any) -> List[Any]:
    """All-gather arbitrary host objects — COLD paths only (string
    dictionary merges, tiny model artifacts). Hot paths use
    all_gather_tensor. Works under nccl (torch pickles via device byte
    tensors) because init set the device."""
    if not is_dist():
        return [obj]
    out = [None] * world_size()
    td.all_gather_object(out, obj)
    return out


def broadcast_(t: torch.Tensor, src: int = 0) -> torch.Tensor:
    """In-place tensor broadcast (model params, cutoffs) on the
    collective device."""
    if not is_dist():
        return t
    if backend() == "nccl" and not t.is_cuda:
        d = t.cuda()
        td.broadcast(d, src=src)
        t.copy_(d)
        return t
    td.broadcast(t, src=src)
    return t


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
