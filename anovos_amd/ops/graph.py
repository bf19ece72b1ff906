"""hipGraph capture for repeated transform chains (serving-style apply).

A fitted transform chain (binning cutoffs, scaler params, imputation
fills, encoders) applied to a stream of fixed-shape batches is a
launch-bound inner loop: tens of small kernels per batch with identical
shapes. ``CapturedTransform`` captures the whole chain into ONE hipGraph
(torch.cuda.CUDAGraph is hipGraph on ROCm) and replays it per batch —
one graph launch instead of N kernel launches + python dispatch.

Use: fit transformers normally, wrap the pure-tensor apply in a function
of the static input block, capture once, then feed batches::

    cap = CapturedTransform(apply_fn, example_cols)   # capture
    outs = cap(batch_cols)                            # replay (any batch)

Constraints (hipGraph semantics): fixed shapes/dtypes, no host syncs or
allocations inside apply_fn (pure device kernels). CPU fallback runs
apply_fn eagerly, so code paths stay identical off-GPU.
"""

from __future__ import annotations

from typing import Callable, List

import torch


class CapturedTransform:
    """Capture fn(List[Tensor]) -> List[Tensor] into a replayable graph."""

    def __init__(self, fn: Callable[[List[torch.Tensor]], List[torch.Tensor]], example: List[torch.Tensor], warmups: int = 2):
        self.fn = fn
        self.on_gpu = bool(example) and example[0].is_cuda
        if not self.on_gpu:
            self._static_in = None
            return
        self._static_in = [t.clone() for t in example]
        # warm up on a side stream so capture sees steady-state allocs
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(warmups):
                out = fn(self._static_in)
        torch.cuda.current_stream().wait_stream(s)
        self._graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self._graph):
            self._static_out = fn(self._static_in)

    def __call__(self, cols: List[torch.Tensor]) -> List[torch.Tensor]:
        if not self.on_gpu:
            return self.fn(cols)
        for dst, src in zip(self._static_in, cols):
            if dst.shape != src.shape or dst.dtype != src.dtype:
                raise ValueError("CapturedTransform requires fixed shapes/dtypes; re-capture for new geometry")
            dst.copy_(src, non_blocking=True)
        self._graph.replay()
        return [t.clone() for t in self._static_out]
