#!/usr/bin/env python3
"""Isolated Gram-kernel profile target: 150 f32 columns, pearson_matrix
3x (for rocprofv3 --pmc counter runs)."""

import sys
import time

import torch

sys.path.insert(0, ".")
from anovos_amd.core.frame import AnovosFrame, Column

rows = int(sys.argv[1]) if len(sys.argv) > 1 else 20_000_000
k = int(sys.argv[2]) if len(sys.argv) > 2 else 150
g = torch.Generator(device="cuda").manual_seed(3)
cols = {}
for i in range(k):
    x = torch.randn(rows, generator=g, device="cuda") * (1 + i % 5) + i
    cols[f"c{i}"] = Column(f"c{i}", "float", x)
idf = AnovosFrame(cols, "cuda")

from anovos_amd.ops import corr

m = corr.pearson_matrix(idf, list(cols))  # warmup (includes moments pass)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(3):
    m = corr.pearson_matrix(idf, list(cols))
torch.cuda.synchronize()
t1 = time.perf_counter()
gb = rows * k * 4 / 1e9
print(f"pearson {k}x{rows}: {(t1 - t0) / 3 * 1000:.2f} ms/call, {gb:.1f} GB/call, "
      f"{gb / ((t1 - t0) / 3) / 1000:.2f} TB/s effective")
