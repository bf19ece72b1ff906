"""Fine-grained timing of the nullRows path (the 72% hotspot)."""

import sys
import time

import torch

sys.path.insert(0, ".")
from bench import make_synthetic_frame
from anovos_amd.ops import rowops, backend
from anovos_amd.shared.context import init_context
from anovos_amd.shared.utils import attributeType_segregation


def t(name, fn, reps=3):
    fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    print(f"{name}: {(time.perf_counter()-t0)/reps*1000:.2f} ms")


ctx = init_context()
idf = make_synthetic_frame(10_000_000, ctx.device, seed=1)
num_cols, cat_cols, _ = attributeType_segregation(idf)
cols = num_cols + cat_cols
ext = backend.hip_ext()

num = [idf.col(c).data.contiguous() for c in cols if idf.col(c).kind == "numerical"]
cat = [idf.col(c).data.contiguous() for c in cols if idf.col(c).kind != "numerical"]
n = idf.local_rows()

out = torch.zeros(n, dtype=torch.int32, device=ctx.device)
t("hip row_null_kernel (150 num cols)", lambda: ext.row_null_counts_num(num, out))


def cat_part():
    o = torch.zeros(n, dtype=torch.int32, device=ctx.device)
    for tt in cat:
        o += (tt == -1).to(torch.int32)
    return o


t("cat part (51 cols torch)", cat_part)
t("row_null_counts full", lambda: rowops.row_null_counts(idf, cols))

counts = rowops.row_null_counts(idf, cols)
flagged = counts > (len(cols) * 0.8)


def hist_part():
    hist = torch.zeros((len(cols) + 1) * 2, dtype=torch.float64, device=counts.device)
    key = counts.to(torch.long) * 2 + flagged.to(torch.long)
    hist.scatter_add_(0, key, torch.ones_like(key, dtype=torch.float64))
    return hist


t("scatter_add hist", hist_part)

from anovos_amd.data_analyzer import quality_checker as qc

t("nullRows_detection end-to-end", lambda: qc.nullRows_detection(ctx, idf, treatment=False), reps=1)
