"""Micro-ablation of the slowest bench sections (run via gpurun)."""

import json
import sys
import time

import torch

sys.path.insert(0, ".")

from bench import make_synthetic_frame
from anovos_amd.data_analyzer import stats_generator as sg
from anovos_amd.data_transformer import transformers as T
from anovos_amd.ops import elementwise, groupby as groupby_ops, histogram as hist_ops, stats as stats_ops
from anovos_amd.ops import bucketize as bucketize_ops
from anovos_amd.shared.context import init_context
from anovos_amd.shared.utils import attributeType_segregation

ctx = init_context()
idf = make_synthetic_frame(10_000_000, ctx.device, seed=99)
num_cols, cat_cols, _ = attributeType_segregation(idf)
times = {}


def t(name, fn, reps=3, clear=True):
    fn()
    ctx.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        if clear:
            idf.clear_stats_cache()
        fn()
    ctx.synchronize()
    times[name] = (time.perf_counter() - t0) / reps * 1000


moments = stats_ops.frame_moments(idf, num_cols)
tensors = [idf.col(c).data for c in num_cols]
a = [moments[c].mean for c in num_cols]
b = [1.0 / max(moments[c].stddev, 1e-9) for c in num_cols]

t("moments", lambda: stats_ops.frame_moments(idf, num_cols))
t("scale_columns_ext_only", lambda: elementwise.scale_columns(tensors, a, b), clear=False)
t("z_standardization_full", lambda: T.z_standardization(ctx, idf, num_cols))
t("fill_nan_ext_only", lambda: elementwise.fill_nan_columns(tensors, [0.0] * len(tensors)), clear=False)
t("quantiles_p50", lambda: hist_ops.approx_quantiles(idf, num_cols, [0.5]))
t("imputation_median_full", lambda: T.imputation_MMM(ctx, idf, method_type="median"))
t("cat_value_counts_50", lambda: groupby_ops.cat_value_counts(idf, cat_cols))
t("hll_ext", lambda: sg.uniqueCount_computation(ctx, idf, num_cols + cat_cols, compute_approx_unique_count=True))

# binning internals
cuts = hist_ops.approx_quantiles(idf, num_cols, [j / 10 for j in range(1, 10)])
cut_t = [torch.tensor(cuts[c], dtype=torch.float64) for c in num_cols]
t("bucketize_float_ext", lambda: bucketize_ops.bucketize_columns_float(tensors, cut_t), clear=False)
t("binning_full", lambda: T.attribute_binning(ctx, idf, num_cols, bin_size=10, output_mode="append"))

times["rows"] = 10_000_000
print(json.dumps(times, indent=1))
