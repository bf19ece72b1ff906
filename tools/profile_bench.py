"""Per-section timing of the bench pipeline on the GPU — run via gpurun.
Prints a JSON dict of section -> ms so optimization targets the real
hotspots (cdna_hip_programming.md rule: ablate before optimizing)."""

import json
import sys
import time

import torch

sys.path.insert(0, ".")

from bench import N_NUM_CONT, make_synthetic_frame
from anovos_amd.core import dist
from anovos_amd.data_analyzer import quality_checker as qc
from anovos_amd.data_analyzer import stats_generator as sg
from anovos_amd.data_transformer import transformers as T
from anovos_amd.drift_stability import drift_detector as dd
from anovos_amd.ops import histogram as hist_ops
from anovos_amd.ops import stats as stats_ops
from anovos_amd.shared.context import init_context
from anovos_amd.shared.utils import attributeType_segregation


def main(rows=10_000_000):
    ctx = init_context()
    idf = make_synthetic_frame(rows, ctx.device, seed=99)
    num_cols, cat_cols, _ = attributeType_segregation(idf)
    int_cols = [c for c in idf.columns if c.startswith("int_")]

    times = {}

    def bench_section(name, fn, reps=2):
        fn()  # warmup
        ctx.synchronize()
        t0 = time.perf_counter()
        for _ in range(reps):
            idf.clear_stats_cache()
            fn()
        ctx.synchronize()
        times[name] = (time.perf_counter() - t0) / reps * 1000

    # prime drift source
    binned0 = T.attribute_binning(ctx, idf, num_cols, bin_size=10, output_mode="append")
    total = idf.count()
    source_hist = {}
    for c in num_cols:
        keys, vals = dd._bin_frequencies(binned0, c + "_binned", total)
        source_hist[c] = dict(zip(keys, vals))

    bench_section("moments_only", lambda: stats_ops.frame_moments(idf, num_cols))
    bench_section(
        "quantiles_9p",
        lambda: hist_ops.approx_quantiles(idf, num_cols, [0.01, 0.05, 0.1, 0.25, 0.5, 0.75, 0.9, 0.95, 0.99]),
    )
    bench_section("measures_of_counts", lambda: sg.measures_of_counts(ctx, idf))
    bench_section("central_tendency_discrete", lambda: sg.measures_of_centralTendency(ctx, idf, int_cols + cat_cols))
    bench_section("cardinality_hll", lambda: sg.measures_of_cardinality(ctx, idf, num_cols + cat_cols, use_approx_unique_count=True))
    bench_section("shape", lambda: sg.measures_of_shape(ctx, idf))
    bench_section("null_rows", lambda: qc.nullRows_detection(ctx, idf, treatment=False))
    bench_section("biasedness", lambda: qc.biasedness_detection(ctx, idf, int_cols + cat_cols, treatment=False, treatment_threshold=0.9))
    bench_section(
        "outlier_detection",
        lambda: qc.outlier_detection(ctx, idf, num_cols, detection_side="both", treatment=True, treatment_method="value_replacement"),
    )
    bench_section("binning_append", lambda: T.attribute_binning(ctx, idf, num_cols, bin_size=10, output_mode="append"))

    def drift_fn():
        binned = T.attribute_binning(ctx, idf, num_cols, bin_size=10, output_mode="append")
        cols50 = num_cols[:: max(1, len(num_cols) // 50)]
        dd.batched_bin_frequencies(binned, [c + "_binned" for c in cols50], total, max_bin=10)

    bench_section("drift_50cols", drift_fn)
    bench_section("z_standardization", lambda: T.z_standardization(ctx, idf, num_cols[:N_NUM_CONT]))
    bench_section("imputation_median", lambda: T.imputation_MMM(ctx, idf, method_type="median"))
    bench_section("cat_label_encoding", lambda: T.cat_to_num_unsupervised(ctx, idf, cat_cols[:25], method_type="label_encoding"))
    bench_section("outlier_categories", lambda: T.outlier_categories(ctx, idf, cat_cols[25:], max_category=20))

    times["TOTAL"] = sum(v for k, v in times.items())
    print(json.dumps(times, indent=1))


if __name__ == "__main__":
    rows = int(sys.argv[1]) if len(sys.argv) > 1 else 10_000_000
    main(rows)
