"""anovos_amd — an MI355X-native columnar feature-engineering engine.

A from-scratch rebuild of the capabilities of anovos/anovos (feature
engineering at scale for tabular ML data) with the Spark/JVM substrate
replaced by:

- an Arrow-compatible column store of PyTorch-ROCm tensors held in HBM3E,
  row-partitioned across the GPUs of one node (``anovos_amd.core.frame``),
- hand-written CDNA4 (gfx950) HIP kernels for the hot columnar operators —
  fused multi-column reductions/moments, LDS-staged histograms, quantile
  sketches, bucketize, drift metrics, MFMA correlation GEMM
  (``anovos_amd.ops``),
- RCCL collectives over xGMI for cross-GPU partial-aggregate merges
  (``anovos_amd.core.dist``).

The YAML workflow API, module/function layout, statistic definitions and
report output contract mirror the reference (see SURVEY.md for the
file:line parity map).
"""

from anovos_amd.version import __version__  # noqa: F401
