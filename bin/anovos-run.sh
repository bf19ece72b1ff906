#!/usr/bin/env bash
# Launch an anovos_amd workflow on the MI355X node (the analog of the
# reference's bin/spark-submit.sh, which provisioned 1000 Spark
# executors x 4 cores for the same job). One process per GPU over
# RCCL/xGMI; single-GPU by default.
#
#   bin/anovos-run.sh config/configs.yaml            # 1 GPU (or CPU)
#   NGPU=8 bin/anovos-run.sh config/configs_full.yaml # whole node
#
# Environment:
#   NGPU       number of GPUs (default 1)
#   RUN_TYPE   local | emr | databricks | ak8s (default local)
#   ANOVOS_AMD_INMEMORY_PIPELINE=1  skip per-stage disk materialization
set -euo pipefail
CONFIG="${1:?usage: anovos-run.sh <config.yaml> [run_type]}"
RUN_TYPE="${2:-${RUN_TYPE:-local}}"
NGPU="${NGPU:-1}"
cd "$(dirname "$0")/.."
if [ "$NGPU" -gt 1 ]; then
  exec python -m torch.distributed.run --nnodes=1 --nproc-per-node "$NGPU" \
    --master-addr 127.0.0.1 --master-port "${MASTER_PORT:-29517}" \
    -m anovos_amd "$CONFIG" "$RUN_TYPE"
fi
exec python -m anovos_amd "$CONFIG" "$RUN_TYPE"
