#!/usr/bin/env bash
# Run a workflow YAML inside the anovos-amd-worker container (reference
# parity: local/run_workload.sh). MI355X-native: the ROCm devices are
# passed through (--device /dev/kfd /dev/dri + video group); the
# workflow lands on the GPUs automatically when they are visible.
set -euo pipefail

SCRIPT_DIR=$(cd -- "$(dirname -- "${BASH_SOURCE[0]}")" &>/dev/null && pwd)

python "${SCRIPT_DIR}/rewrite_configuration.py" "${1}"

CONFIG_PATH="${PWD}/config.yaml.tmp"
DATA_ROOT=$(cat data_directory.tmp)
OUTPUT_ROOT="${PWD}/output/"
mkdir -p "${OUTPUT_ROOT}"

IMAGE_NAME="${2:-anovos-amd-worker}"
NGPU="${NGPU:-}"

docker run \
  --device=/dev/kfd --device=/dev/dri \
  --security-opt seccomp=unconfined --group-add video \
  ${NGPU:+-e NGPU="${NGPU}"} \
  -e HSA_ENABLE_IPC_MODE_LEGACY=0 \
  --mount type=bind,source="${CONFIG_PATH}",target=/config.yaml \
  --mount type=bind,source="${DATA_ROOT}",target=/data \
  --mount type=bind,source="${OUTPUT_ROOT}",target=/output \
  "${IMAGE_NAME}"
