#!/bin/bash
# Containerized launch on a ROCm host (mirrors the reference's
# docker/run.sh: ipc=host, data mounts, device visibility).
set -euo pipefail
IMAGE=${IMAGE:-rocm/pytorch:latest}
DATA_DIR=${DATA_DIR:-$PWD/.datasets}
LOG_DIR=${LOG_DIR:-$PWD/runs}
GPUS=${GPUS:-all}   # or a comma list for HIP_VISIBLE_DEVICES

docker run --rm -it \
  --device=/dev/kfd --device=/dev/dri \
  --security-opt seccomp=unconfined --group-add video \
  --ipc=host --shm-size 64G \
  -e HSA_ENABLE_IPC_MODE_LEGACY=0 \
  -e HIP_VISIBLE_DEVICES="${GPUS}" \
  -v "$PWD":/workspace -v "${DATA_DIR}":/workspace/.datasets \
  -v "${LOG_DIR}":/workspace/runs \
  -w /workspace "${IMAGE}" \
  bash -c "PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace && python main.py $*"
