#!/bin/bash
# SLURM array-job launcher for multi-node BYOL-MI355X training:
# 1 task = 1 process = 1 MI355X (mirrors the reference's slurm/run.sh
# 16-GPU recipe, adapted to ROCm).  Submit with e.g.:
#   sbatch --array=0-15 --gres=gpu:1 --ntasks-per-node=1 deploy/slurm_run.sh
set -euo pipefail

NUM_REPLICAS=${NUM_REPLICAS:-16}
GLOBAL_BATCH=${GLOBAL_BATCH:-1024}
PORT=${PORT:-29300}

# master = host of array task 0
MASTER=$(squeue -j "${SLURM_ARRAY_JOB_ID}" -O minnodelist,arraytaskid -h |
         awk '$2 == 0 {print $1}' | head -1)
RANK=${SLURM_ARRAY_TASK_ID}

export HSA_ENABLE_IPC_MODE_LEGACY=0
export MIOPEN_FIND_MODE=${MIOPEN_FIND_MODE:-HYBRID}

exec python main.py \
  --task multi_augment_image_folder \
  --batch-size "${GLOBAL_BATCH}" \
  --num-replicas "${NUM_REPLICAS}" \
  --distributed-master "${MASTER}" \
  --distributed-rank "${RANK}" \
  --distributed-port "${PORT}" \
  --convert-to-sync-bn --channels-last \
  "$@"
