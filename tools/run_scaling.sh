#!/usr/bin/env bash
# Multi-GPU scaling runbook: B4-299 bf16 bench at 1/2/4/8 GPUs on ONE node,
# RCCL over xGMI, one rank per GPU (the driver runs
# the same shape at round end -> SCALE_rNN.json).
#
#   bash tools/run_scaling.sh [STEPS] [WARMUP]
#
# Env pinning for the 7-link xGMI p2p fabric (tune with sweep_buckets.sh):
#   DFD_AMD_BUCKET_MB   DDP bucket size (default 60)
#   NCCL_MIN_NCHANNELS  RCCL channel floor — xGMI rings are per-link bound;
#                       more channels spread a ring across links (sweep 4..32)
set -euo pipefail
cd "$(dirname "$0")/.."
STEPS=${1:-20}
WARMUP=${2:-8}
export HSA_ENABLE_IPC_MODE_LEGACY=0
export NCCL_MIN_NCHANNELS=${NCCL_MIN_NCHANNELS:-16}
for N in 1 2 4 8; do
  AVAIL=$(python -c "import torch; print(torch.cuda.device_count())")
  if [ "$N" -gt "$AVAIL" ]; then echo "skip N=$N (only $AVAIL GPUs)"; continue; fi
  echo "== N=$N =="
  if [ "$N" = 1 ]; then
    python bench.py --gpus 1 --steps "$STEPS" --warmup "$WARMUP"
  else
    python -m torch.distributed.run --standalone --local-addr 127.0.0.1 \
      --nproc-per-node "$N" bench.py --gpus "$N" --steps "$STEPS" --warmup "$WARMUP"
  fi
done
