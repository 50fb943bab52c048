#!/usr/bin/env bash
# DDP bucket-size + RCCL channel sweep on an N-GPU node (default 8): finds
# the xGMI-optimal DFD_AMD_BUCKET_MB / NCCL_MIN_NCHANNELS for the B4-299
# gradient all-reduce (~250 MB fp32 grads per step over 7 p2p links).
#
#   bash tools/sweep_buckets.sh [NPROC]
set -euo pipefail
cd "$(dirname "$0")/.."
NPROC=${1:-8}
export HSA_ENABLE_IPC_MODE_LEGACY=0
for MB in 25 60 100 250; do
  for CH in 8 16 32; do
    echo "== bucket=${MB}MB channels=${CH} =="
    DFD_AMD_BUCKET_MB=$MB NCCL_MIN_NCHANNELS=$CH \
      python -m torch.distributed.run --standalone --local-addr 127.0.0.1 \
      --nproc-per-node "$NPROC" bench.py --gpus "$NPROC" --steps 10 --warmup 4 \
      | grep '"value"' || true
  done
done
