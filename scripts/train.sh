#!/usr/bin/env bash
# Production training launch — parity with reference scripts/train.sh:3-22,
# adapted to the MI355X stack (torchrun over RCCL, bf16 autocast).
# Single 8-GPU node:
set -e
cd "$(dirname "$0")/.."

DATA=${DATA:-/data/deepfake}
NPROC=${NPROC:-8}

python -m torch.distributed.run --standalone --local-addr 127.0.0.1 \
    --nproc-per-node "$NPROC" \
    -m deepfake_detection_amd.runners.train \
    --data "$DATA" \
    --model efficientnet_deepfake_v4 \
    --model-version v4.0 \
    --input-size-v2 12,600,600 \
    --num-classes 2 \
    --class_names fake,real \
    --label_balance \
    -b 48 \
    --opt rmsproptf --opt-eps .001 \
    --sched step --decay-epochs 2 --decay-rate .92 \
    --basic_lr .0000625 \
    --warmup-epochs 1 --epochs 100 \
    --weight-decay 1e-5 \
    --bn-momentum 0.001 \
    --color-jitter 0.2 --rotate_range 10 --reprob 0.2 --remode pixel \
    --model-ema --model-ema-decay 0.9998 \
    --dist-bn reduce \
    --eval-metric loss \
    -j 8 \
    "$@"
