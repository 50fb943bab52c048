#!/usr/bin/env bash
# Inference smoke run (parity with reference scripts/test.sh): scores the
# bundled sample images with the fp16 checkpoint.
set -e
cd "$(dirname "$0")/.."
CKPT=${CKPT:-models/model_half.pth.tar}
python -m deepfake_detection_amd.runners.test --checkpoint "$CKPT" "$@"
