"""Inference runner (CLI).

Parity: reference dfd/runners/test.py — loads `efficientnet_deepfake_v4`
from models/model_half.pth.tar, fp16 inference on GPU (bf16/fp32 optional),
stacks one image x4 into 12 channels, prints per-image fake score
(scores[:, 0]; class 0 = fake).

Usage: python -m deepfake_detection_amd.runners.test img1.png img2.jpg ...
"""

import argparse
import sys

import numpy as np
import torch
from PIL import Image

from .. import create_deepfake_model_v4
from ..params import DeepFakeModel, preprocess_image


def test_img(img_paths, checkpoint_path="../models/model_half.pth.tar",
             device=None, dtype=None):
    if device is None:
        device = "cuda" if torch.cuda.is_available() else "cpu"
    if dtype is None:
        dtype = torch.float16 if device == "cuda" else torch.float32

    basemodel = create_deepfake_model_v4(
        "efficientnet_deepfake_v4", num_classes=2, in_chans=12,
        checkpoint_path=checkpoint_path, strict=False)
    model = DeepFakeModel(basemodel)
    model = model.to(device=device, dtype=dtype)
    if device == "cuda":
        model = model.to(memory_format=torch.channels_last)
    model.eval()

    results = []
    for img_path in img_paths:
        img = np.asarray(Image.open(img_path).convert("RGB"))
        x = preprocess_image(img, device=device, dtype=dtype)
        if device == "cuda":
            x = x.contiguous(memory_format=torch.channels_last)
        with torch.no_grad():
            scores = model(x)
        fake_score = scores[:, 0].float().item()
        results.append((img_path, fake_score))
        print(f"{img_path}: fake score {fake_score:.4f}")
    return results


def main(argv=None):
    parser = argparse.ArgumentParser(description="Deepfake inference")
    parser.add_argument("images", nargs="+", help="image paths")
    parser.add_argument("--checkpoint", default="../models/model_half.pth.tar")
    parser.add_argument("--device", default=None)
    args = parser.parse_args(argv)
    test_img(args.images, checkpoint_path=args.checkpoint, device=args.device)


if __name__ == "__main__":
    main(sys.argv[1:])
