#!/usr/bin/env python3
"""Exact per-op shape inventory for any registered model.

Hooks every depthwise conv, pointwise (1x1) conv, other conv, and BatchNorm
in a forward pass and prints the deduplicated (shape, count) table — the
ground truth for tools/bench_kernels.py shape lists and kernel-tuning
priorities. NOTE: BatchNorm rows appear only for modules whose own forward
runs; the fused bn_act path consumes BN parameters directly, so BN shapes
there equal the producing conv's output shapes.

Usage: python tools/model_shapes.py --model efficientnet_b4 --img-size 299
       python tools/model_shapes.py --model efficientnet_deepfake_v4 --img-size 600 --in-chans 12
"""

import argparse
import collections
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn as nn


def collect(model, img_size=224, in_chans=3):
    rows = collections.Counter()
    hooks = []

    def on_conv(mod, inp, out):
        x = inp[0]
        k = mod.kernel_size[0]
        if mod.groups == mod.in_channels == mod.out_channels and mod.groups > 1:
            kind = "dw"
            desc = (kind, mod.in_channels, x.shape[-1], k, mod.stride[0])
        elif k == 1 and mod.groups == 1:
            kind = "pw"
            desc = (kind, mod.in_channels, mod.out_channels, x.shape[-1])
        else:
            kind = "conv"
            desc = (kind, mod.in_channels, mod.out_channels, x.shape[-1], k,
                    mod.stride[0], mod.groups)
        rows[desc] += 1

    def on_bn(mod, inp, out):
        x = inp[0]
        rows[("bn", x.shape[1], x.shape[-1])] += 1

    for m in model.modules():
        if isinstance(m, nn.Conv2d):
            hooks.append(m.register_forward_hook(on_conv))
        elif isinstance(m, nn.BatchNorm2d):
            hooks.append(m.register_forward_hook(on_bn))

    model.eval()
    with torch.no_grad():
        model(torch.randn(1, in_chans, img_size, img_size))
    for h in hooks:
        h.remove()
    return rows


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="efficientnet_b4")
    p.add_argument("--img-size", type=int, default=299)
    p.add_argument("--in-chans", type=int, default=3)
    p.add_argument("--kind", default="", help="filter: dw / pw / conv / bn")
    args = p.parse_args()

    import deepfake_detection_amd as dfd

    model = dfd.create_model(args.model, num_classes=2, in_chans=args.in_chans)
    rows = collect(model, args.img_size, args.in_chans)
    total = collections.Counter()
    for desc, count in sorted(rows.items()):
        if args.kind and desc[0] != args.kind:
            continue
        total[desc[0]] += count
        if desc[0] == "dw":
            _, c, h, k, s = desc
            print(f"dw   C={c:5d} H={h:4d} k{k} s{s}  x{count}")
        elif desc[0] == "pw":
            _, cin, cout, h = desc
            print(f"pw   {cin:5d}->{cout:<5d} H={h:4d}      x{count}")
        elif desc[0] == "bn":
            _, c, h = desc
            print(f"bn   C={c:5d} H={h:4d}          x{count}")
        else:
            _, cin, cout, h, k, s, g = desc
            print(f"conv {cin:5d}->{cout:<5d} H={h:4d} k{k} s{s} g{g} x{count}")
    print("totals:", dict(total))


if __name__ == "__main__":
    main()
