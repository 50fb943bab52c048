#!/usr/bin/env python3
"""Per-shape microbenchmark of the gfx950 HIP kernels on the EfficientNet-B4
299px batch-192 layer shapes (the bench.py hot path). Prints one line per
(op, shape): HIP time, torch-eager reference time, effective HBM bandwidth.

Run on a GPU box:  python tools/bench_kernels.py [--batch 192] [--ops dw,bn,se]
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

# (C, H_in, kernel, stride, count) for every depthwise conv in B4-299 —
# exact inventory from `python tools/model_shapes.py --model efficientnet_b4
# --img-size 299 --kind dw`.
DW_SHAPES = [
    (24, 150, 3, 1, 1),
    (48, 150, 3, 1, 1),
    (144, 150, 3, 2, 1),
    (192, 75, 3, 1, 3),
    (192, 75, 5, 2, 1),
    (336, 38, 3, 2, 1),
    (336, 38, 5, 1, 3),
    (672, 19, 3, 1, 5),
    (672, 19, 5, 1, 1),
    (960, 19, 5, 1, 5),
    (960, 19, 5, 2, 1),
    (1632, 10, 3, 1, 1),
    (1632, 10, 5, 1, 7),
    (2688, 10, 3, 1, 1),
]

# BN shapes: (C, H, count) — pw-expand outs, dw outs, pwl outs (approx)
BN_SHAPES = [
    (48, 150, 2),
    (144, 150, 1),
    (24, 150, 1),
    (192, 75, 4),
    (32, 75, 4),
    (288, 38, 4),
    (48, 38, 4),
    (672, 19, 6),
    (112, 19, 6),
    (960, 19, 6),
    (160, 19, 6),
    (1632, 10, 8),
    (272, 10, 8),
    (2688, 10, 2),
    (448, 10, 2),
]

SE_SHAPES = [(c, h, n) for (c, h, k, s, n) in DW_SHAPES]


def timeit(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters  # ms


def cl(x):
    return x.contiguous(memory_format=torch.channels_last)


def bench_dw(B, dtype):
    from deepfake_detection_amd.ops.extension import load_extension
    ext = load_extension()
    print(f"== depthwise conv (B={B}, {dtype}) | HIP ms | torch ms | GB/s eff (HIP)")
    tot_hip = tot_ref = 0.0
    for C, H, k, s, count in DW_SHAPES:
        pad = (k - 1) // 2
        Ho = (H + 2 * pad - k) // s + 1
        x = cl(torch.randn(B, C, H, W := H, device="cuda", dtype=dtype))
        w = torch.randn(C, 1, k, k, device="cuda", dtype=dtype)
        wp = w.reshape(C, k, k).permute(1, 2, 0).contiguous()
        dy = cl(torch.randn(B, C, Ho, Ho, device="cuda", dtype=dtype))
        bytes_x = x.numel() * x.element_size()
        bytes_y = dy.numel() * dy.element_size()

        t_f = timeit(lambda: ext.dw_conv2d_fwd(x, wp, s, s, pad, pad))
        t_bd = timeit(lambda: ext.dw_conv2d_bwd_data(dy, wp, H, W, s, s, pad, pad))
        t_bw = timeit(lambda: ext.dw_conv2d_bwd_weight(dy, x, k, s, s, pad, pad))

        tr_f = timeit(lambda: torch.nn.functional.conv2d(x, w, None, s, pad, 1, C))
        gbs_f = (bytes_x + bytes_y) / t_f / 1e6
        gbs_bw = (bytes_x + bytes_y) / t_bw / 1e6
        tot = count * (t_f + t_bd + t_bw)
        tot_hip += tot
        tot_ref += count * tr_f
        print(f"dw C={C:4d} H={H:3d} k{k} s{s} x{count}: fwd {t_f:7.3f} bwd_d {t_bd:7.3f} "
              f"bwd_w {t_bw:7.3f} ms | torch fwd {tr_f:7.3f} | f {gbs_f:6.0f} bw {gbs_bw:6.0f} GB/s")
    print(f"dw total per step (fwd+bwd, weighted): {tot_hip:.2f} ms  (torch fwd only: {tot_ref:.2f})")


def bench_bn(B, dtype):
    from deepfake_detection_amd.ops.bn_act import fused_bn_act
    print(f"== BN+SiLU (B={B}, {dtype}) | fwd/bwd HIP ms | GB/s")
    tot = 0.0
    for C, H, count in BN_SHAPES:
        x = cl(torch.randn(B, C, H, H, device="cuda", dtype=dtype))
        bn = torch.nn.BatchNorm2d(C, momentum=0.01, eps=1e-3).cuda().float()
        nbytes = x.numel() * x.element_size()

        def fwd():
            return fused_bn_act(x, bn.weight, bn.bias, bn.running_mean,
                                bn.running_var, True, 0.01, 1e-3, "silu")

        t_f = timeit(fwd)
        xg = x.detach().requires_grad_(True)
        y = fused_bn_act(xg, bn.weight, bn.bias, bn.running_mean, bn.running_var,
                         True, 0.01, 1e-3, "silu")
        g = torch.ones_like(y)

        def bwd():
            xg.grad = None
            y.backward(g, retain_graph=True)

        t_b = timeit(bwd, iters=5)
        tot += count * (t_f + t_b)
        print(f"bn C={C:4d} H={H:3d} x{count}: fwd {t_f:7.3f} bwd {t_b:7.3f} ms | "
              f"fwd {2*nbytes/t_f/1e6:6.0f} bwd {3*nbytes/t_b/1e6:6.0f} GB/s")
    print(f"bn total per step: {tot:.2f} ms")


def bench_se(B, dtype):
    from deepfake_detection_amd.ops.se import fused_se
    print(f"== SE chain (B={B}, {dtype})")
    tot = 0.0
    for C, H, count in SE_SHAPES:
        Cr = max(1, C // 24)  # se_ratio 0.25 of block in-chs approx
        x = cl(torch.randn(B, C, H, H, device="cuda", dtype=dtype))
        w1 = torch.randn(Cr, C, 1, 1, device="cuda", dtype=torch.float32)
        b1 = torch.randn(Cr, device="cuda")
        w2 = torch.randn(C, Cr, 1, 1, device="cuda", dtype=torch.float32)
        b2 = torch.randn(C, device="cuda")
        t_f = timeit(lambda: fused_se(x, w1, b1, w2, b2, "silu"))
        xg = x.detach().requires_grad_(True)
        y = fused_se(xg, w1, b1, w2, b2, "silu")
        g = torch.ones_like(y)

        def bwd():
            xg.grad = None
            y.backward(g, retain_graph=True)

        t_b = timeit(bwd, iters=5)
        tot += count * (t_f + t_b)
        nbytes = x.numel() * x.element_size()
        print(f"se C={C:4d} H={H:3d} x{count}: fwd {t_f:7.3f} bwd {t_b:7.3f} ms | "
              f"fwd {2*nbytes/t_f/1e6:6.0f} GB/s")
    print(f"se total per step: {tot:.2f} ms")


def bench_pw(B=192):
    """A/B the experimental MFMA 1x1 GEMM vs MIOpen conv and rocBLAS matmul
    on the B4-299 pointwise shapes."""
    from deepfake_detection_amd.ops.pwconv import pw_conv2d_fwd
    shapes = [  # (Cin, Cout, H)
        (48, 24, 150), (24, 144, 150), (32, 192, 75), (144, 32, 75), (192, 32, 75),
        (192, 288, 38), (288, 48, 38), (672, 112, 19), (960, 160, 19),
        (960, 272, 10), (1632, 272, 10), (1632, 448, 10), (448, 2688, 10),
        (2688, 448, 10),
    ]
    print(f"== pointwise conv fwd (B={B}, bf16) | mfma ms | miopen ms | matmul ms")
    for k, n, h in shapes:
        x = cl(torch.randn(B, k, h, h, device="cuda", dtype=torch.bfloat16))
        w = torch.randn(n, k, 1, 1, device="cuda", dtype=torch.bfloat16)
        t_mfma = timeit(lambda: pw_conv2d_fwd(x, w))
        t_mi = timeit(lambda: torch.nn.functional.conv2d(x, w))
        x2 = x.permute(0, 2, 3, 1).reshape(-1, k)
        w2 = w.reshape(n, k)
        t_mm = timeit(lambda: x2 @ w2.t())
        gb = (x.numel() + B * n * h * h) * 2 / 1e9
        tf = 2 * x2.shape[0] * k * n / 1e12
        print(f"pw K={k:4d} N={n:4d} H={h:3d}: mfma {t_mfma:7.3f} miopen {t_mi:7.3f} "
              f"matmul {t_mm:7.3f} ms | mfma {gb/t_mfma*1000:5.0f} GB/s {tf/t_mfma*1000:6.1f} TF")

    from deepfake_detection_amd.ops.extension import load_extension
    ext = load_extension()
    print(f"== pointwise conv bwd-weight (B={B}, bf16) | mfma ms | matmul ms")
    for k, n, h in shapes:
        x = cl(torch.randn(B, k, h, h, device="cuda", dtype=torch.bfloat16))
        dy = cl(torch.randn(B, n, h, h, device="cuda", dtype=torch.bfloat16))
        t_mfma = timeit(lambda: ext.pw_conv2d_bwd_weight_mfma(dy, x))
        dy2 = dy.permute(0, 2, 3, 1).reshape(-1, n)
        x2 = x.permute(0, 2, 3, 1).reshape(-1, k)
        t_mm = timeit(lambda: dy2.t() @ x2)
        gb = (x.numel() + dy.numel()) * 2 / 1e9
        print(f"wg K={k:4d} N={n:4d} H={h:3d}: mfma {t_mfma:7.3f} matmul {t_mm:7.3f} ms "
              f"| mfma {gb/t_mfma*1000:5.0f} GB/s")


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--batch", type=int, default=192)
    p.add_argument("--ops", default="dw,bn,se")  # add "pw" for the MFMA 1x1 A/B
    p.add_argument("--dtype", default="bf16")
    args = p.parse_args()
    dtype = {"bf16": torch.bfloat16, "fp16": torch.float16, "fp32": torch.float32}[args.dtype]
    assert torch.cuda.is_available()
    ops = args.ops.split(",")
    if "dw" in ops:
        bench_dw(args.batch, dtype)
    if "bn" in ops:
        bench_bn(args.batch, dtype)
    if "se" in ops:
        bench_se(args.batch, dtype)
    if "pw" in ops:
        bench_pw(args.batch)


if __name__ == "__main__":
    main()
