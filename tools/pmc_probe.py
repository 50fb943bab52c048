#!/usr/bin/env python3
"""Tiny targeted kernel driver for PMC counter capture.

Runs a handful of invocations of ONE kernel family on its production shape
so a rocprofv3 --pmc pass stays seconds long and the counter CSV stays
small. Usage (on a GPU box):

  rocprofv3 --pmc TCC_EA0_RDREQ_sum TCC_EA0_WRREQ_sum -d out -o hbm -- \
      python tools/pmc_probe.py dw_fwd_k5
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def cl(x):
    return x.contiguous(memory_format=torch.channels_last)


def dw_fwd(k, c, h, b=384, iters=5):
    from deepfake_detection_amd.ops.dwconv import dw_conv2d

    x = cl(torch.randn(b, c, h, h, device="cuda", dtype=torch.bfloat16))
    w = torch.randn(c, 1, k, k, device="cuda", dtype=torch.bfloat16)
    for _ in range(iters):
        y = dw_conv2d(x, w, None, (1, 1), (k // 2, k // 2), (1, 1))
    torch.cuda.synchronize()
    print("dw_fwd", k, c, h, tuple(y.shape))


def bn_bwd(c, h, b=384, iters=5):
    from deepfake_detection_amd.ops.bn_act import fused_bn_act

    x = cl(torch.randn(b, c, h, h, device="cuda", dtype=torch.bfloat16)).requires_grad_(True)
    bn = torch.nn.BatchNorm2d(c, momentum=0.01, eps=1e-3).cuda()
    for _ in range(iters):
        y = fused_bn_act(x, bn.weight, bn.bias, bn.running_mean, bn.running_var,
                         True, 0.01, 1e-3, "silu")
        y.backward(y.detach())
        x.grad = None
    torch.cuda.synchronize()
    print("bn_bwd", c, h)


def pw_fwd(k, n, h, b=384, iters=5):
    from deepfake_detection_amd.ops.pwconv import pw_conv2d_fwd

    x = cl(torch.randn(b, k, h, h, device="cuda", dtype=torch.bfloat16))
    w = torch.randn(n, k, 1, 1, device="cuda", dtype=torch.bfloat16)
    for _ in range(iters):
        pw_conv2d_fwd(x, w)
    torch.cuda.synchronize()
    print("pw_fwd", k, n, h)


def pw_wgrad(k, n, h, b=384, iters=5):
    from deepfake_detection_amd.ops.extension import load_extension

    ext = load_extension()
    x = cl(torch.randn(b, k, h, h, device="cuda", dtype=torch.bfloat16))
    dy = cl(torch.randn(b, n, h, h, device="cuda", dtype=torch.bfloat16))
    for _ in range(iters):
        ext.pw_conv2d_bwd_weight_mfma(dy, x)
    torch.cuda.synchronize()
    print("pw_wgrad", k, n, h)


def stem_fwd(cin, n, img, b=384, iters=5):
    from deepfake_detection_amd.ops.stemconv import stem_conv2d

    x = cl(torch.randn(b, cin, img, img, device="cuda", dtype=torch.bfloat16))
    w = torch.randn(n, cin, 3, 3, device="cuda", dtype=torch.bfloat16)
    for _ in range(iters):
        stem_conv2d(x, w, stride=2, padding=1)
    torch.cuda.synchronize()
    print("stem_fwd", cin, n, img)


CASES = {
    "dw_fwd_k5": lambda: dw_fwd(5, 336, 38),
    "dw_fwd_k3": lambda: dw_fwd(3, 192, 75),
    "bn_bwd_144": lambda: bn_bwd(144, 150),
    "bn_bwd_672": lambda: bn_bwd(672, 19),
    "pw_fwd_expand": lambda: pw_fwd(192, 288, 38),
    "pw_fwd_lowk": lambda: pw_fwd(24, 144, 150),
    "pw_wgrad_tall": lambda: pw_wgrad(24, 144, 150),
    "stem_fwd_b4": lambda: stem_fwd(3, 48, 299),
}

if __name__ == "__main__":
    assert torch.cuda.is_available()
    for case in sys.argv[1:] or ["dw_fwd_k5"]:
        CASES[case]()
