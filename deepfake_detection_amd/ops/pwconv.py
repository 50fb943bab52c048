"""Experimental: 1x1 (pointwise) conv forward as an MFMA GEMM
(ops/hip/pwconv.hip). Off by default — MIOpen's igemm kernels serve the 1x1
convs in the main path; set DFD_AMD_PW_MFMA=1 to A/B this kernel. The
BASELINE.json north star names this op as one to own natively; enable it
per-shape once it measures faster than MIOpen (tools/bench_kernels.py).
"""

import os

import torch

from .extension import load_extension


def pw_mfma_enabled() -> bool:
    return os.environ.get("DFD_AMD_PW_MFMA", "0") == "1"


def pw_conv2d_fwd(x, weight):
    """bf16 NHWC 1x1 conv forward on matrix cores. No autograd (A/B use)."""
    ext = load_extension()
    x = x.contiguous(memory_format=torch.channels_last)
    return ext.pw_conv2d_fwd_mfma(x, weight)


class _PwConv2d(torch.autograd.Function):
    """1x1 conv with the MFMA GEMM kernel on both data passes.

    bwd-data reuses the SAME (GPU-validated) kernel: dX[M,K] = dY[M,N] @ W
    viewed as a [K,N]-weighted forward (weight transposed once, tiny).
    bwd-weight is a [N,M]x[M,K] reduction over the huge M — that shape fits
    rocBLAS's split-K GEMM better than this block tiling, so it stays on
    torch.matmul.
    """

    @staticmethod
    def forward(ctx, x, weight):
        ext = load_extension()
        x = x.contiguous(memory_format=torch.channels_last)
        y = ext.pw_conv2d_fwd_mfma(x, weight)
        ctx.save_for_backward(x, weight)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = load_extension()
        x, weight = ctx.saved_tensors
        dy = dy.contiguous(memory_format=torch.channels_last)
        N, K = weight.shape[0], weight.shape[1]
        dx = dw = None
        if ctx.needs_input_grad[0]:
            wt = weight.reshape(N, K).t().contiguous().view(K, N, 1, 1)
            dx = ext.pw_conv2d_fwd_mfma(dy, wt)
        if ctx.needs_input_grad[1]:
            m_dy = dy.permute(0, 2, 3, 1).reshape(-1, N)  # [M, N] (view: NHWC)
            m_x = x.permute(0, 2, 3, 1).reshape(-1, K)
            dw = (m_dy.t() @ m_x).view(N, K, 1, 1)
        return dx, dw


def pw_conv2d(x, weight, bias=None):
    """Autograd-enabled 1x1 conv on the MFMA kernel (bf16 NHWC)."""
    if weight.dtype != x.dtype:
        weight = weight.to(x.dtype)
    y = _PwConv2d.apply(x, weight)
    if bias is not None:
        y = y + bias.to(y.dtype).view(1, -1, 1, 1)
    return y
