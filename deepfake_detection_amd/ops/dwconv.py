"""Depthwise conv2d on the hand-written gfx950 NHWC kernels.

Replaces the MIOpen grouped-conv path (reference delegates depthwise convs
to cuDNN via groups=C, dfd/timm/models/layers/create_conv2d.py:24-25). The
MIOpen/CK grouped bwd-weight kernel was 87% of step time on MI355X
(profiles/r01_bench_b4_299_bs192_top_kernels.md) — these kernels own all
three passes.

Weight handling: torch keeps depthwise weight as (C, 1, K, K); the kernels
want (K, K, C) so channel loads vectorize. The pack/unpack permutes touch
C*K*K elements (≤ ~70 KB) — negligible next to the conv itself.
"""

import os

import torch

from .extension import load_extension

_SUPPORTED_K = (3, 5, 7, 9, 11)


def dw_stats_enabled() -> bool:
    """The dw stats-epilogue variant stays OFF by default. Two designs were
    measured (r02): fixed-channel threads with per-use weight loads (the
    compiler hoists the K*K tile -> occupancy loss, 2x slower) and with an
    LDS-staged weight slice (hoist gone, but the channel-blocked spatial
    striding itself loses the wo-tile x locality: k5 C=336 1.89 vs 0.51 ms,
    whole bench 2609 vs 2752 img/s) — both lose more than the bn_stats pass
    they save. Kept correct and GPU-tested behind DFD_AMD_DW_STATS=1."""
    return os.environ.get("DFD_AMD_DW_STATS", "0") == "1"


def dw_supported(weight, stride, padding, dilation, groups) -> bool:
    """True when the HIP depthwise path covers this conv geometry."""
    C, one, kh, kw = weight.shape
    if one != 1 or groups != C:
        return False
    if kh != kw or kh not in _SUPPORTED_K:
        return False
    if isinstance(dilation, (tuple, list)):
        if any(d != 1 for d in dilation):
            return False
    elif dilation != 1:
        return False
    return True


STATS_BUCKETS = 64


class _DwConv2d(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, sh, sw, ph, pw, stats_out):
        ext = load_extension()
        x = x.contiguous(memory_format=torch.channels_last)
        C, _, K, _ = weight.shape
        w_packed = weight.reshape(C, K, K).permute(1, 2, 0).contiguous()
        stats = None
        if (stats_out is not None and K in (3, 5) and sh == 1 and sw == 1
                and dw_stats_enabled()):
            stats = torch.zeros(STATS_BUCKETS, 2, C, device=x.device,
                                dtype=torch.float32)
            stats_out.append(stats)
        y = ext.dw_conv2d_fwd(x, w_packed, sh, sw, ph, pw, stats)
        ctx.save_for_backward(x, w_packed)
        ctx.geom = (sh, sw, ph, pw, K)
        ctx.w_dtype = weight.dtype
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = load_extension()
        x, w_packed = ctx.saved_tensors
        sh, sw, ph, pw, K = ctx.geom
        dy = dy.contiguous(memory_format=torch.channels_last)
        dx = dw = None
        if ctx.needs_input_grad[0]:
            dx = ext.dw_conv2d_bwd_data(dy, w_packed, x.size(2), x.size(3), sh, sw, ph, pw)
        if ctx.needs_input_grad[1]:
            dw_kkc = ext.dw_conv2d_bwd_weight(dy, x, K, sh, sw, ph, pw)  # (K,K,C) fp32
            dw = dw_kkc.permute(2, 0, 1).reshape(-1, 1, K, K).to(ctx.w_dtype).contiguous()
        return dx, dw, None, None, None, None, None


def _pair(v):
    return (v, v) if isinstance(v, int) else tuple(v)


def dw_conv2d(x, weight, bias=None, stride=1, padding=0, dilation=1,
              want_stats=False):
    """Depthwise conv2d (groups == C) on the HIP NHWC kernels.

    `weight` is the torch-native (C, 1, K, K) tensor; dtype mismatch with x
    (fp32 master weights under bf16 autocast) is resolved by a differentiable
    cast so weight grads land on the master dtype.

    want_stats: for k3 stride-1, also emit per-channel (sum, sumsq) of y for
    the following fused BatchNorm (attached as ``y._dfd_bn_stats``).
    """
    sh, sw = _pair(stride)
    ph, pw = _pair(padding)
    if weight.dtype != x.dtype:
        weight = weight.to(x.dtype)
    holder = [] if (want_stats and bias is None) else None
    y = _DwConv2d.apply(x, weight, sh, sw, ph, pw, holder)
    if bias is not None:
        y = y + bias.to(y.dtype).view(1, -1, 1, 1)
    elif holder:
        y._dfd_bn_stats = (holder[0],
                           y.shape[0] * y.shape[2] * y.shape[3], y.shape[1])
    return y
