"""Stem convolution (kxk stride-2, few input channels) on the implicit-GEMM
MFMA kernels (ops/hip/stemconv.hip).

The reference's stem (3x3 s2, 12->256 for deepfake_v4 / 3->48 for B4) goes
through create_conv2d to cuDNN (reference efficientnet.py:275); these were
the last MIOpen kernels in the MI355X training step. Forward gathers input
patches per fragment element; bwd-weight is a split-M two-stage reduction;
bwd-data is not needed (the stem input carries no grad) and falls back to
torch on the rare occasions it is requested.
"""

import torch

from .extension import load_extension

STATS_BUCKETS = 64


def stem_supported(x, weight, stride, padding, dilation, groups) -> bool:
    # bf16/fp16 (fragments compute in bf16 either way); fp32 falls back to
    # torch for full-precision parity
    if groups != 1 or x.dtype not in (torch.bfloat16, torch.float16):
        return False
    n, cin, kh, kw = weight.shape
    if kh != kw or cin > 16:
        return False
    sh, sw = (stride, stride) if isinstance(stride, int) else tuple(stride)
    ph, pw = (padding, padding) if isinstance(padding, int) else tuple(padding)
    dh, dw_ = (dilation, dilation) if isinstance(dilation, int) else tuple(dilation)
    return dh == 1 and dw_ == 1 and sh == sw and ph == pw and n % 8 == 0


def _pack_weight(weight):
    """(N, Cin, KH, KW) -> bf16 [N, Kpad], k = (kh*KW + kw)*Cin + c."""
    n, cin, kh, kw = weight.shape
    kpad = (kh * kw * cin + 31) // 32 * 32
    flat = weight.permute(0, 2, 3, 1).reshape(n, kh * kw * cin)
    packed = torch.zeros(n, kpad, device=weight.device, dtype=torch.bfloat16)
    packed[:, : kh * kw * cin] = flat.to(torch.bfloat16)
    return packed


class _StemConv2d(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, sh, ph, stats_out):
        ext = load_extension()
        x = x.contiguous(memory_format=torch.channels_last)
        n, cin, kh, kw = weight.shape
        packed = _pack_weight(weight)
        stats = None
        if stats_out is not None:
            stats = torch.zeros(STATS_BUCKETS, 2, n, device=x.device,
                                dtype=torch.float32)
            stats_out.append(stats)
        y = ext.stem_conv2d_fwd(x, packed, n, kh, kw, sh, sh, ph, ph, stats)
        ctx.save_for_backward(x)
        ctx.meta = (n, cin, kh, kw, sh, ph, weight.dtype)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = load_extension()
        (x,) = ctx.saved_tensors
        n, cin, kh, kw, sh, ph, wdtype = ctx.meta
        dy = dy.contiguous(memory_format=torch.channels_last)
        dx = dw = None
        if ctx.needs_input_grad[0]:
            # the module falls back to F.conv2d when x requires grad
            raise RuntimeError(
                "stem_conv2d: input gradient not supported (stem input has no "
                "grad in training); use F.conv2d for this case")
        if ctx.needs_input_grad[1]:
            dw_pack = ext.stem_conv2d_bwd_weight(dy, x, kh, kw, sh, sh, ph, ph)
            kk = kh * kw * cin
            dw = (dw_pack[:, :kk].view(n, kh, kw, cin)
                  .permute(0, 3, 1, 2).contiguous().to(wdtype))
        return dx, dw, None, None, None


def stem_conv2d(x, weight, bias=None, stride=2, padding=1, want_stats=False):
    """Stem conv on the MFMA implicit-GEMM kernels (bf16 NHWC).

    want_stats attaches bucketed per-channel (sum, sumsq) of y as
    ``y._dfd_bn_stats`` for the following fused BatchNorm.
    """
    sh = stride if isinstance(stride, int) else stride[0]
    ph = padding if isinstance(padding, int) else padding[0]
    if weight.dtype != x.dtype:
        weight = weight.to(x.dtype)
    holder = [] if (want_stats and bias is None) else None
    y = _StemConv2d.apply(x, weight, sh, ph, holder)
    if bias is not None:
        y = y + bias.to(y.dtype).view(1, -1, 1, 1)
    elif holder:
        y._dfd_bn_stats = (holder[0],
                           y.shape[0] * y.shape[2] * y.shape[3], y.shape[1])
    return y
