"""Data-path device ops: fused uint8 -> {bf16,fp16,fp32} normalize with
NCHW->NHWC relayout.

GPU: one HIP kernel (ops/hip/normalize.hip) replaces the reference's
half()/sub_()/div_() chain (reference loader.py:246-253).
CPU: torch math fallback for tests.
"""

import torch

from .extension import gpu_ops_required, load_extension


def normalize_uint8(x_u8, mean, std, out_dtype=torch.float32, channels_last=True):
    """(B, C, H, W) uint8 -> normalized float tensor.

    mean/std are (1, C, 1, 1) fp32 in 0..255 scale. Output is channels_last
    (NHWC) when requested — the layout the whole CNN hot path runs in.
    """
    if x_u8.is_cuda and gpu_ops_required():
        ext = load_extension()
        return ext.normalize_uint8_nhwc(
            x_u8, mean.flatten(), std.flatten(),
            str(out_dtype).replace("torch.", ""), channels_last)
    x = x_u8.to(out_dtype)
    x = (x - mean.to(x.dtype)) / std.to(x.dtype)
    if channels_last and x.dim() == 4:
        x = x.contiguous(memory_format=torch.channels_last)
    return x
