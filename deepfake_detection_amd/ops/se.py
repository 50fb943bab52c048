"""Autograd wrapper for the fused squeeze-excite HIP chain.

Forward: one fused kernel (pool -> reduce GEMV -> SiLU -> expand GEMV ->
sigmoid) + one gate-apply kernel. Backward: HW-scale reductions in HIP
(dx-direct + per-(n,c) dgate; pooled-ds broadcast add), the tiny dense
algebra (Cr x C GEMMs over the batch) in torch/rocBLAS.
"""

import torch

from .extension import load_extension


def _act_bwd(z, act):
    if act == "relu":
        return (z > 0).to(z.dtype)
    s = torch.sigmoid(z)
    return s * (1 + z * (1 - s))


class _FusedSE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w1, b1, w2, b2, act):
        ext = load_extension()
        x = x.contiguous(memory_format=torch.channels_last)
        w1_2d = w1.flatten(1).float()  # (Cr, C)
        w2_2d = w2.flatten(1).float()  # (C, Cr)
        y, s, z1, r, g = ext.se_fwd(x, w1_2d, b1.float(), w2_2d, b2.float(), act)
        ctx.save_for_backward(x, w1_2d, w2_2d, s, z1, r, g)
        ctx.act = act
        ctx.w_shapes = (w1.shape, w2.shape)
        ctx.w_dtypes = (w1.dtype, w2.dtype)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = load_extension()
        x, w1, w2, s, z1, r, g = ctx.saved_tensors
        act = ctx.act
        dx, dg = ext.se_bwd_reduce(dy, x, g)
        # dense chain (all (N,C)/(N,Cr) fp32 — tiny GEMMs)
        dz2 = dg * g * (1 - g)                      # (N, C)
        dw2 = dz2.t() @ r                           # (C, Cr)
        db2 = dz2.sum(0)
        dr = dz2 @ w2                               # (N, Cr)
        dz1 = dr * _act_bwd(z1, act)                # (N, Cr)
        dw1 = dz1.t() @ s                           # (Cr, C)
        db1 = dz1.sum(0)
        ds = dz1 @ w1                               # (N, C)
        ext.se_bwd_add_pool(dx, ds)

        w1_shape, w2_shape = ctx.w_shapes
        w1_dtype, w2_dtype = ctx.w_dtypes
        return (dx,
                dw1.view(w1_shape).to(w1_dtype),
                db1.to(w1_dtype),
                dw2.view(w2_shape).to(w2_dtype),
                db2.to(w2_dtype),
                None)


def fused_se(x, w_reduce, b_reduce, w_expand, b_expand, act="silu"):
    return _FusedSE.apply(x, w_reduce, b_reduce, w_expand, b_expand, act)
