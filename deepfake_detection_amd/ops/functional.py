"""Dispatch layer: model blocks call these; on ROCm devices they run the
hand-written gfx950 HIP kernels, on CPU the PyTorch reference path.

The HIP path is NOT optional on a GPU machine: if a CUDA(ROCm) tensor
arrives and the extension is missing, we raise (see ops/extension.py).
"""

import torch
import torch.nn.functional as F

from .extension import gpu_ops_required

_FUSED_ACTS = ("none", "relu", "silu")


def fusable_bn(bn) -> bool:
    """True only for a plain ``nn.BatchNorm2d`` (exact type).

    Subclasses (SplitBatchNorm2d, SyncBatchNorm, ...) override ``forward``
    with semantics the fused kernel does not implement — an isinstance
    check would silently bypass them (aux-BN stats never updated), so the
    fused path requires the exact base type.
    """
    import torch.nn as nn

    return type(bn) is nn.BatchNorm2d


def act_name_of(module) -> str:
    """Map an activation module instance to a fused-kernel act name."""
    import torch.nn as nn

    from ..models import layers as L

    if module is None or isinstance(module, nn.Identity):
        return "none"
    if isinstance(module, (nn.SiLU, L.Swish)):
        return "silu"
    if isinstance(module, (nn.ReLU, nn.ReLU6)):
        return "relu"
    return "other"


def bn_act(x, bn, act: str = "silu", residual=None, drop_path_mask=None):
    """BatchNorm2d + activation (+ optional fused drop_path scale and
    residual add), fused on GPU (HIP kernel, NHWC, bf16 I/O, fp32 stats —
    SURVEY.md §2.6 items 5 and 7), torch ops on CPU.

    drop_path_mask: fp32 [B] per-sample 0-or-1/keep values (stochastic
    depth, reference drop.py:84-100), applied to the BN+act output before
    the residual add."""
    if not fusable_bn(bn):
        # Subclass with its own semantics (SplitBatchNorm2d, SyncBatchNorm...):
        # module dispatch, then the named activation.
        y = bn(x)
        if act == "silu":
            y = F.silu(y)
        elif act == "relu":
            y = F.relu(y)
        if drop_path_mask is not None:
            y = y * drop_path_mask.to(y.dtype).view(-1, 1, 1, 1)
        if residual is not None:
            y = y + residual
        return y
    if x.is_cuda and gpu_ops_required() and act in _FUSED_ACTS:
        from .bn_act import fused_bn_act

        # the kernel requires fp32 BN params/stats. A .half()'ed model (the
        # reference's model_half.pth.tar inference path, test.py:44-47)
        # carries fp16 BN params: cast read-only copies for EVAL (no
        # write-back of running stats); a halved model in TRAINING falls
        # through to the torch path below.
        w, b = bn.weight, bn.bias
        rm, rv = bn.running_mean, bn.running_var
        halved = any(t is not None and t.dtype != torch.float32
                     for t in (w, rm, rv))
        use_fused = True
        if halved:
            if bn.training:
                use_fused = False  # halved training: eager fallback below
            else:
                w = w.float() if w is not None else w
                b = b.float() if b is not None else b
                rm = rm.float() if rm is not None else rm
                rv = rv.float() if rv is not None else rv
        if use_fused:
            # producer-fused BN stats: the conv kernel that wrote x may have
            # attached bucketed per-channel (sum, sumsq) partials
            stats = None
            attached = getattr(x, "_dfd_bn_stats", None)
            if attached is not None and bn.training:
                buckets, m, c = attached
                if c == x.shape[1] and m == x.shape[0] * x.shape[2] * x.shape[3]:
                    stats = buckets
            return fused_bn_act(
                x, w, b, rm, rv,
                bn.training, bn.momentum, bn.eps, act, residual, stats,
                drop_path_mask,
            )
    y = F.batch_norm(
        x, bn.running_mean, bn.running_var, bn.weight, bn.bias,
        bn.training, bn.momentum if bn.momentum is not None else 0.1, bn.eps,
    )
    if act == "silu":
        y = F.silu(y)
    elif act == "relu":
        y = F.relu(y)
    if drop_path_mask is not None:
        y = y * drop_path_mask.to(y.dtype).view(-1, 1, 1, 1)
    if residual is not None:
        y = y + residual
    return y


def se(x, conv_reduce, act_module, conv_expand):
    """Squeeze-excite chain (pool -> 1x1 -> act -> 1x1 -> sigmoid -> mul).

    GPU: fused HIP kernel chain (ops/hip/se.hip); CPU: reference path.
    Falls back to composable torch ops for non-silu/relu gates.
    """
    act = act_name_of(act_module)
    if x.is_cuda and gpu_ops_required() and act in _FUSED_ACTS:
        from .se import fused_se

        return fused_se(x, conv_reduce.weight, conv_reduce.bias,
                        conv_expand.weight, conv_expand.bias, act)
    s = x.mean(dim=(2, 3), keepdim=True)
    s = conv_reduce(s)
    s = act_module(s)
    s = conv_expand(s)
    return x * torch.sigmoid(s)


def global_avg_pool(x):
    """Global average pool NxCxHxW -> NxC (head pooling)."""
    if x.is_cuda and gpu_ops_required():
        from .pool import fused_global_avg_pool

        return fused_global_avg_pool(x)
    return x.mean(dim=(2, 3))
