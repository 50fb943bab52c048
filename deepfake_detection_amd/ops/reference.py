"""Pure-PyTorch fp32 reference implementations of every HIP op.

These are the numerics oracles: each GPU kernel test compares the HIP op
against the matching function here (run in fp32), per the test strategy
SURVEY.md §4. They also serve as the CPU execution path (no GPU available).
"""

import torch
import torch.nn.functional as F

ACTS = {
    "none": lambda x: x,
    "relu": F.relu,
    "silu": F.silu,
}


def bn_act_forward(x, weight, bias, running_mean, running_var, training, momentum, eps, act="silu"):
    """BatchNorm2d (+ activation) exactly as torch — the semantics the fused
    HIP kernel must reproduce (fp32 stats, biased batch var for normalize,
    unbiased var into running stats — torch.nn.BatchNorm2d contract)."""
    y = F.batch_norm(x, running_mean, running_var, weight, bias, training, momentum, eps)
    return ACTS[act](y)


def se_forward(x, w_reduce, b_reduce, w_expand, b_expand, act="silu", gate="sigmoid"):
    """Squeeze-excite chain: global-avg-pool -> 1x1 reduce -> act -> 1x1
    expand -> sigmoid gate -> broadcast multiply
    (reference efficientnet_blocks.py:93-110)."""
    s = x.mean(dim=(2, 3), keepdim=True)
    s = F.conv2d(s, w_reduce, b_reduce)
    s = ACTS[act](s)
    s = F.conv2d(s, w_expand, b_expand)
    assert gate == "sigmoid"
    return x * torch.sigmoid(s)


def global_avg_pool(x):
    return x.mean(dim=(2, 3))


def rmsprop_tf_step(p, grad, square_avg, momentum_buffer, lr, alpha, eps, momentum, weight_decay,
                    decoupled_decay=False, lr_in_momentum=True):
    """One RMSpropTF update, TF semantics (reference rmsprop_tf.py:80-120):
    square_avg initialized to ONES by the optimizer, eps added INSIDE the
    sqrt, LR folded into the momentum buffer."""
    if weight_decay != 0:
        if decoupled_decay:
            p.add_(p, alpha=-weight_decay)
        else:
            grad = grad.add(p, alpha=weight_decay)
    one_minus_alpha = 1.0 - alpha
    square_avg.add_(grad.pow(2) - square_avg, alpha=one_minus_alpha)
    avg = square_avg.add(eps).sqrt_()
    if momentum > 0:
        if lr_in_momentum:
            momentum_buffer.mul_(momentum).addcdiv_(grad, avg, value=lr)
            p.add_(-momentum_buffer)
        else:
            momentum_buffer.mul_(momentum).addcdiv_(grad, avg)
            p.add_(momentum_buffer, alpha=-lr)
    else:
        p.addcdiv_(grad, avg, value=-lr)


def adamw_step(p, grad, exp_avg, exp_avg_sq, step, lr, beta1, beta2, eps, weight_decay):
    """Decoupled AdamW update (torch semantics)."""
    p.mul_(1 - lr * weight_decay)
    exp_avg.mul_(beta1).add_(grad, alpha=1 - beta1)
    exp_avg_sq.mul_(beta2).addcmul_(grad, grad, value=1 - beta2)
    bias_c1 = 1 - beta1 ** step
    bias_c2 = 1 - beta2 ** step
    denom = (exp_avg_sq / bias_c2).sqrt_().add_(eps)
    p.addcdiv_(exp_avg, denom, value=-lr / bias_c1)


def ema_update(ema_p, model_p, decay):
    """EMA: ema = ema*d + m*(1-d) (reference timm/utils.py:329-340)."""
    ema_p.mul_(decay).add_(model_p.to(ema_p.dtype), alpha=1.0 - decay)


def normalize_uint8(x_u8, mean, std, out_dtype=torch.float32):
    """Prefetcher device op: uint8 (N,C,H,W) -> float, (x - mean)/std with
    (1,C,1,1) broadcast (reference loader.py:246-253)."""
    x = x_u8.to(out_dtype)
    return (x - mean.to(out_dtype)) / std.to(out_dtype)
