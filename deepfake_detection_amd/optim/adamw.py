"""AdamW (decoupled weight decay), with a fused multi-tensor HIP path on
ROCm devices (BASELINE.json names fused AdamW as a deliverable).

Parity: reference dfd/timm/optim/adamw.py (decoupled `p *= 1 - lr*wd`).
"""

import math

import torch
from torch.optim import Optimizer


class AdamW(Optimizer):
    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=1e-2, amsgrad=False):
        if not 0.0 <= lr:
            raise ValueError("Invalid learning rate: {}".format(lr))
        if not 0.0 <= eps:
            raise ValueError("Invalid epsilon value: {}".format(eps))
        if not 0.0 <= betas[0] < 1.0:
            raise ValueError("Invalid beta parameter at index 0: {}".format(betas[0]))
        if not 0.0 <= betas[1] < 1.0:
            raise ValueError("Invalid beta parameter at index 1: {}".format(betas[1]))
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay, amsgrad=amsgrad)
        super().__init__(params, defaults)

    def __setstate__(self, state):
        super().__setstate__(state)
        for group in self.param_groups:
            group.setdefault("amsgrad", False)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        for group in self.param_groups:
            fused_bucket = []
            for p in group["params"]:
                if p.grad is None:
                    continue
                grad = p.grad
                if grad.is_sparse:
                    raise RuntimeError("AdamW does not support sparse gradients")
                amsgrad = group["amsgrad"]
                state = self.state[p]

                if len(state) == 0:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(p)
                    state["exp_avg_sq"] = torch.zeros_like(p)
                    if amsgrad:
                        state["max_exp_avg_sq"] = torch.zeros_like(p)
                state["step"] += 1

                from .rmsprop_tf import _fusable

                if p.is_cuda and not amsgrad and _fusable(p, grad, state):
                    fused_bucket.append((p, grad, state))
                    continue

                beta1, beta2 = group["betas"]
                # decoupled weight decay
                p.mul_(1 - group["lr"] * group["weight_decay"])
                exp_avg, exp_avg_sq = state["exp_avg"], state["exp_avg_sq"]
                exp_avg.mul_(beta1).add_(grad, alpha=1 - beta1)
                exp_avg_sq.mul_(beta2).addcmul_(grad, grad, value=1 - beta2)
                bias_correction1 = 1 - beta1 ** state["step"]
                bias_correction2 = 1 - beta2 ** state["step"]
                if amsgrad:
                    max_exp_avg_sq = state["max_exp_avg_sq"]
                    torch.maximum(max_exp_avg_sq, exp_avg_sq, out=max_exp_avg_sq)
                    denom = (max_exp_avg_sq.sqrt() / math.sqrt(bias_correction2)).add_(group["eps"])
                else:
                    denom = (exp_avg_sq.sqrt() / math.sqrt(bias_correction2)).add_(group["eps"])
                step_size = group["lr"] / bias_correction1
                p.addcdiv_(exp_avg, denom, value=-step_size)

            if fused_bucket:
                from ..ops.optim_kernels import adamw_multi_tensor

                beta1, beta2 = group["betas"]
                # all tensors in the bucket share the same step count in
                # steady state; group by step to stay exact after resume
                steps = {}
                for p, g, s in fused_bucket:
                    steps.setdefault(s["step"], []).append((p, g, s))
                for step_val, items in steps.items():
                    adamw_multi_tensor(
                        params=[p for p, _, _ in items],
                        grads=[g for _, g, _ in items],
                        exp_avgs=[s["exp_avg"] for _, _, s in items],
                        exp_avg_sqs=[s["exp_avg_sq"] for _, _, s in items],
                        step=step_val, lr=group["lr"], beta1=beta1, beta2=beta2,
                        eps=group["eps"], weight_decay=group["weight_decay"])

        return loss
