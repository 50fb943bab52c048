"""Secondary optimizers: Nadam, RAdam, NovoGrad, Lookahead.

Capability parity with reference dfd/timm/optim/{nadam,radam,novograd,
nvnovograd,lookahead}.py — modernized to torch>=2 APIs.
"""

import math
from collections import defaultdict

import torch
from torch.optim import Optimizer


class Nadam(Optimizer):
    """Adam with Nesterov momentum."""

    def __init__(self, params, lr=2e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0, schedule_decay=4e-3):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay, schedule_decay=schedule_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            for p in group["params"]:
                if p.grad is None:
                    continue
                grad = p.grad
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    state["m_schedule"] = 1.0
                    state["exp_avg"] = torch.zeros_like(p)
                    state["exp_avg_sq"] = torch.zeros_like(p)

                m_schedule = state["m_schedule"]
                schedule_decay = group["schedule_decay"]
                exp_avg, exp_avg_sq = state["exp_avg"], state["exp_avg_sq"]
                beta1, beta2 = group["betas"]
                eps = group["eps"]
                state["step"] += 1
                t = state["step"]

                if group["weight_decay"] != 0:
                    grad = grad.add(p, alpha=group["weight_decay"])

                momentum_cache_t = beta1 * (1.0 - 0.5 * (0.96 ** (t * schedule_decay)))
                momentum_cache_t_1 = beta1 * (1.0 - 0.5 * (0.96 ** ((t + 1) * schedule_decay)))
                m_schedule_new = m_schedule * momentum_cache_t
                m_schedule_next = m_schedule * momentum_cache_t * momentum_cache_t_1
                state["m_schedule"] = m_schedule_new

                exp_avg.mul_(beta1).add_(grad, alpha=1.0 - beta1)
                exp_avg_sq.mul_(beta2).addcmul_(grad, grad, value=1.0 - beta2)
                exp_avg_sq_prime = exp_avg_sq.div(1.0 - beta2 ** t)
                denom = exp_avg_sq_prime.sqrt_().add_(eps)

                p.addcdiv_(grad, denom, value=-group["lr"] * (1.0 - momentum_cache_t) / (1.0 - m_schedule_new))
                p.addcdiv_(exp_avg, denom, value=-group["lr"] * momentum_cache_t_1 / (1.0 - m_schedule_next))
        return loss


class RAdam(Optimizer):
    """Rectified Adam (variance warmup)."""

    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8, weight_decay=0):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        self.buffer = [[None, None, None] for _ in range(10)]
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            for p in group["params"]:
                if p.grad is None:
                    continue
                grad = p.grad.float()
                p_fp32 = p.float()
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(p_fp32)
                    state["exp_avg_sq"] = torch.zeros_like(p_fp32)
                else:
                    state["exp_avg"] = state["exp_avg"].type_as(p_fp32)
                    state["exp_avg_sq"] = state["exp_avg_sq"].type_as(p_fp32)

                exp_avg, exp_avg_sq = state["exp_avg"], state["exp_avg_sq"]
                beta1, beta2 = group["betas"]

                exp_avg_sq.mul_(beta2).addcmul_(grad, grad, value=1 - beta2)
                exp_avg.mul_(beta1).add_(grad, alpha=1 - beta1)

                state["step"] += 1
                buffered = self.buffer[int(state["step"] % 10)]
                if state["step"] == buffered[0]:
                    N_sma, step_size = buffered[1], buffered[2]
                else:
                    buffered[0] = state["step"]
                    beta2_t = beta2 ** state["step"]
                    N_sma_max = 2 / (1 - beta2) - 1
                    N_sma = N_sma_max - 2 * state["step"] * beta2_t / (1 - beta2_t)
                    buffered[1] = N_sma
                    if N_sma >= 5:
                        step_size = group["lr"] * math.sqrt(
                            (1 - beta2_t) * (N_sma - 4) / (N_sma_max - 4)
                            * (N_sma - 2) / N_sma * N_sma_max / (N_sma_max - 2)
                        ) / (1 - beta1 ** state["step"])
                    else:
                        step_size = group["lr"] / (1 - beta1 ** state["step"])
                    buffered[2] = step_size

                if group["weight_decay"] != 0:
                    p_fp32.add_(p_fp32, alpha=-group["weight_decay"] * group["lr"])

                if N_sma >= 5:
                    denom = exp_avg_sq.sqrt().add_(group["eps"])
                    p_fp32.addcdiv_(exp_avg, denom, value=-step_size)
                else:
                    p_fp32.add_(exp_avg, alpha=-step_size)

                p.copy_(p_fp32)
        return loss


class NovoGrad(Optimizer):
    """NovoGrad: layer-wise second moment + gradient normalization."""

    def __init__(self, params, lr=1e-3, betas=(0.95, 0.98), eps=1e-8,
                 weight_decay=0, grad_averaging=False, amsgrad=False):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay,
                        grad_averaging=grad_averaging, amsgrad=amsgrad)
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
            for p in group["params"]:
                if p.grad is None:
                    continue
                grad = p.grad
                state = self.state[p]
                g2 = grad.pow(2).sum()
                if len(state) == 0:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(p)
                    state["exp_avg_sq"] = g2.clone()
                    if group["amsgrad"]:
                        state["max_exp_avg_sq"] = g2.clone()
                exp_avg, exp_avg_sq = state["exp_avg"], state["exp_avg_sq"]
                beta1, beta2 = group["betas"]
                state["step"] += 1

                exp_avg_sq.mul_(beta2).add_(g2, alpha=1 - beta2)
                if group["amsgrad"]:
                    max_exp_avg_sq = state["max_exp_avg_sq"]
                    torch.maximum(max_exp_avg_sq, exp_avg_sq, out=max_exp_avg_sq)
                    denom = max_exp_avg_sq.sqrt().add_(group["eps"])
                else:
                    denom = exp_avg_sq.sqrt().add_(group["eps"])

                norm_grad = grad / denom
                if group["weight_decay"] != 0:
                    norm_grad = norm_grad.add(p, alpha=group["weight_decay"])
                if group["grad_averaging"]:
                    norm_grad = norm_grad.mul(1 - beta1)
                exp_avg.mul_(beta1).add_(norm_grad)
                p.add_(exp_avg, alpha=-group["lr"])
        return loss


class PlainRAdam(Optimizer):
    """RAdam without the per-step-modulo buffer cache: the rectification
    term is recomputed every step (reference radam.py PlainRAdam, :88-152).
    Same math as RAdam; kept as a separate registry entry for parity."""

    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8, weight_decay=0):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            beta1, beta2 = group["betas"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                grad = p.grad.float()
                p_fp32 = p.float()
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(p_fp32)
                    state["exp_avg_sq"] = torch.zeros_like(p_fp32)
                else:
                    state["exp_avg"] = state["exp_avg"].type_as(p_fp32)
                    state["exp_avg_sq"] = state["exp_avg_sq"].type_as(p_fp32)
                exp_avg, exp_avg_sq = state["exp_avg"], state["exp_avg_sq"]

                exp_avg_sq.mul_(beta2).addcmul_(grad, grad, value=1 - beta2)
                exp_avg.mul_(beta1).add_(grad, alpha=1 - beta1)

                state["step"] += 1
                t = state["step"]
                beta2_t = beta2 ** t
                n_sma_max = 2 / (1 - beta2) - 1
                n_sma = n_sma_max - 2 * t * beta2_t / (1 - beta2_t)

                if group["weight_decay"] != 0:
                    p_fp32.add_(p_fp32, alpha=-group["weight_decay"] * group["lr"])

                if n_sma >= 5:
                    step_size = group["lr"] * math.sqrt(
                        (1 - beta2_t) * (n_sma - 4) / (n_sma_max - 4)
                        * (n_sma - 2) / n_sma * n_sma_max / (n_sma_max - 2)
                    ) / (1 - beta1 ** t)
                    denom = exp_avg_sq.sqrt().add_(group["eps"])
                    p_fp32.addcdiv_(exp_avg, denom, value=-step_size)
                else:
                    step_size = group["lr"] / (1 - beta1 ** t)
                    p_fp32.add_(exp_avg, alpha=-step_size)
                p.copy_(p_fp32)
        return loss


class NvNovoGrad(Optimizer):
    """Nvidia Jasper NovoGrad variant (reference nvnovograd.py:13-120):
    per-tensor SCALAR second moment initialized lazily — zeros, then copies
    the first grad-norm — versus NovoGrad's construction-time init."""

    def __init__(self, params, lr=1e-3, betas=(0.95, 0.98), eps=1e-8,
                 weight_decay=0, grad_averaging=False, amsgrad=False):
        if not 0.0 <= lr:
            raise ValueError("Invalid learning rate: {}".format(lr))
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay,
                        grad_averaging=grad_averaging, amsgrad=amsgrad)
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
            beta1, beta2 = group["betas"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                grad = p.grad
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(p)
                    state["exp_avg_sq"] = torch.zeros([], device=p.device)
                    if group["amsgrad"]:
                        state["max_exp_avg_sq"] = torch.zeros([], device=p.device)
                exp_avg, exp_avg_sq = state["exp_avg"], state["exp_avg_sq"]
                state["step"] += 1

                norm = grad.pow(2).sum()
                if exp_avg_sq == 0:
                    exp_avg_sq.copy_(norm)
                else:
                    exp_avg_sq.mul_(beta2).add_(norm, alpha=1 - beta2)

                if group["amsgrad"]:
                    max_exp_avg_sq = state["max_exp_avg_sq"]
                    torch.maximum(max_exp_avg_sq, exp_avg_sq, out=max_exp_avg_sq)
                    denom = max_exp_avg_sq.sqrt().add_(group["eps"])
                else:
                    denom = exp_avg_sq.sqrt().add_(group["eps"])

                ngrad = grad / denom
                if group["weight_decay"] != 0:
                    ngrad = ngrad.add(p, alpha=group["weight_decay"])
                if group["grad_averaging"]:
                    ngrad = ngrad.mul(1 - beta1)
                exp_avg.mul_(beta1).add_(ngrad)
                p.add_(exp_avg, alpha=-group["lr"])
        return loss


class Lookahead(Optimizer):
    """Lookahead wrapper (k slow steps) — reference lookahead.py:10; the
    trainer calls `sync_lookahead()` at epoch end (reference train.py:697-698)."""

    def __init__(self, base_optimizer, alpha=0.5, k=6):
        if not 0.0 <= alpha <= 1.0:
            raise ValueError("Invalid slow update rate: {}".format(alpha))
        if not 1 <= k:
            raise ValueError("Invalid lookahead steps: {}".format(k))
        defaults = dict(lookahead_alpha=alpha, lookahead_k=k, lookahead_step=0)
        self.base_optimizer = base_optimizer
        self.param_groups = self.base_optimizer.param_groups
        self.defaults = base_optimizer.defaults
        self.defaults.update(defaults)
        self.state = defaultdict(dict)
        for name, default in defaults.items():
            for group in self.param_groups:
                group.setdefault(name, default)

    @torch.no_grad()
    def update_slow(self, group):
        for fast_p in group["params"]:
            if fast_p.grad is None:
                continue
            param_state = self.state[fast_p]
            if "slow_buffer" not in param_state:
                param_state["slow_buffer"] = torch.empty_like(fast_p)
                param_state["slow_buffer"].copy_(fast_p)
            slow = param_state["slow_buffer"]
            slow.add_(fast_p - slow, alpha=group["lookahead_alpha"])
            fast_p.copy_(slow)

    def sync_lookahead(self):
        for group in self.param_groups:
            self.update_slow(group)

    def step(self, closure=None):
        loss = self.base_optimizer.step(closure)
        for group in self.param_groups:
            group["lookahead_step"] += 1
            if group["lookahead_step"] % group["lookahead_k"] == 0:
                self.update_slow(group)
        return loss

    def state_dict(self):
        fast_state_dict = self.base_optimizer.state_dict()
        slow_state = {
            (id(k) if isinstance(k, torch.Tensor) else k): v
            for k, v in self.state.items()
        }
        return {
            "state": fast_state_dict["state"],
            "slow_state": slow_state,
            "param_groups": fast_state_dict["param_groups"],
        }

    def load_state_dict(self, state_dict):
        fast_state_dict = {
            "state": state_dict["state"],
            "param_groups": state_dict["param_groups"],
        }
        self.base_optimizer.load_state_dict(fast_state_dict)
        self.param_groups = self.base_optimizer.param_groups

    def zero_grad(self, set_to_none=True):
        self.base_optimizer.zero_grad(set_to_none=set_to_none)
