"""RMSprop with TensorFlow-style semantics — the production optimizer
(reference scripts/train.sh:6 `--opt rmsproptf --opt-eps .001`).

Exact-parity semantics (reference dfd/timm/optim/rmsprop_tf.py):
  * square_avg state initialized to ONES, not zeros (:80)
  * eps added INSIDE the sqrt (:105-107)
  * LR folded into the momentum buffer (`lr_in_momentum`, :112-114)
  * one-minus-alpha update order: sa += (1-a)*(g^2 - sa) (:98)

On ROCm devices the per-parameter elementwise chain runs as one fused
multi-tensor HIP kernel (ops/hip/optim.hip) instead of the ~6 eager kernels
per tensor; numerics are the same and tested against this module.
"""

import torch
from torch.optim import Optimizer


def _fusable(p, grad, state):
    """Fused path needs param/grad/state to share one dense memory layout
    (plain or channels_last contiguous — the elementwise update is
    layout-agnostic as long as all tensors agree)."""
    dense = p.is_contiguous() or p.is_contiguous(memory_format=torch.channels_last) \
        or (p.dim() == 5 and p.is_contiguous(memory_format=torch.channels_last_3d))
    if not (p.dtype == torch.float32 and dense):
        return False
    if grad.stride() != p.stride():
        return False
    for v in state.values():
        if isinstance(v, torch.Tensor) and v.stride() != p.stride():
            return False
    return True


class RMSpropTF(Optimizer):
    def __init__(self, params, lr=1e-2, alpha=0.9, eps=1e-10, weight_decay=0,
                 momentum=0.0, centered=False, decoupled_decay=False, lr_in_momentum=True):
        if not 0.0 <= lr:
            raise ValueError("Invalid learning rate: {}".format(lr))
        if not 0.0 <= eps:
            raise ValueError("Invalid epsilon value: {}".format(eps))
        if not 0.0 <= momentum:
            raise ValueError("Invalid momentum value: {}".format(momentum))
        if not 0.0 <= weight_decay:
            raise ValueError("Invalid weight_decay value: {}".format(weight_decay))
        if not 0.0 <= alpha:
            raise ValueError("Invalid alpha value: {}".format(alpha))

        defaults = dict(
            lr=lr, momentum=momentum, alpha=alpha, eps=eps, centered=centered,
            weight_decay=weight_decay, decoupled_decay=decoupled_decay,
            lr_in_momentum=lr_in_momentum)
        super().__init__(params, defaults)

    def __setstate__(self, state):
        super().__setstate__(state)
        for group in self.param_groups:
            group.setdefault("momentum", 0)
            group.setdefault("centered", False)

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
                    raise RuntimeError("RMSpropTF does not support sparse gradients")
                state = self.state[p]

                if len(state) == 0:
                    state["step"] = 0
                    # TF inits the RMS accumulator to ones (PyTorch uses zeros)
                    state["square_avg"] = torch.ones_like(p)
                    if group["momentum"] > 0:
                        state["momentum_buffer"] = torch.zeros_like(p)
                    if group["centered"]:
                        state["grad_avg"] = torch.zeros_like(p)
                state["step"] += 1

                if p.is_cuda and not group["centered"] and _fusable(p, grad, state):
                    fused_bucket.append((p, grad, state))
                    continue

                self._step_one(group, p, grad, state)

            if fused_bucket:
                self._step_fused(group, fused_bucket)

        return loss

    @staticmethod
    def _step_one(group, p, grad, state):
        square_avg = state["square_avg"]
        one_minus_alpha = 1.0 - group["alpha"]

        if group["weight_decay"] != 0:
            if group["decoupled_decay"]:
                p.add_(p, alpha=-group["weight_decay"])
            else:
                grad = grad.add(p, alpha=group["weight_decay"])

        # TF order of ops for the squared-gradient accumulator
        square_avg.add_(grad.pow(2) - square_avg, alpha=one_minus_alpha)

        if group["centered"]:
            grad_avg = state["grad_avg"]
            grad_avg.add_(grad - grad_avg, alpha=one_minus_alpha)
            avg = square_avg.addcmul(grad_avg, grad_avg, value=-1).add(group["eps"]).sqrt_()
        else:
            avg = square_avg.add(group["eps"]).sqrt_()  # eps inside sqrt

        if group["momentum"] > 0:
            buf = state["momentum_buffer"]
            if group["lr_in_momentum"]:
                # TF accumulates the LR scaling inside the momentum buffer
                buf.mul_(group["momentum"]).addcdiv_(grad, avg, value=group["lr"])
                p.add_(-buf)
            else:
                buf.mul_(group["momentum"]).addcdiv_(grad, avg)
                p.add_(buf, alpha=-group["lr"])
        else:
            p.addcdiv_(grad, avg, value=-group["lr"])

    def _step_fused(self, group, bucket):
        """Fused multi-tensor HIP path (one kernel for the whole group)."""
        from ..ops.optim_kernels import rmsprop_tf_multi_tensor

        rmsprop_tf_multi_tensor(
            params=[p for p, _, _ in bucket],
            grads=[g for _, g, _ in bucket],
            square_avgs=[s["square_avg"] for _, _, s in bucket],
            momentum_buffers=[
                s.get("momentum_buffer") for _, _, s in bucket
            ] if group["momentum"] > 0 else None,
            lr=group["lr"], alpha=group["alpha"], eps=group["eps"],
            momentum=group["momentum"], weight_decay=group["weight_decay"],
            decoupled_decay=group["decoupled_decay"],
            lr_in_momentum=group["lr_in_momentum"],
        )
