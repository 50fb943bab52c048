"""Model EMA (reference dfd/timm/utils.py:277-340): deepcopy'd eval model,
`ema = ema*d + m*(1-d)` per step, optional CPU placement, checkpoint
`state_dict_ema` load.

MI355X path: the per-tensor update loop collapses into ONE fused
multi-tensor HIP kernel (ops/hip/optim.hip ema_multi_tensor) when the EMA
model lives on the GPU.
"""

import logging
from collections import OrderedDict
from copy import deepcopy

import torch

_logger = logging.getLogger(__name__)


class ModelEma:
    def __init__(self, model, decay=0.9999, device="", resume=""):
        self.ema = deepcopy(model)
        self.ema.eval()
        self.decay = decay
        self.device = device
        if device:
            self.ema.to(device=device)
        self.ema_has_module = hasattr(self.ema, "module")
        if resume:
            self._load_checkpoint(resume)
        for p in self.ema.parameters():
            p.requires_grad_(False)

    def _load_checkpoint(self, checkpoint_path):
        checkpoint = torch.load(checkpoint_path, map_location="cpu", weights_only=False)
        assert isinstance(checkpoint, dict)
        if "state_dict_ema" in checkpoint:
            new_state_dict = OrderedDict()
            for k, v in checkpoint["state_dict_ema"].items():
                if self.ema_has_module:
                    name = "module." + k if not k.startswith("module") else k
                else:
                    name = k
                new_state_dict[name] = v
            self.ema.load_state_dict(new_state_dict)
            _logger.info("Loaded state_dict_ema")
        else:
            _logger.warning("Failed to find state_dict_ema, starting from loaded model weights")

    @torch.no_grad()
    def update(self, model):
        needs_module = hasattr(model, "module") and not self.ema_has_module
        msd = model.state_dict()
        esd = self.ema.state_dict()

        ema_f, model_f = [], []
        for k, ema_v in esd.items():
            mk = "module." + k if needs_module else k
            model_v = msd[mk]
            if self.device:
                model_v = model_v.to(device=self.device)
            if ema_v.is_cuda and model_v.is_cuda and ema_v.is_floating_point() \
                    and model_v.dtype == ema_v.dtype:
                ema_f.append(ema_v)
                model_f.append(model_v)
            else:
                ema_v.copy_(ema_v * self.decay + (1.0 - self.decay) * model_v)

        if ema_f:
            from ..ops.extension import has_extension

            if has_extension():
                from ..ops.optim_kernels import ema_multi_tensor

                ema_multi_tensor(ema_f, model_f, self.decay)
            else:
                torch._foreach_mul_(ema_f, self.decay)
                torch._foreach_add_(ema_f, model_f, alpha=1.0 - self.decay)
