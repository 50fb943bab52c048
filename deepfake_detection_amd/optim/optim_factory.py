"""Optimizer factory.

Capability parity with reference dfd/timm/optim/optim_factory.py:
`add_weight_decay` routes 1-D / bias params to a wd=0 group (:11-23),
`create_optimizer` name dispatch with AdamW/RAdam wd/lr compensation
(:29-33) and `lookahead_` prefix wrapping (:96-98). The reference's apex
Fused* optimizers (:42-91) are replaced by our own fused HIP multi-tensor
paths inside RMSpropTF / AdamW (always on for ROCm tensors).
"""

import torch.optim as optim

from .adamw import AdamW
from .extras import Lookahead, Nadam, NovoGrad, NvNovoGrad, PlainRAdam, RAdam
from .rmsprop_tf import RMSpropTF


def add_weight_decay(model, weight_decay=1e-5, skip_list=()):
    decay = []
    no_decay = []
    for name, param in model.named_parameters():
        if not param.requires_grad:
            continue
        if len(param.shape) == 1 or name.endswith(".bias") or name in skip_list:
            no_decay.append(param)
        else:
            decay.append(param)
    return [
        {"params": no_decay, "weight_decay": 0.0},
        {"params": decay, "weight_decay": weight_decay},
    ]


def create_optimizer(args, model, filter_bias_and_bn=True):
    opt_lower = args.opt.lower()
    weight_decay = getattr(args, "weight_decay", 0.0)
    if "adamw" in opt_lower or "radam" in opt_lower:
        # decoupled decay in AdamW/RAdam is scaled by LR inside the update;
        # compensate so the CLI wd means the same thing (reference :29-33)
        weight_decay /= args.lr
    if weight_decay and filter_bias_and_bn:
        parameters = add_weight_decay(model, weight_decay)
        weight_decay = 0.0
    else:
        parameters = model.parameters()

    opt_split = opt_lower.split("_")
    opt_lower = opt_split[-1]
    opt_args = dict(lr=args.lr, weight_decay=weight_decay)
    if hasattr(args, "opt_eps") and args.opt_eps is not None:
        opt_args["eps"] = args.opt_eps
    if hasattr(args, "opt_betas") and args.opt_betas is not None:
        opt_args["betas"] = args.opt_betas

    if opt_lower == "sgd" or opt_lower == "nesterov":
        opt_args.pop("eps", None)
        optimizer = optim.SGD(parameters, momentum=args.momentum, nesterov=True, **opt_args)
    elif opt_lower == "momentum":
        opt_args.pop("eps", None)
        optimizer = optim.SGD(parameters, momentum=args.momentum, nesterov=False, **opt_args)
    elif opt_lower == "adam":
        optimizer = optim.Adam(parameters, **opt_args)
    elif opt_lower == "adamw":
        optimizer = AdamW(parameters, **opt_args)
    elif opt_lower == "nadam":
        optimizer = Nadam(parameters, **opt_args)
    elif opt_lower == "radam":
        optimizer = RAdam(parameters, **opt_args)
    elif opt_lower == "plainradam":
        optimizer = PlainRAdam(parameters, **opt_args)
    elif opt_lower == "adadelta":
        optimizer = optim.Adadelta(parameters, **opt_args)
    elif opt_lower == "rmsprop":
        optimizer = optim.RMSprop(parameters, alpha=0.9, momentum=args.momentum, **opt_args)
    elif opt_lower == "rmsproptf":
        optimizer = RMSpropTF(parameters, alpha=0.9, momentum=args.momentum, **opt_args)
    elif opt_lower == "novograd":
        optimizer = NovoGrad(parameters, **opt_args)
    elif opt_lower == "nvnovograd":
        optimizer = NvNovoGrad(parameters, **opt_args)
    else:
        raise ValueError("Invalid optimizer: %s" % args.opt)

    if len(opt_split) > 1 and opt_split[0] == "lookahead":
        optimizer = Lookahead(optimizer)

    return optimizer
