"""Data config resolver (reference dfd/timm/data/config.py:5-101): merges CLI
args with the model default_cfg into {input_size, interpolation, mean, std,
crop_pct}; supports the `input_size_v2` "12,600,600" string form (:17-21)."""

import logging

from .constants import (
    DEFAULT_CROP_PCT,
    IMAGENET_DEFAULT_MEAN,
    IMAGENET_DEFAULT_STD,
    IMAGENET_DPN_MEAN,
    IMAGENET_DPN_STD,
    IMAGENET_INCEPTION_MEAN,
    IMAGENET_INCEPTION_STD,
)

_logger = logging.getLogger(__name__)


def resolve_data_config(args, default_cfg={}, model=None, verbose=False):
    new_config = {}
    default_cfg = dict(default_cfg)
    if not default_cfg and model is not None and hasattr(model, "default_cfg"):
        default_cfg = model.default_cfg

    # input_size
    in_chans = 3
    if "chps" in args and args["chps"] is not None:
        in_chans = args["chps"]
    elif "img_channels" in args and args["img_channels"] is not None:
        in_chans = args["img_channels"]

    input_size = (in_chans, 224, 224)
    if "input_size_v2" in args and args["input_size_v2"] is not None:
        # "12,600,600" string form (reference config.py:17-21)
        parts = [int(x) for x in str(args["input_size_v2"]).split(",")]
        assert len(parts) == 3
        input_size = tuple(parts)
        in_chans = input_size[0]
    elif "input_size" in args and args["input_size"] is not None:
        if isinstance(args["input_size"], (tuple, list)) and len(args["input_size"]) == 3:
            input_size = tuple(args["input_size"])
            in_chans = input_size[0]
        else:
            img_size = int(args["input_size"]) if not isinstance(args["input_size"], (tuple, list)) \
                else args["input_size"][-1]
            input_size = (in_chans, img_size, img_size)
    elif "img_size" in args and args["img_size"] is not None:
        input_size = (in_chans, args["img_size"], args["img_size"])
    elif "input_size" in default_cfg:
        input_size = default_cfg["input_size"]
    new_config["input_size"] = input_size

    # interpolation
    new_config["interpolation"] = "bicubic"
    if "interpolation" in args and args["interpolation"]:
        new_config["interpolation"] = args["interpolation"]
    elif "interpolation" in default_cfg:
        new_config["interpolation"] = default_cfg["interpolation"]

    # mean/std — model-name heuristics (reference config.py:84-101)
    model_name = args.get("model", "") or ""
    if "mean" in args and args["mean"] is not None:
        mean = tuple(args["mean"])
        if len(mean) == 1:
            mean = tuple(list(mean) * 3)
        new_config["mean"] = mean
    elif "inception" in model_name or "nasnet" in model_name:
        new_config["mean"] = IMAGENET_INCEPTION_MEAN
    elif "dpn" in model_name:
        new_config["mean"] = IMAGENET_DPN_MEAN
    elif "mean" in default_cfg:
        new_config["mean"] = default_cfg["mean"]
    else:
        new_config["mean"] = IMAGENET_DEFAULT_MEAN

    if "std" in args and args["std"] is not None:
        std = tuple(args["std"])
        if len(std) == 1:
            std = tuple(list(std) * 3)
        new_config["std"] = std
    elif "inception" in model_name or "nasnet" in model_name:
        new_config["std"] = IMAGENET_INCEPTION_STD
    elif "dpn" in model_name:
        new_config["std"] = IMAGENET_DPN_STD
    elif "std" in default_cfg:
        new_config["std"] = default_cfg["std"]
    else:
        new_config["std"] = IMAGENET_DEFAULT_STD

    # crop_pct
    crop_pct = DEFAULT_CROP_PCT
    if "crop_pct" in args and args["crop_pct"] is not None:
        crop_pct = args["crop_pct"]
    elif "crop_pct" in default_cfg:
        crop_pct = default_cfg["crop_pct"]
    new_config["crop_pct"] = crop_pct

    if verbose:
        _logger.info("Data processing configuration:")
        for n, v in new_config.items():
            _logger.info("\t%s: %s", n, str(v))

    return new_config
