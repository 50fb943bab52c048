"""Checkpoint load/resume helpers.

Capability parity with reference dfd/timm/models/helpers.py: `load_state_dict`
strips the `module.` DDP prefix (:8-28), `load_checkpoint` non-strict load
drops shape-mismatched keys (:31-44), `resume_checkpoint` restores model and
returns optimizer/amp state + next epoch (version>=2 -> epoch+1) (:47-73),
`load_pretrained` with input-conv collapse and classifier discard (:76-109).
"""

import logging
import os
from collections import OrderedDict

import torch

_logger = logging.getLogger(__name__)


def load_state_dict(checkpoint_path, use_ema=False):
    if checkpoint_path and os.path.isfile(checkpoint_path):
        checkpoint = torch.load(checkpoint_path, map_location="cpu", weights_only=False)
        state_dict_key = ""
        if isinstance(checkpoint, dict):
            if use_ema and "state_dict_ema" in checkpoint:
                state_dict_key = "state_dict_ema"
            elif "state_dict" in checkpoint:
                state_dict_key = "state_dict"
        if state_dict_key:
            new_state_dict = OrderedDict()
            for k, v in checkpoint[state_dict_key].items():
                name = k[7:] if k.startswith("module.") else k
                new_state_dict[name] = v
            state_dict = new_state_dict
        else:
            state_dict = checkpoint
        _logger.info("Loaded %s from checkpoint '%s'", state_dict_key or "weights", checkpoint_path)
        return state_dict
    _logger.error("No checkpoint found at '%s'", checkpoint_path)
    raise FileNotFoundError(checkpoint_path)


def load_checkpoint(model, checkpoint_path, use_ema=False, strict=True, ignore_keys=None):
    state_dict = load_state_dict(checkpoint_path, use_ema)
    if ignore_keys:
        state_dict = {k: v for k, v in state_dict.items() if k not in set(ignore_keys)}
        strict = False
    if not strict:
        # drop keys whose shapes mismatch the model (reference helpers.py:39-44)
        model_sd = model.state_dict()
        state_dict = {
            k: v for k, v in state_dict.items()
            if k in model_sd and model_sd[k].shape == v.shape
        }
    model.load_state_dict(state_dict, strict=strict)


def resume_checkpoint(model, checkpoint_path):
    """Restore model weights; return (other_state, resume_epoch) where
    other_state carries optimizer / amp / scaler entries for the trainer."""
    other_state = {}
    resume_epoch = None
    if not os.path.isfile(checkpoint_path):
        _logger.error("No checkpoint found at '%s'", checkpoint_path)
        raise FileNotFoundError(checkpoint_path)
    checkpoint = torch.load(checkpoint_path, map_location="cpu", weights_only=False)
    if isinstance(checkpoint, dict) and "state_dict" in checkpoint:
        new_state_dict = OrderedDict()
        for k, v in checkpoint["state_dict"].items():
            name = k[7:] if k.startswith("module.") else k
            new_state_dict[name] = v
        model.load_state_dict(new_state_dict)
        if "optimizer" in checkpoint:
            other_state["optimizer"] = checkpoint["optimizer"]
        if "amp" in checkpoint:
            other_state["amp"] = checkpoint["amp"]
        if "epoch" in checkpoint:
            resume_epoch = checkpoint["epoch"]
            if "version" in checkpoint and checkpoint["version"] > 1:
                resume_epoch += 1  # checkpoint saved at end of epoch
        _logger.info("Loaded checkpoint '%s' (epoch %s)", checkpoint_path, checkpoint.get("epoch"))
    else:
        model.load_state_dict(checkpoint)
        _logger.info("Loaded checkpoint '%s'", checkpoint_path)
    return other_state, resume_epoch


def load_pretrained(model, default_cfg, num_classes=1000, in_chans=3, filter_fn=None, strict=True):
    """Load pretrained weights from default_cfg['url'] or ['file'].

    Offline environment note: URL download is unavailable; a local 'file'
    entry works. Handles in_chans!=3 input-conv adaptation (1-chan sum
    collapse) and classifier discard for num_classes mismatch
    (reference helpers.py:76-109).
    """
    url = default_cfg.get("url", None)
    local = default_cfg.get("file", None)
    if local and os.path.isfile(local):
        state_dict = torch.load(local, map_location="cpu", weights_only=False)
    elif url:
        from torch.hub import load_state_dict_from_url

        state_dict = load_state_dict_from_url(url, map_location="cpu", progress=False)
    else:
        _logger.warning("Pretrained model URL/file is invalid, using random initialization.")
        return

    if filter_fn is not None:
        state_dict = filter_fn(state_dict)

    if in_chans == 1:
        conv1_name = default_cfg["first_conv"]
        conv1_weight = state_dict[conv1_name + ".weight"]
        state_dict[conv1_name + ".weight"] = conv1_weight.sum(dim=1, keepdim=True)
    elif in_chans != 3:
        conv1_name = default_cfg["first_conv"]
        del state_dict[conv1_name + ".weight"]
        strict = False

    classifier_name = default_cfg["classifier"]
    if num_classes != default_cfg.get("num_classes", 1000):
        state_dict.pop(classifier_name + ".weight", None)
        state_dict.pop(classifier_name + ".bias", None)
        strict = False

    model.load_state_dict(state_dict, strict=strict)
