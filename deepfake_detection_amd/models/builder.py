"""Arch-string decoder + stage builder for the EfficientNet family.

Capability parity with reference dfd/timm/models/efficientnet_builder.py:
`_decode_block_str` (:20-138), `_scale_stage_depth` ceil depth scaling
(:139), `decode_arch_def` (:180), `EfficientNetBuilder` (:194-362) with
per-block linearly-scaled drop_path and stride->dilation fixup, and the
TF-style fan-out init `_init_weight_goog` (:537-575).
"""

import logging
import math
import re
from copy import deepcopy

import torch.nn as nn

from .blocks import (
    ConvBnAct,
    CondConvResidual,
    DepthwiseSeparableConv,
    EdgeResidual,
    InvertedResidual,
    round_channels,
)
from .layers import CondConv2d, HardSwish, Swish, get_condconv_initializer

__all__ = ["decode_arch_def", "EfficientNetBuilder", "efficientnet_init_weights"]

_logger = logging.getLogger(__name__)


def _parse_ksize(ss):
    if ss.isdigit():
        return int(ss)
    return [int(k) for k in ss.split(".")]


def _decode_block_str(block_str):
    """Decode one block string, e.g. 'ir_r2_k3_s2_e6_c24_se0.25'.

    Leading token = block type (ir / ds / dsa / er / cn); options: r repeats,
    k kernel, s stride, e expansion, c out_chs, se ratio, n act fn, a/p
    exp/pw kernel size, cc experts, noskip.
    """
    assert isinstance(block_str, str)
    ops = block_str.split("_")
    block_type = ops[0]
    ops = ops[1:]
    options = {}
    noskip = False
    for op in ops:
        if op == "noskip":
            noskip = True
        elif op.startswith("n"):
            key = op[0]
            v = op[1:]
            if v == "re":
                value = nn.ReLU
            elif v == "r6":
                value = nn.ReLU6
            elif v == "hs":
                value = HardSwish
            elif v == "sw":
                value = Swish
            else:
                continue
            options[key] = value
        else:
            splits = re.split(r"(\d.*)", op)
            if len(splits) >= 2:
                key, value = splits[:2]
                options[key] = value

    act_layer = options["n"] if "n" in options else None
    exp_kernel_size = _parse_ksize(options["a"]) if "a" in options else 1
    pw_kernel_size = _parse_ksize(options["p"]) if "p" in options else 1
    fake_in_chs = int(options["fc"]) if "fc" in options else 0

    num_repeat = int(options["r"])
    if block_type == "ir":
        block_args = dict(
            block_type=block_type,
            dw_kernel_size=_parse_ksize(options["k"]),
            exp_kernel_size=exp_kernel_size,
            pw_kernel_size=pw_kernel_size,
            out_chs=int(options["c"]),
            exp_ratio=float(options["e"]),
            se_ratio=float(options["se"]) if "se" in options else None,
            stride=int(options["s"]),
            act_layer=act_layer,
            noskip=noskip,
        )
        if "cc" in options:
            block_args["num_experts"] = int(options["cc"])
    elif block_type in ("ds", "dsa"):
        block_args = dict(
            block_type=block_type,
            dw_kernel_size=_parse_ksize(options["k"]),
            pw_kernel_size=pw_kernel_size,
            out_chs=int(options["c"]),
            se_ratio=float(options["se"]) if "se" in options else None,
            stride=int(options["s"]),
            act_layer=act_layer,
            pw_act=block_type == "dsa",
            noskip=block_type == "dsa" or noskip,
        )
    elif block_type == "er":
        block_args = dict(
            block_type=block_type,
            exp_kernel_size=_parse_ksize(options["k"]),
            pw_kernel_size=pw_kernel_size,
            out_chs=int(options["c"]),
            exp_ratio=float(options["e"]),
            fake_in_chs=fake_in_chs,
            se_ratio=float(options["se"]) if "se" in options else None,
            stride=int(options["s"]),
            act_layer=act_layer,
            noskip=noskip,
        )
    elif block_type == "cn":
        block_args = dict(
            block_type=block_type,
            kernel_size=int(options["k"]),
            out_chs=int(options["c"]),
            stride=int(options["s"]),
            act_layer=act_layer,
        )
    else:
        raise AssertionError("Unknown block type (%s)" % block_type)

    return block_args, num_repeat


def _scale_stage_depth(stack_args, repeats, depth_multiplier=1.0, depth_trunc="ceil"):
    """EfficientNet depth scaling: scale the stage's total repeat count
    (ceil), distributing in reverse so the first block scales last."""
    num_repeat = sum(repeats)
    if depth_trunc == "round":
        num_repeat_scaled = max(1, round(num_repeat * depth_multiplier))
    else:
        num_repeat_scaled = int(math.ceil(num_repeat * depth_multiplier))

    # distribute the scaled budget back-to-front (exact-parity algorithm:
    # the LAST block definitions absorb rounding first, so the stage's first
    # block scales last)
    remaining, budget = num_repeat, num_repeat_scaled
    scaled = {}
    for idx in range(len(repeats) - 1, -1, -1):
        share = max(1, round(repeats[idx] / remaining * budget))
        scaled[idx] = share
        remaining -= repeats[idx]
        budget -= share

    expanded = []
    for idx, ba in enumerate(stack_args):
        expanded.extend(deepcopy(ba) for _ in range(scaled[idx]))
    return expanded


def decode_arch_def(arch_def, depth_multiplier=1.0, depth_trunc="ceil", experts_multiplier=1):
    arch_args = []
    for stack_idx, block_strings in enumerate(arch_def):
        assert isinstance(block_strings, list)
        stack_args = []
        repeats = []
        for block_str in block_strings:
            assert isinstance(block_str, str)
            ba, rep = _decode_block_str(block_str)
            if ba.get("num_experts", 0) > 0 and experts_multiplier > 1:
                ba["num_experts"] *= experts_multiplier
            stack_args.append(ba)
            repeats.append(rep)
        arch_args.append(_scale_stage_depth(stack_args, repeats, depth_multiplier, depth_trunc))
    return arch_args


class EfficientNetBuilder:
    """Builds the block stages from decoded arch args.

    Per-block drop_path scales linearly with global block index; only the
    first block of a stage keeps stride>1; stride converts to dilation past
    the requested output_stride.
    """

    def __init__(self, channel_multiplier=1.0, channel_divisor=8, channel_min=None,
                 output_stride=32, pad_type="", act_layer=None, se_kwargs=None,
                 norm_layer=nn.BatchNorm2d, norm_kwargs=None, drop_path_rate=0.0,
                 feature_location="", verbose=False):
        self.channel_multiplier = channel_multiplier
        self.channel_divisor = channel_divisor
        self.channel_min = channel_min
        self.output_stride = output_stride
        self.pad_type = pad_type
        self.act_layer = act_layer
        self.se_kwargs = se_kwargs
        self.norm_layer = norm_layer
        self.norm_kwargs = norm_kwargs
        self.drop_path_rate = drop_path_rate
        self.feature_location = feature_location
        assert feature_location in ("pre_pwl", "post_exp", "")
        self.verbose = verbose

        self.in_chs = None
        self.features = {}

    def _round_channels(self, chs):
        return round_channels(chs, self.channel_multiplier, self.channel_divisor, self.channel_min)

    def _make_block(self, ba, block_idx, block_count):
        drop_path_rate = self.drop_path_rate * block_idx / block_count
        bt = ba.pop("block_type")
        ba["in_chs"] = self.in_chs
        ba["out_chs"] = self._round_channels(ba["out_chs"])
        if "fake_in_chs" in ba and ba["fake_in_chs"]:
            ba["fake_in_chs"] = self._round_channels(ba["fake_in_chs"])
        ba["norm_layer"] = self.norm_layer
        ba["norm_kwargs"] = self.norm_kwargs
        ba["pad_type"] = self.pad_type
        ba["act_layer"] = ba["act_layer"] if ba["act_layer"] is not None else self.act_layer
        assert ba["act_layer"] is not None
        if bt == "ir":
            ba["drop_path_rate"] = drop_path_rate
            ba["se_kwargs"] = self.se_kwargs
            if ba.get("num_experts", 0) > 0:
                block = CondConvResidual(**ba)
            else:
                block = InvertedResidual(**ba)
        elif bt in ("ds", "dsa"):
            ba["drop_path_rate"] = drop_path_rate
            ba["se_kwargs"] = self.se_kwargs
            block = DepthwiseSeparableConv(**ba)
        elif bt == "er":
            ba["drop_path_rate"] = drop_path_rate
            ba["se_kwargs"] = self.se_kwargs
            block = EdgeResidual(**ba)
        elif bt == "cn":
            block = ConvBnAct(**ba)
        else:
            raise AssertionError("Unknown block type (%s) while building model." % bt)
        self.in_chs = ba["out_chs"]
        return block

    def __call__(self, in_chs, model_block_args):
        self.in_chs = in_chs
        total_block_count = sum(len(x) for x in model_block_args)
        total_block_idx = 0
        current_stride = 2
        current_dilation = 1
        feature_idx = 0
        stages = []
        for stage_idx, stage_block_args in enumerate(model_block_args):
            last_stack = stage_idx == (len(model_block_args) - 1)
            assert isinstance(stage_block_args, list)

            blocks = []
            for block_idx, block_args in enumerate(stage_block_args):
                last_block = block_idx == (len(stage_block_args) - 1)
                extract_features = ""

                assert block_args["stride"] in (1, 2)
                if block_idx >= 1:
                    block_args["stride"] = 1

                do_extract = False
                if self.feature_location == "pre_pwl":
                    if last_block:
                        next_stage_idx = stage_idx + 1
                        if next_stage_idx >= len(model_block_args):
                            do_extract = True
                        else:
                            do_extract = model_block_args[next_stage_idx][0]["stride"] > 1
                elif self.feature_location == "post_exp":
                    if block_args["stride"] > 1 or (last_stack and last_block):
                        do_extract = True
                if do_extract:
                    extract_features = self.feature_location

                next_dilation = current_dilation
                if block_args["stride"] > 1:
                    next_output_stride = current_stride * block_args["stride"]
                    if next_output_stride > self.output_stride:
                        next_dilation = current_dilation * block_args["stride"]
                        block_args["stride"] = 1
                    else:
                        current_stride = next_output_stride
                block_args["dilation"] = current_dilation
                if next_dilation != current_dilation:
                    current_dilation = next_dilation

                block = self._make_block(block_args, total_block_idx, total_block_count)
                blocks.append(block)

                if extract_features:
                    feature_module = block.feature_module(extract_features)
                    if feature_module:
                        feature_module = "blocks.{}.{}.".format(stage_idx, block_idx) + feature_module
                    feature_channels = block.feature_channels(extract_features)
                    self.features[feature_idx] = dict(name=feature_module, num_chs=feature_channels)
                    feature_idx += 1

                total_block_idx += 1
            stages.append(nn.Sequential(*blocks))
        return stages


def _conv_fan_out(conv, fix_group_fanout):
    fan = conv.kernel_size[0] * conv.kernel_size[1] * conv.out_channels
    return fan // conv.groups if fix_group_fanout else fan


def _init_weight_goog(m, n="", fix_group_fanout=True):
    """TF-official fan-out normal init; exact semantic parity with reference
    efficientnet_builder.py:537-575 (normal(0, sqrt(2/fan_out)) convs,
    unit-BN, uniform classifier with routing_fn fan-in special case)."""
    if isinstance(m, CondConv2d):
        std = math.sqrt(2.0 / _conv_fan_out(m, fix_group_fanout))
        expert_init = get_condconv_initializer(
            lambda w: nn.init.normal_(w, 0, std), m.num_experts, m.weight_shape)
        expert_init(m.weight)
        if m.bias is not None:
            nn.init.zeros_(m.bias)
    elif isinstance(m, nn.Conv2d):
        std = math.sqrt(2.0 / _conv_fan_out(m, fix_group_fanout))
        nn.init.normal_(m.weight, 0, std)
        if m.bias is not None:
            nn.init.zeros_(m.bias)
    elif isinstance(m, nn.BatchNorm2d):
        nn.init.ones_(m.weight)
        nn.init.zeros_(m.bias)
    elif isinstance(m, nn.Linear):
        fan_out = m.weight.size(0)
        fan_in = m.weight.size(1) if "routing_fn" in n else 0
        bound = 1.0 / math.sqrt(fan_in + fan_out)
        nn.init.uniform_(m.weight, -bound, bound)
        nn.init.zeros_(m.bias)


def efficientnet_init_weights(model: nn.Module, init_fn=None):
    init_fn = init_fn or _init_weight_goog
    for n, m in model.named_modules():
        init_fn(m, n)
