"""EfficientNet-family building blocks (hot path).

Capability parity with reference dfd/timm/models/efficientnet_blocks.py:
SqueezeExcite (:93-110), ConvBnAct (:113), DepthwiseSeparableConv (:136-194),
InvertedResidual (:260-348), CondConvResidual (:431), EdgeResidual (:484),
channel rounding helpers (:55-69) and TF BN defaults (:13-15).

MI355X-first design: the BN->act pair and the SE chain route through
deepfake_detection_amd.ops.functional, which on ROCm runs fused NHWC HIP
kernels (BN+SiLU fwd/bwd with fp32 stats; pooled SE chain) instead of the
reference's separate cuDNN BN / jit-scripted Swish kernels
(reference activations.py:19-48). State-dict key names are byte-identical to
the reference (conv_pw/bn1/conv_dw/bn2/se.conv_reduce/se.conv_expand/
conv_pwl/bn3 ...), so `.pth.tar` checkpoints interchange.
"""

import torch
import torch.nn as nn

from ..ops import functional as O
from .layers import (DepthwiseConv2d, PointwiseConv2d, StemConv2d,
                     create_conv2d, drop_path, sigmoid)

__all__ = [
    "BN_MOMENTUM_TF_DEFAULT",
    "BN_EPS_TF_DEFAULT",
    "get_bn_args_tf",
    "resolve_bn_args",
    "resolve_se_args",
    "resolve_act_layer",
    "make_divisible",
    "round_channels",
    "SqueezeExcite",
    "ConvBnAct",
    "DepthwiseSeparableConv",
    "InvertedResidual",
    "CondConvResidual",
    "EdgeResidual",
]

# TF-reference BatchNorm defaults (reference efficientnet_blocks.py:13-15)
BN_MOMENTUM_TF_DEFAULT = 1 - 0.99
BN_EPS_TF_DEFAULT = 1e-3
_BN_ARGS_TF = dict(momentum=BN_MOMENTUM_TF_DEFAULT, eps=BN_EPS_TF_DEFAULT)


def get_bn_args_tf():
    return _BN_ARGS_TF.copy()


def resolve_bn_args(kwargs):
    bn_args = get_bn_args_tf() if kwargs.pop("bn_tf", False) else {}
    bn_momentum = kwargs.pop("bn_momentum", None)
    if bn_momentum is not None:
        bn_args["momentum"] = bn_momentum
    bn_eps = kwargs.pop("bn_eps", None)
    if bn_eps is not None:
        bn_args["eps"] = bn_eps
    return bn_args


# SE defaults for the EfficientNet family (reference efficientnet_blocks.py:33-37):
# gate = sigmoid, act = containing block's act, reduction base = block in_chs,
# divisor = 1.
_SE_ARGS_DEFAULT = dict(gate_fn=sigmoid, act_layer=None, reduce_mid=False, divisor=1)


def resolve_se_args(kwargs, in_chs, act_layer=None):
    se_kwargs = kwargs.copy() if kwargs is not None else {}
    for k, v in _SE_ARGS_DEFAULT.items():
        se_kwargs.setdefault(k, v)
    if not se_kwargs.pop("reduce_mid"):
        se_kwargs["reduced_base_chs"] = in_chs
    if se_kwargs["act_layer"] is None:
        assert act_layer is not None
        se_kwargs["act_layer"] = act_layer
    return se_kwargs


def resolve_act_layer(kwargs, default="relu"):
    from .layers import HardSwish, Swish

    act_layer = kwargs.pop("act_layer", default)
    if isinstance(act_layer, str):
        act_layer = {
            "relu": nn.ReLU,
            "relu6": nn.ReLU6,
            "swish": Swish,
            "hard_swish": HardSwish,
        }[act_layer]
    return act_layer


def make_divisible(v, divisor=8, min_value=None):
    """Round v to the nearest multiple of divisor, never below 90% of v
    (TF MobileNet rule; reference efficientnet_blocks.py:55-62)."""
    min_value = min_value or divisor
    new_v = max(min_value, int(v + divisor / 2) // divisor * divisor)
    if new_v < 0.9 * v:
        new_v += divisor
    return new_v


def round_channels(channels, multiplier=1.0, divisor=8, channel_min=None):
    if not multiplier:
        return channels
    channels *= multiplier
    return make_divisible(channels, divisor, channel_min)


class SqueezeExcite(nn.Module):
    """SE: global-avg-pool -> 1x1 reduce -> act -> 1x1 expand -> sigmoid gate
    (reference efficientnet_blocks.py:93-110). Executed as one fused HIP
    chain on GPU via ops.functional.se."""

    def __init__(self, in_chs, se_ratio=0.25, reduced_base_chs=None,
                 act_layer=nn.ReLU, gate_fn=sigmoid, divisor=1, **_):
        super().__init__()
        self.gate_fn = gate_fn
        reduced_chs = make_divisible((reduced_base_chs or in_chs) * se_ratio, divisor)
        self.conv_reduce = nn.Conv2d(in_chs, reduced_chs, 1, bias=True)
        self.act1 = act_layer(inplace=True)
        self.conv_expand = nn.Conv2d(reduced_chs, in_chs, 1, bias=True)

    def forward(self, x):
        if self.gate_fn is sigmoid:
            return O.se(x, self.conv_reduce, self.act1, self.conv_expand)
        x_se = x.mean(dim=(2, 3), keepdim=True)
        x_se = self.conv_reduce(x_se)
        x_se = self.act1(x_se)
        x_se = self.conv_expand(x_se)
        return x * self.gate_fn(x_se)


class ConvBnAct(nn.Module):
    def __init__(self, in_chs, out_chs, kernel_size,
                 stride=1, dilation=1, pad_type="", act_layer=nn.ReLU,
                 norm_layer=nn.BatchNorm2d, norm_kwargs=None):
        super().__init__()
        norm_kwargs = norm_kwargs or {}
        self.conv = create_conv2d(in_chs, out_chs, kernel_size, stride=stride,
                                  dilation=dilation, padding=pad_type)
        self.bn1 = norm_layer(out_chs, **norm_kwargs)
        self.act1 = act_layer(inplace=True)
        self._act_name = O.act_name_of(self.act1)
        _mark_bn_producer(self.conv)

    def feature_module(self, location):
        return "act1"

    def feature_channels(self, location):
        return self.conv.out_channels

    def forward(self, x):
        x = self.conv(x)
        if self._act_name != "other" and O.fusable_bn(self.bn1):
            return O.bn_act(x, self.bn1, self._act_name)
        return self.act1(self.bn1(x))


def _mark_bn_producer(conv):
    """Ask a conv whose output feeds a fused BatchNorm to emit the BN stats
    from its epilogue — the BN then skips its own stats pass over y."""
    if isinstance(conv, (PointwiseConv2d, DepthwiseConv2d, StemConv2d)):
        conv.emit_bn_stats = True


def _bn_act(bn, act_module, act_name, x, residual=None, drop_path_mask=None):
    if act_name != "other" and O.fusable_bn(bn):
        return O.bn_act(x, bn, act_name, residual, drop_path_mask)
    y = act_module(bn(x))
    if drop_path_mask is not None:
        y = y * drop_path_mask.to(y.dtype).view(-1, 1, 1, 1)
    return y if residual is None else y + residual


def _drop_path_mask(x, drop_prob, training):
    """Per-sample stochastic-depth keep mask (0 or 1/keep_prob), drawn with
    the reference's floor(keep + U[0,1)) binarization (reference
    drop.py:84-100); fused into the block-tail BN kernel."""
    if drop_prob <= 0.0 or not training:
        return None
    keep = 1.0 - drop_prob
    mask = torch.floor(keep + torch.rand(x.shape[0], device=x.device,
                                         dtype=torch.float32))
    return mask.div_(keep)


class DepthwiseSeparableConv(nn.Module):
    """DS conv: dw kxk -> BN+act -> (SE) -> pw 1x1 -> BN (+act if pw_act)
    (reference efficientnet_blocks.py:136-194)."""

    def __init__(self, in_chs, out_chs, dw_kernel_size=3,
                 stride=1, dilation=1, pad_type="", act_layer=nn.ReLU, noskip=False,
                 pw_kernel_size=1, pw_act=False, se_ratio=0.0, se_kwargs=None,
                 norm_layer=nn.BatchNorm2d, norm_kwargs=None, drop_path_rate=0.0):
        super().__init__()
        norm_kwargs = norm_kwargs or {}
        has_se = se_ratio is not None and se_ratio > 0.0
        self.has_residual = (stride == 1 and in_chs == out_chs) and not noskip
        self.has_pw_act = pw_act
        self.drop_path_rate = drop_path_rate

        self.conv_dw = create_conv2d(
            in_chs, in_chs, dw_kernel_size, stride=stride, dilation=dilation,
            padding=pad_type, depthwise=True)
        _mark_bn_producer(self.conv_dw)
        self.bn1 = norm_layer(in_chs, **norm_kwargs)
        self.act1 = act_layer(inplace=True)

        if has_se:
            se_kwargs = resolve_se_args(se_kwargs, in_chs, act_layer)
            self.se = SqueezeExcite(in_chs, se_ratio=se_ratio, **se_kwargs)
        else:
            self.se = None

        self.conv_pw = create_conv2d(in_chs, out_chs, pw_kernel_size, padding=pad_type)
        _mark_bn_producer(self.conv_pw)
        self.bn2 = norm_layer(out_chs, **norm_kwargs)
        self.act2 = act_layer(inplace=True) if self.has_pw_act else nn.Identity()
        self._act_name = O.act_name_of(self.act1)
        self._act2_name = O.act_name_of(self.act2)

    def feature_module(self, location):
        return "conv_pw"

    def feature_channels(self, location):
        return self.conv_pw.in_channels

    def forward(self, x):
        residual = x
        x = self.conv_dw(x)
        x = _bn_act(self.bn1, self.act1, self._act_name, x)
        if self.se is not None:
            x = self.se(x)
        x = self.conv_pw(x)
        if self.has_residual and self._act2_name != "other" and x.is_cuda:
            # fused BN(+act) + drop_path scale + residual in one pass
            dp = _drop_path_mask(x, self.drop_path_rate, self.training)
            x = _bn_act(self.bn2, self.act2, self._act2_name, x, residual, dp)
        else:
            x = _bn_act(self.bn2, self.act2, self._act2_name, x)
            if self.has_residual:
                if self.drop_path_rate > 0.0:
                    x = drop_path(x, self.drop_path_rate, self.training)
                x = x + residual
        return x


class InvertedResidual(nn.Module):
    """MBConv: pw-expand -> BN+act -> dw -> BN+act -> (SE) -> pw-linear -> BN
    (+residual w/ drop_path) (reference efficientnet_blocks.py:260-348)."""

    def __init__(self, in_chs, out_chs, dw_kernel_size=3,
                 stride=1, dilation=1, pad_type="", act_layer=nn.ReLU, noskip=False,
                 exp_ratio=1.0, exp_kernel_size=1, pw_kernel_size=1,
                 se_ratio=0.0, se_kwargs=None, norm_layer=nn.BatchNorm2d, norm_kwargs=None,
                 conv_kwargs=None, drop_path_rate=0.0):
        super().__init__()
        norm_kwargs = norm_kwargs or {}
        conv_kwargs = conv_kwargs or {}
        mid_chs = make_divisible(in_chs * exp_ratio)
        has_se = se_ratio is not None and se_ratio > 0.0
        self.has_residual = (in_chs == out_chs and stride == 1) and not noskip
        self.drop_path_rate = drop_path_rate

        self.conv_pw = create_conv2d(in_chs, mid_chs, exp_kernel_size, padding=pad_type, **conv_kwargs)
        _mark_bn_producer(self.conv_pw)
        self.bn1 = norm_layer(mid_chs, **norm_kwargs)
        self.act1 = act_layer(inplace=True)

        self.conv_dw = create_conv2d(
            mid_chs, mid_chs, dw_kernel_size, stride=stride, dilation=dilation,
            padding=pad_type, depthwise=True, **conv_kwargs)
        _mark_bn_producer(self.conv_dw)
        self.bn2 = norm_layer(mid_chs, **norm_kwargs)
        self.act2 = act_layer(inplace=True)

        if has_se:
            se_kwargs = resolve_se_args(se_kwargs, in_chs, act_layer)
            self.se = SqueezeExcite(mid_chs, se_ratio=se_ratio, **se_kwargs)
        else:
            self.se = None

        self.conv_pwl = create_conv2d(mid_chs, out_chs, pw_kernel_size, padding=pad_type, **conv_kwargs)
        _mark_bn_producer(self.conv_pwl)
        self.bn3 = norm_layer(out_chs, **norm_kwargs)
        self._act_name = O.act_name_of(self.act1)

    def feature_module(self, location):
        if location == "post_exp":
            return "act1"
        return "conv_pwl"

    def feature_channels(self, location):
        if location == "post_exp":
            return self.conv_pw.out_channels
        return self.conv_pwl.in_channels

    def forward(self, x):
        residual = x
        x = self.conv_pw(x)
        x = _bn_act(self.bn1, self.act1, self._act_name, x)
        x = self.conv_dw(x)
        x = _bn_act(self.bn2, self.act2, self._act_name, x)
        if self.se is not None:
            x = self.se(x)
        x = self.conv_pwl(x)
        if self.has_residual and x.is_cuda:
            # fused BN + drop_path scale + residual add in one pass
            dp = _drop_path_mask(x, self.drop_path_rate, self.training)
            x = _bn_act(self.bn3, nn.Identity(), "none", x, residual, dp)
        else:
            x = _bn_act(self.bn3, nn.Identity(), "none", x)
            if self.has_residual:
                if self.drop_path_rate > 0.0:
                    x = drop_path(x, self.drop_path_rate, self.training)
                x = x + residual
        return x


class CondConvResidual(InvertedResidual):
    """Inverted residual with CondConv per-sample expert routing
    (reference efficientnet_blocks.py:431)."""

    def __init__(self, in_chs, out_chs, dw_kernel_size=3,
                 stride=1, dilation=1, pad_type="", act_layer=nn.ReLU, noskip=False,
                 exp_ratio=1.0, exp_kernel_size=1, pw_kernel_size=1,
                 se_ratio=0.0, se_kwargs=None, norm_layer=nn.BatchNorm2d, norm_kwargs=None,
                 num_experts=0, drop_path_rate=0.0):
        self.num_experts = num_experts
        conv_kwargs = dict(num_experts=self.num_experts)
        super().__init__(
            in_chs, out_chs, dw_kernel_size=dw_kernel_size, stride=stride,
            dilation=dilation, pad_type=pad_type, act_layer=act_layer, noskip=noskip,
            exp_ratio=exp_ratio, exp_kernel_size=exp_kernel_size,
            pw_kernel_size=pw_kernel_size, se_ratio=se_ratio, se_kwargs=se_kwargs,
            norm_layer=norm_layer, norm_kwargs=norm_kwargs, conv_kwargs=conv_kwargs,
            drop_path_rate=drop_path_rate)
        self.routing_fn = nn.Linear(in_chs, self.num_experts)

    def forward(self, x):
        residual = x
        pooled_inputs = x.mean(dim=(2, 3))
        routing_weights = torch.sigmoid(self.routing_fn(pooled_inputs))
        x = self.conv_pw(x, routing_weights)
        x = _bn_act(self.bn1, self.act1, self._act_name, x)
        x = self.conv_dw(x, routing_weights)
        x = _bn_act(self.bn2, self.act2, self._act_name, x)
        if self.se is not None:
            x = self.se(x)
        x = self.conv_pwl(x, routing_weights)
        if self.has_residual and x.is_cuda:
            # fused BN + drop_path scale + residual add in one pass
            dp = _drop_path_mask(x, self.drop_path_rate, self.training)
            x = _bn_act(self.bn3, nn.Identity(), "none", x, residual, dp)
        else:
            x = _bn_act(self.bn3, nn.Identity(), "none", x)
            if self.has_residual:
                if self.drop_path_rate > 0.0:
                    x = drop_path(x, self.drop_path_rate, self.training)
                x = x + residual
        return x


class EdgeResidual(nn.Module):
    """EdgeTPU residual: expansion kxk conv -> BN+act -> (SE) -> pw-linear -> BN
    (reference efficientnet_blocks.py:484)."""

    def __init__(self, in_chs, out_chs, exp_kernel_size=3, exp_ratio=1.0,
                 fake_in_chs=0, stride=1, dilation=1, pad_type="", act_layer=nn.ReLU,
                 noskip=False, pw_kernel_size=1, se_ratio=0.0, se_kwargs=None,
                 norm_layer=nn.BatchNorm2d, norm_kwargs=None, drop_path_rate=0.0):
        super().__init__()
        norm_kwargs = norm_kwargs or {}
        if fake_in_chs > 0:
            mid_chs = make_divisible(fake_in_chs * exp_ratio)
        else:
            mid_chs = make_divisible(in_chs * exp_ratio)
        has_se = se_ratio is not None and se_ratio > 0.0
        self.has_residual = (in_chs == out_chs and stride == 1) and not noskip
        self.drop_path_rate = drop_path_rate

        self.conv_exp = create_conv2d(in_chs, mid_chs, exp_kernel_size, padding=pad_type)
        self.bn1 = norm_layer(mid_chs, **norm_kwargs)
        self.act1 = act_layer(inplace=True)

        if has_se:
            se_kwargs = resolve_se_args(se_kwargs, in_chs, act_layer)
            self.se = SqueezeExcite(mid_chs, se_ratio=se_ratio, **se_kwargs)
        else:
            self.se = None

        self.conv_pwl = create_conv2d(
            mid_chs, out_chs, pw_kernel_size, stride=stride, dilation=dilation, padding=pad_type)
        _mark_bn_producer(self.conv_pwl)
        self.bn2 = norm_layer(out_chs, **norm_kwargs)
        self._act_name = O.act_name_of(self.act1)

    def feature_module(self, location):
        if location == "post_exp":
            return "act1"
        return "conv_pwl"

    def feature_channels(self, location):
        if location == "post_exp":
            return self.conv_exp.out_channels
        return self.conv_pwl.in_channels

    def forward(self, x):
        residual = x
        x = self.conv_exp(x)
        x = _bn_act(self.bn1, self.act1, self._act_name, x)
        if self.se is not None:
            x = self.se(x)
        x = self.conv_pwl(x)
        x = _bn_act(self.bn2, nn.Identity(), "none", x)
        if self.has_residual:
            if self.drop_path_rate > 0.0:
                x = drop_path(x, self.drop_path_rate, self.training)
            x = x + residual
        return x
