"""Dual-Path Networks (reference dfd/timm/models/dpn.py, 323 LoC,
6 entrypoints)."""

from collections import OrderedDict

import torch
import torch.nn as nn
import torch.nn.functional as F

from .layers import SelectAdaptivePool2d
from .registry import register_model

__all__ = ["DPN"]


def _cfg(**kwargs):
    return {
        "url": "", "num_classes": 1000, "input_size": (3, 224, 224),
        "pool_size": (7, 7), "crop_pct": 0.875, "interpolation": "bicubic",
        "mean": (124 / 255, 117 / 255, 104 / 255),
        "std": tuple([1 / (0.0167 * 255)] * 3),
        "first_conv": "features.conv1_1.conv", "classifier": "classifier",
        **kwargs,
    }


default_cfgs = {k: _cfg() for k in [
    "dpn68", "dpn68b", "dpn92", "dpn98", "dpn131", "dpn107",
]}


class CatBnAct(nn.Module):
    def __init__(self, in_chs, activation_fn=nn.ReLU(inplace=True)):
        super().__init__()
        self.bn = nn.BatchNorm2d(in_chs, eps=0.001)
        self.act = activation_fn

    def forward(self, x):
        x = torch.cat(x, dim=1) if isinstance(x, tuple) else x
        return self.act(self.bn(x))


class BnActConv2d(nn.Module):
    def __init__(self, in_chs, out_chs, kernel_size, stride, padding=0, groups=1,
                 activation_fn=nn.ReLU(inplace=True)):
        super().__init__()
        self.bn = nn.BatchNorm2d(in_chs, eps=0.001)
        self.act = activation_fn
        self.conv = nn.Conv2d(in_chs, out_chs, kernel_size, stride, padding,
                              groups=groups, bias=False)

    def forward(self, x):
        return self.conv(self.act(self.bn(x)))


class InputBlock(nn.Module):
    def __init__(self, num_init_features, kernel_size=7, padding=3, in_chans=3,
                 activation_fn=nn.ReLU(inplace=True)):
        super().__init__()
        self.conv = nn.Conv2d(in_chans, num_init_features, kernel_size=kernel_size,
                              stride=2, padding=padding, bias=False)
        self.bn = nn.BatchNorm2d(num_init_features, eps=0.001)
        self.act = activation_fn
        self.pool = nn.MaxPool2d(kernel_size=3, stride=2, padding=1)

    def forward(self, x):
        x = self.conv(x)
        x = self.bn(x)
        x = self.act(x)
        return self.pool(x)


class DualPathBlock(nn.Module):
    """Dual-path unit: a residual lane of width num_1x1_c plus a dense lane
    growing by `inc` per block. State carries as the (residual, dense)
    tuple; module attribute names match the reference state dict
    (c1x1_w_s{1,2} shortcut, c1x1_a -> c3x3_b -> c1x1_c main path, with the
    'b' variant splitting the tail into c1x1_c1/c1x1_c2)."""

    def __init__(self, in_chs, num_1x1_a, num_3x3_b, num_1x1_c, inc, groups,
                 block_type="normal", b=False):
        super().__init__()
        assert block_type in ("proj", "down", "normal")
        self.num_1x1_c = num_1x1_c
        self.inc = inc
        self.b = b
        self.key_stride = 2 if block_type == "down" else 1
        self.has_proj = block_type != "normal"

        if self.has_proj:
            # the shortcut projection produces both lanes at once
            proj = BnActConv2d(in_chs=in_chs, out_chs=num_1x1_c + 2 * inc,
                               kernel_size=1, stride=self.key_stride)
            self.add_module("c1x1_w_s2" if self.key_stride == 2 else "c1x1_w_s1",
                            proj)
        self.c1x1_a = BnActConv2d(in_chs=in_chs, out_chs=num_1x1_a, kernel_size=1, stride=1)
        self.c3x3_b = BnActConv2d(
            in_chs=num_1x1_a, out_chs=num_3x3_b, kernel_size=3,
            stride=self.key_stride, padding=1, groups=groups)
        if b:
            self.c1x1_c = CatBnAct(in_chs=num_3x3_b)
            self.c1x1_c1 = nn.Conv2d(num_3x3_b, num_1x1_c, kernel_size=1, bias=False)
            self.c1x1_c2 = nn.Conv2d(num_3x3_b, inc, kernel_size=1, bias=False)
        else:
            self.c1x1_c = BnActConv2d(
                in_chs=num_3x3_b, out_chs=num_1x1_c + inc, kernel_size=1, stride=1)

    def _shortcut(self, joined, lanes):
        if not self.has_proj:
            return lanes  # identity: lanes pass through unchanged
        proj = getattr(self, "c1x1_w_s2" if self.key_stride == 2 else "c1x1_w_s1")
        both = proj(joined)
        return both[:, : self.num_1x1_c], both[:, self.num_1x1_c:]

    def forward(self, x):
        lanes = x if isinstance(x, tuple) else (x[:, : self.num_1x1_c], x[:, self.num_1x1_c:])
        joined = torch.cat(x, dim=1) if isinstance(x, tuple) else x
        short_res, short_dense = self._shortcut(joined, lanes)

        mid = self.c3x3_b(self.c1x1_a(joined))
        if self.b:
            mid = self.c1x1_c(mid)
            main_res, main_dense = self.c1x1_c1(mid), self.c1x1_c2(mid)
        else:
            tail = self.c1x1_c(mid)
            main_res, main_dense = tail[:, : self.num_1x1_c], tail[:, self.num_1x1_c:]
        return short_res + main_res, torch.cat([short_dense, main_dense], dim=1)


class DPN(nn.Module):
    def __init__(self, small=False, num_init_features=64, k_r=96, groups=32,
                 b=False, k_sec=(3, 4, 20, 3), inc_sec=(16, 32, 24, 128),
                 num_classes=1000, in_chans=3, drop_rate=0.0, global_pool="avg"):
        super().__init__()
        self.num_classes = num_classes
        self.drop_rate = drop_rate
        self.b = b
        bw_factor = 1 if small else 4

        blocks = OrderedDict()
        if small:
            blocks["conv1_1"] = InputBlock(num_init_features, 3, 1, in_chans)
        else:
            blocks["conv1_1"] = InputBlock(num_init_features, 7, 3, in_chans)

        bw = 64 * bw_factor
        inc = inc_sec[0]
        r = (k_r * bw) // (64 * bw_factor)
        blocks["conv2_1"] = DualPathBlock(num_init_features, r, r, bw, inc, groups, "proj", b)
        in_chs = bw + 3 * inc
        for i in range(2, k_sec[0] + 1):
            blocks["conv2_" + str(i)] = DualPathBlock(in_chs, r, r, bw, inc, groups, "normal", b)
            in_chs += inc

        bw = 128 * bw_factor
        inc = inc_sec[1]
        r = (k_r * bw) // (64 * bw_factor)
        blocks["conv3_1"] = DualPathBlock(in_chs, r, r, bw, inc, groups, "down", b)
        in_chs = bw + 3 * inc
        for i in range(2, k_sec[1] + 1):
            blocks["conv3_" + str(i)] = DualPathBlock(in_chs, r, r, bw, inc, groups, "normal", b)
            in_chs += inc

        bw = 256 * bw_factor
        inc = inc_sec[2]
        r = (k_r * bw) // (64 * bw_factor)
        blocks["conv4_1"] = DualPathBlock(in_chs, r, r, bw, inc, groups, "down", b)
        in_chs = bw + 3 * inc
        for i in range(2, k_sec[2] + 1):
            blocks["conv4_" + str(i)] = DualPathBlock(in_chs, r, r, bw, inc, groups, "normal", b)
            in_chs += inc

        bw = 512 * bw_factor
        inc = inc_sec[3]
        r = (k_r * bw) // (64 * bw_factor)
        blocks["conv5_1"] = DualPathBlock(in_chs, r, r, bw, inc, groups, "down", b)
        in_chs = bw + 3 * inc
        for i in range(2, k_sec[3] + 1):
            blocks["conv5_" + str(i)] = DualPathBlock(in_chs, r, r, bw, inc, groups, "normal", b)
            in_chs += inc
        blocks["conv5_bn_ac"] = CatBnAct(in_chs)

        self.num_features = in_chs
        self.features = nn.Sequential(blocks)
        self.global_pool = SelectAdaptivePool2d(pool_type=global_pool)
        self.classifier = nn.Conv2d(self.num_features, num_classes, kernel_size=1, bias=True)

    def get_classifier(self):
        return self.classifier

    def reset_classifier(self, num_classes, global_pool="avg"):
        self.num_classes = num_classes
        self.global_pool = SelectAdaptivePool2d(pool_type=global_pool)
        self.classifier = nn.Conv2d(
            self.num_features, num_classes, kernel_size=1, bias=True) if num_classes else None

    def forward_features(self, x):
        return self.features(x)

    def forward(self, x):
        x = self.forward_features(x)
        x = self.global_pool(x)
        if self.drop_rate > 0.0:
            x = F.dropout(x, p=self.drop_rate, training=self.training)
        out = self.classifier(x)
        return out.flatten(1)


def _make(variant, pretrained=False, **kwargs):
    model = DPN(**kwargs)
    model.default_cfg = default_cfgs[variant]
    return model


@register_model
def dpn68(pretrained=False, **kwargs):
    return _make("dpn68", pretrained, small=True, num_init_features=10, k_r=128,
                 groups=32, b=False, k_sec=(3, 4, 12, 3), inc_sec=(16, 32, 32, 64), **kwargs)


@register_model
def dpn68b(pretrained=False, **kwargs):
    return _make("dpn68b", pretrained, small=True, num_init_features=10, k_r=128,
                 groups=32, b=True, k_sec=(3, 4, 12, 3), inc_sec=(16, 32, 32, 64), **kwargs)


@register_model
def dpn92(pretrained=False, **kwargs):
    return _make("dpn92", pretrained, num_init_features=64, k_r=96, groups=32,
                 k_sec=(3, 4, 20, 3), inc_sec=(16, 32, 24, 128), **kwargs)


@register_model
def dpn98(pretrained=False, **kwargs):
    return _make("dpn98", pretrained, num_init_features=96, k_r=160, groups=40,
                 k_sec=(3, 6, 20, 3), inc_sec=(16, 32, 32, 128), **kwargs)


@register_model
def dpn131(pretrained=False, **kwargs):
    return _make("dpn131", pretrained, num_init_features=128, k_r=160, groups=40,
                 k_sec=(4, 8, 28, 3), inc_sec=(16, 32, 32, 128), **kwargs)


@register_model
def dpn107(pretrained=False, **kwargs):
    return _make("dpn107", pretrained, num_init_features=128, k_r=200, groups=50,
                 k_sec=(4, 8, 20, 3), inc_sec=(20, 64, 64, 128), **kwargs)
