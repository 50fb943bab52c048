"""Res2Net (reference dfd/timm/models/res2net.py, 236 LoC, 7 entrypoints) —
multi-scale bottleneck on the ResNet trunk."""

import math

import torch
import torch.nn as nn

from .registry import register_model
from .resnet import ResNet


def _cfg(**kwargs):
    return {
        "url": "", "num_classes": 1000, "input_size": (3, 224, 224),
        "pool_size": (7, 7), "crop_pct": 0.875, "interpolation": "bilinear",
        "mean": (0.485, 0.456, 0.406), "std": (0.229, 0.224, 0.225),
        "first_conv": "conv1", "classifier": "fc", **kwargs,
    }


default_cfgs = {k: _cfg() for k in [
    "res2net50_26w_4s", "res2net50_48w_2s", "res2net50_14w_8s",
    "res2net50_26w_6s", "res2net50_26w_8s", "res2net101_26w_4s", "res2next50",
]}


class Bottle2neck(nn.Module):
    """Res2Net bottleneck: the 3x3 stage is split into `scale` groups with
    hierarchical residual connections."""

    expansion = 4

    def __init__(self, inplanes, planes, stride=1, downsample=None, cardinality=1,
                 base_width=26, scale=4, use_se=False, act_layer=nn.ReLU,
                 norm_layer=nn.BatchNorm2d, dilation=1, first_dilation=None, **_):
        super().__init__()
        self.scale = scale
        self.is_first = stride > 1 or downsample is not None
        self.num_scales = max(1, scale - 1)
        width = int(math.floor(planes * (base_width / 64.0))) * cardinality
        self.width = width
        outplanes = planes * self.expansion
        first_dilation = first_dilation or dilation

        self.conv1 = nn.Conv2d(inplanes, width * scale, kernel_size=1, bias=False)
        self.bn1 = norm_layer(width * scale)

        convs = []
        bns = []
        for _i in range(self.num_scales):
            convs.append(nn.Conv2d(width, width, kernel_size=3, stride=stride,
                                   padding=first_dilation, dilation=first_dilation,
                                   groups=cardinality, bias=False))
            bns.append(norm_layer(width))
        self.convs = nn.ModuleList(convs)
        self.bns = nn.ModuleList(bns)
        if self.is_first:
            self.pool = nn.AvgPool2d(kernel_size=3, stride=stride, padding=1)
        else:
            self.pool = None

        self.conv3 = nn.Conv2d(width * scale, outplanes, kernel_size=1, bias=False)
        self.bn3 = norm_layer(outplanes)
        self.se = None
        self.relu = act_layer(inplace=True)
        self.downsample = downsample

    def forward(self, x):
        residual = x
        out = self.conv1(x)
        out = self.bn1(out)
        out = self.relu(out)

        spx = torch.split(out, self.width, 1)
        spo = []
        sp = spx[0]
        for i, (conv, bn) in enumerate(zip(self.convs, self.bns)):
            if i == 0 or self.is_first:
                sp = spx[i]
            else:
                sp = sp + spx[i]
            sp = conv(sp)
            sp = bn(sp)
            sp = self.relu(sp)
            spo.append(sp)
        if self.scale > 1:
            if self.pool is not None:
                spo.append(self.pool(spx[-1]))
            else:
                spo.append(spx[-1])
        out = torch.cat(spo, 1)

        out = self.conv3(out)
        out = self.bn3(out)
        if self.downsample is not None:
            residual = self.downsample(x)
        out = out + residual
        return self.relu(out)


def _make(variant, layers, base_width=26, scale=4, cardinality=1, pretrained=False, **kwargs):
    import functools

    block = functools.partial(Bottle2neck, scale=scale)
    block.expansion = Bottle2neck.expansion
    model = ResNet(block, layers, base_width=base_width, cardinality=cardinality, **kwargs)
    model.default_cfg = default_cfgs[variant]
    return model


@register_model
def res2net50_26w_4s(pretrained=False, **kwargs):
    return _make("res2net50_26w_4s", [3, 4, 6, 3], 26, 4, 1, pretrained, **kwargs)


@register_model
def res2net50_48w_2s(pretrained=False, **kwargs):
    return _make("res2net50_48w_2s", [3, 4, 6, 3], 48, 2, 1, pretrained, **kwargs)


@register_model
def res2net50_14w_8s(pretrained=False, **kwargs):
    return _make("res2net50_14w_8s", [3, 4, 6, 3], 14, 8, 1, pretrained, **kwargs)


@register_model
def res2net50_26w_6s(pretrained=False, **kwargs):
    return _make("res2net50_26w_6s", [3, 4, 6, 3], 26, 6, 1, pretrained, **kwargs)


@register_model
def res2net50_26w_8s(pretrained=False, **kwargs):
    return _make("res2net50_26w_8s", [3, 4, 6, 3], 26, 8, 1, pretrained, **kwargs)


@register_model
def res2net101_26w_4s(pretrained=False, **kwargs):
    return _make("res2net101_26w_4s", [3, 4, 23, 3], 26, 4, 1, pretrained, **kwargs)


@register_model
def res2next50(pretrained=False, **kwargs):
    return _make("res2next50", [3, 4, 6, 3], 4, 4, 8, pretrained, **kwargs)
