"""Selective-Kernel Networks (reference dfd/timm/models/sknet.py, 237 LoC,
5 entrypoints) — SK convs on the ResNet trunk."""

import math

import torch.nn as nn

from .layers_extra import SelectiveKernelConv
from .registry import register_model
from .resnet import ResNet


def _cfg(**kwargs):
    return {
        "url": "", "num_classes": 1000, "input_size": (3, 224, 224),
        "pool_size": (7, 7), "crop_pct": 0.875, "interpolation": "bicubic",
        "mean": (0.485, 0.456, 0.406), "std": (0.229, 0.224, 0.225),
        "first_conv": "conv1", "classifier": "fc", **kwargs,
    }


default_cfgs = {k: _cfg() for k in [
    "skresnet18", "skresnet34", "skresnet50", "skresnet50d", "skresnext50_32x4d",
]}


class SelectiveKernelBasic(nn.Module):
    expansion = 1

    def __init__(self, inplanes, planes, stride=1, downsample=None, cardinality=1,
                 base_width=64, sk_kwargs=None, reduce_first=1, dilation=1,
                 first_dilation=None, act_layer=nn.ReLU, norm_layer=nn.BatchNorm2d, **_):
        super().__init__()
        sk_kwargs = sk_kwargs or {}
        assert cardinality == 1 and base_width == 64
        first_planes = planes // reduce_first
        outplanes = planes * self.expansion
        first_dilation = first_dilation or dilation

        self.conv1 = SelectiveKernelConv(
            inplanes, first_planes, stride=stride, dilation=first_dilation,
            act_layer=act_layer, norm_layer=norm_layer, **sk_kwargs)
        self.conv2 = nn.Sequential(
            nn.Conv2d(first_planes, outplanes, kernel_size=3, padding=dilation,
                      dilation=dilation, bias=False),
            norm_layer(outplanes))
        self.act = act_layer(inplace=True)
        self.downsample = downsample

    def forward(self, x):
        residual = x
        out = self.conv1(x)
        out = self.conv2(out)
        if self.downsample is not None:
            residual = self.downsample(x)
        out = out + residual
        return self.act(out)


class SelectiveKernelBottleneck(nn.Module):
    expansion = 4

    def __init__(self, inplanes, planes, stride=1, downsample=None, cardinality=1,
                 base_width=64, sk_kwargs=None, reduce_first=1, dilation=1,
                 first_dilation=None, act_layer=nn.ReLU, norm_layer=nn.BatchNorm2d, **_):
        super().__init__()
        sk_kwargs = sk_kwargs or {}
        width = int(math.floor(planes * (base_width / 64)) * cardinality)
        first_planes = width // reduce_first
        outplanes = planes * self.expansion
        first_dilation = first_dilation or dilation

        self.conv1 = nn.Sequential(
            nn.Conv2d(inplanes, first_planes, 1, bias=False),
            norm_layer(first_planes),
            act_layer(inplace=True))
        self.conv2 = SelectiveKernelConv(
            first_planes, width, stride=stride, dilation=first_dilation,
            groups=cardinality, act_layer=act_layer, norm_layer=norm_layer, **sk_kwargs)
        self.conv3 = nn.Sequential(
            nn.Conv2d(width, outplanes, 1, bias=False),
            norm_layer(outplanes))
        self.act = act_layer(inplace=True)
        self.downsample = downsample

    def forward(self, x):
        residual = x
        out = self.conv1(x)
        out = self.conv2(out)
        out = self.conv3(out)
        if self.downsample is not None:
            residual = self.downsample(x)
        out = out + residual
        return self.act(out)


def _make(variant, block, layers, pretrained=False, sk_kwargs=None, **kwargs):
    import functools

    blk = functools.partial(block, sk_kwargs=sk_kwargs)
    blk.expansion = block.expansion
    model = ResNet(blk, layers, **kwargs)
    model.default_cfg = default_cfgs[variant]
    return model


@register_model
def skresnet18(pretrained=False, **kwargs):
    sk_kwargs = dict(min_attn_channels=16, attn_reduction=8, split_input=True)
    return _make("skresnet18", SelectiveKernelBasic, [2, 2, 2, 2], pretrained,
                 sk_kwargs=sk_kwargs, **kwargs)


@register_model
def skresnet34(pretrained=False, **kwargs):
    sk_kwargs = dict(min_attn_channels=16, attn_reduction=8, split_input=True)
    return _make("skresnet34", SelectiveKernelBasic, [3, 4, 6, 3], pretrained,
                 sk_kwargs=sk_kwargs, **kwargs)


@register_model
def skresnet50(pretrained=False, **kwargs):
    sk_kwargs = dict(split_input=True)
    return _make("skresnet50", SelectiveKernelBottleneck, [3, 4, 6, 3], pretrained,
                 sk_kwargs=sk_kwargs, **kwargs)


@register_model
def skresnet50d(pretrained=False, **kwargs):
    sk_kwargs = dict(split_input=True)
    return _make("skresnet50d", SelectiveKernelBottleneck, [3, 4, 6, 3], pretrained,
                 sk_kwargs=sk_kwargs, stem_width=32, deep_stem=True, avg_down=True, **kwargs)


@register_model
def skresnext50_32x4d(pretrained=False, **kwargs):
    return _make("skresnext50_32x4d", SelectiveKernelBottleneck, [3, 4, 6, 3],
                 pretrained, cardinality=32, base_width=4, **kwargs)
