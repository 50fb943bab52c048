"""Deep Layer Aggregation (DLA / DLA-X, plus Res2Net-DLA hybrids) —
capability parity with reference dfd/timm/models/dla.py (467 LoC, 12
entrypoints). Yu et al., CVPR 2018; hierarchical tree aggregation over
conv stages. Module names (base_layer, level0..5, tree1/tree2/root, fc)
match the published checkpoints.
"""

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from .layers import SelectAdaptivePool2d
from .registry import register_model

__all__ = ["DLA"]


def _cfg(**kwargs):
    return {
        "url": "", "num_classes": 1000, "input_size": (3, 224, 224),
        "pool_size": (7, 7), "crop_pct": 0.875, "interpolation": "bilinear",
        "mean": (0.485, 0.456, 0.406), "std": (0.229, 0.224, 0.225),
        "first_conv": "base_layer.0", "classifier": "fc", **kwargs,
    }


default_cfgs = {n: _cfg() for n in [
    "dla34", "dla46_c", "dla46x_c", "dla60x_c", "dla60", "dla60x", "dla102",
    "dla102x", "dla102x2", "dla169", "dla60_res2net", "dla60_res2next"]}


class DlaBasic(nn.Module):
    """Two 3x3 convs with an externally supplied residual."""

    def __init__(self, inplanes, planes, stride=1, dilation=1, **_):
        super().__init__()
        self.conv1 = nn.Conv2d(inplanes, planes, 3, stride=stride,
                               padding=dilation, dilation=dilation, bias=False)
        self.bn1 = nn.BatchNorm2d(planes)
        self.relu = nn.ReLU(inplace=True)
        self.conv2 = nn.Conv2d(planes, planes, 3, stride=1,
                               padding=dilation, dilation=dilation, bias=False)
        self.bn2 = nn.BatchNorm2d(planes)
        self.stride = stride

    def forward(self, x, residual=None):
        if residual is None:
            residual = x
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.bn2(self.conv2(out))
        return self.relu(out + residual)


class DlaBottleneck(nn.Module):
    """1-3-1 bottleneck; cardinality/base_width give the DLA-X variants."""

    expansion = 2

    def __init__(self, inplanes, outplanes, stride=1, dilation=1, cardinality=1,
                 base_width=64):
        super().__init__()
        self.stride = stride
        mid = int(math.floor(outplanes * (base_width / 64)) * cardinality) // self.expansion
        self.conv1 = nn.Conv2d(inplanes, mid, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(mid)
        self.conv2 = nn.Conv2d(mid, mid, 3, stride=stride, padding=dilation,
                               dilation=dilation, groups=cardinality, bias=False)
        self.bn2 = nn.BatchNorm2d(mid)
        self.conv3 = nn.Conv2d(mid, outplanes, 1, bias=False)
        self.bn3 = nn.BatchNorm2d(outplanes)
        self.relu = nn.ReLU(inplace=True)

    def forward(self, x, residual=None):
        if residual is None:
            residual = x
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        return self.relu(out + residual)


class DlaBottle2neck(nn.Module):
    """Res2Net-style multi-scale bottleneck for the dla60_res2net/res2next
    hybrids (scale-wise 3x3 convs over channel splits)."""

    expansion = 2

    def __init__(self, inplanes, outplanes, stride=1, dilation=1, scale=4,
                 cardinality=8, base_width=4):
        super().__init__()
        self.is_first = stride > 1
        self.scale = scale
        mid = int(math.floor(outplanes * (base_width / 64)) * cardinality) // self.expansion
        self.width = mid
        self.conv1 = nn.Conv2d(inplanes, mid * scale, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(mid * scale)
        n_convs = max(1, scale - 1)
        self.convs = nn.ModuleList([
            nn.Conv2d(mid, mid, 3, stride=stride, padding=dilation,
                      dilation=dilation, groups=cardinality, bias=False)
            for _ in range(n_convs)])
        self.bns = nn.ModuleList([nn.BatchNorm2d(mid) for _ in range(n_convs)])
        if self.is_first:
            self.pool = nn.AvgPool2d(3, stride=stride, padding=1)
        self.conv3 = nn.Conv2d(mid * scale, outplanes, 1, bias=False)
        self.bn3 = nn.BatchNorm2d(outplanes)
        self.relu = nn.ReLU(inplace=True)

    def forward(self, x, residual=None):
        if residual is None:
            residual = x
        out = self.relu(self.bn1(self.conv1(x)))
        splits = torch.split(out, self.width, 1)
        pieces = []
        sp = None
        for i, (conv, bn) in enumerate(zip(self.convs, self.bns)):
            sp = splits[i] if i == 0 or self.is_first else sp + splits[i]
            sp = self.relu(bn(conv(sp)))
            pieces.append(sp)
        if self.scale > 1:
            pieces.append(self.pool(splits[-1]) if self.is_first else splits[-1])
        out = self.bn3(self.conv3(torch.cat(pieces, 1)))
        return self.relu(out + residual)


class DlaRoot(nn.Module):
    def __init__(self, in_channels, out_channels, kernel_size, residual):
        super().__init__()
        self.conv = nn.Conv2d(in_channels, out_channels, 1, stride=1, bias=False,
                              padding=(kernel_size - 1) // 2)
        self.bn = nn.BatchNorm2d(out_channels)
        self.relu = nn.ReLU(inplace=True)
        self.residual = residual

    def forward(self, *children):
        x = self.bn(self.conv(torch.cat(children, 1)))
        if self.residual:
            x = x + children[0]
        return self.relu(x)


class DlaTree(nn.Module):
    def __init__(self, levels, block, in_channels, out_channels, stride=1,
                 dilation=1, cardinality=1, base_width=64, level_root=False,
                 root_dim=0, root_kernel_size=1, root_residual=False):
        super().__init__()
        if root_dim == 0:
            root_dim = 2 * out_channels
        if level_root:
            root_dim += in_channels
        cargs = dict(dilation=dilation, cardinality=cardinality, base_width=base_width)
        if levels == 1:
            self.tree1 = block(in_channels, out_channels, stride, **cargs)
            self.tree2 = block(out_channels, out_channels, 1, **cargs)
            self.root = DlaRoot(root_dim, out_channels, root_kernel_size, root_residual)
        else:
            cargs.update(root_kernel_size=root_kernel_size, root_residual=root_residual)
            self.tree1 = DlaTree(levels - 1, block, in_channels, out_channels,
                                 stride, root_dim=0, **cargs)
            self.tree2 = DlaTree(levels - 1, block, out_channels, out_channels,
                                 root_dim=root_dim + out_channels, **cargs)
        self.level_root = level_root
        self.levels = levels
        self.downsample = nn.MaxPool2d(stride, stride=stride) if stride > 1 else None
        self.project = None
        if in_channels != out_channels:
            self.project = nn.Sequential(
                nn.Conv2d(in_channels, out_channels, 1, stride=1, bias=False),
                nn.BatchNorm2d(out_channels))

    def forward(self, x, residual=None, children=None):
        children = [] if children is None else children
        bottom = self.downsample(x) if self.downsample else x
        residual = self.project(bottom) if self.project else bottom
        if self.level_root:
            children.append(bottom)
        x1 = self.tree1(x, residual)
        if self.levels == 1:
            x2 = self.tree2(x1)
            return self.root(x2, x1, *children)
        children.append(x1)
        return self.tree2(x1, children=children)


class DLA(nn.Module):
    def __init__(self, levels, channels, num_classes=1000, in_chans=3,
                 cardinality=1, base_width=64, block=DlaBottle2neck,
                 residual_root=False, drop_rate=0.0, global_pool="avg"):
        super().__init__()
        self.channels = channels
        self.num_classes = num_classes
        self.cardinality = cardinality
        self.base_width = base_width
        self.drop_rate = drop_rate

        self.base_layer = nn.Sequential(
            nn.Conv2d(in_chans, channels[0], 7, stride=1, padding=3, bias=False),
            nn.BatchNorm2d(channels[0]),
            nn.ReLU(inplace=True))
        self.level0 = self._conv_level(channels[0], channels[0], levels[0])
        self.level1 = self._conv_level(channels[0], channels[1], levels[1], stride=2)
        cargs = dict(cardinality=cardinality, base_width=base_width,
                     root_residual=residual_root)
        self.level2 = DlaTree(levels[2], block, channels[1], channels[2], 2,
                              level_root=False, **cargs)
        self.level3 = DlaTree(levels[3], block, channels[2], channels[3], 2,
                              level_root=True, **cargs)
        self.level4 = DlaTree(levels[4], block, channels[3], channels[4], 2,
                              level_root=True, **cargs)
        self.level5 = DlaTree(levels[5], block, channels[4], channels[5], 2,
                              level_root=True, **cargs)

        self.num_features = channels[-1]
        self.global_pool = SelectAdaptivePool2d(pool_type=global_pool)
        self.fc = nn.Conv2d(self.num_features * self.global_pool.feat_mult(),
                            num_classes, 1, bias=True)

        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                n = m.kernel_size[0] * m.kernel_size[1] * m.out_channels
                m.weight.data.normal_(0, math.sqrt(2.0 / n))
            elif isinstance(m, nn.BatchNorm2d):
                m.weight.data.fill_(1)
                m.bias.data.zero_()

    @staticmethod
    def _conv_level(inplanes, planes, convs, stride=1, dilation=1):
        layers = []
        for i in range(convs):
            layers += [
                nn.Conv2d(inplanes, planes, 3, stride=stride if i == 0 else 1,
                          padding=dilation, bias=False, dilation=dilation),
                nn.BatchNorm2d(planes),
                nn.ReLU(inplace=True)]
            inplanes = planes
        return nn.Sequential(*layers)

    def get_classifier(self):
        return self.fc

    def reset_classifier(self, num_classes, global_pool="avg"):
        self.num_classes = num_classes
        self.global_pool = SelectAdaptivePool2d(pool_type=global_pool)
        self.fc = nn.Conv2d(
            self.num_features * self.global_pool.feat_mult(), num_classes, 1,
            bias=True) if num_classes else None

    def forward_features(self, x):
        x = self.base_layer(x)
        for name in ("level0", "level1", "level2", "level3", "level4", "level5"):
            x = getattr(self, name)(x)
        return x

    def forward(self, x):
        x = self.forward_features(x)
        x = self.global_pool(x)
        if self.drop_rate > 0.0:
            x = F.dropout(x, p=self.drop_rate, training=self.training)
        return self.fc(x).flatten(1)


def _make(variant, levels, channels, block, pretrained=False, **kwargs):
    model = DLA(levels, channels, block=block, **kwargs)
    model.default_cfg = default_cfgs[variant]
    return model


@register_model
def dla34(pretrained=False, **kwargs):
    return _make("dla34", [1, 1, 1, 2, 2, 1], [16, 32, 64, 128, 256, 512],
                 DlaBasic, pretrained, **kwargs)


@register_model
def dla46_c(pretrained=False, **kwargs):
    return _make("dla46_c", [1, 1, 1, 2, 2, 1], [16, 32, 64, 64, 128, 256],
                 DlaBottleneck, pretrained, **kwargs)


@register_model
def dla46x_c(pretrained=False, **kwargs):
    return _make("dla46x_c", [1, 1, 1, 2, 2, 1], [16, 32, 64, 64, 128, 256],
                 DlaBottleneck, pretrained, cardinality=32, base_width=4, **kwargs)


@register_model
def dla60x_c(pretrained=False, **kwargs):
    return _make("dla60x_c", [1, 1, 1, 2, 3, 1], [16, 32, 64, 64, 128, 256],
                 DlaBottleneck, pretrained, cardinality=32, base_width=4, **kwargs)


@register_model
def dla60(pretrained=False, **kwargs):
    return _make("dla60", [1, 1, 1, 2, 3, 1], [16, 32, 128, 256, 512, 1024],
                 DlaBottleneck, pretrained, **kwargs)


@register_model
def dla60x(pretrained=False, **kwargs):
    return _make("dla60x", [1, 1, 1, 2, 3, 1], [16, 32, 128, 256, 512, 1024],
                 DlaBottleneck, pretrained, cardinality=32, base_width=4, **kwargs)


@register_model
def dla102(pretrained=False, **kwargs):
    return _make("dla102", [1, 1, 1, 3, 4, 1], [16, 32, 128, 256, 512, 1024],
                 DlaBottleneck, pretrained, residual_root=True, **kwargs)


@register_model
def dla102x(pretrained=False, **kwargs):
    return _make("dla102x", [1, 1, 1, 3, 4, 1], [16, 32, 128, 256, 512, 1024],
                 DlaBottleneck, pretrained, cardinality=32, base_width=4,
                 residual_root=True, **kwargs)


@register_model
def dla102x2(pretrained=False, **kwargs):
    return _make("dla102x2", [1, 1, 1, 3, 4, 1], [16, 32, 128, 256, 512, 1024],
                 DlaBottleneck, pretrained, cardinality=64, base_width=4,
                 residual_root=True, **kwargs)


@register_model
def dla169(pretrained=False, **kwargs):
    return _make("dla169", [1, 1, 2, 3, 5, 1], [16, 32, 128, 256, 512, 1024],
                 DlaBottleneck, pretrained, residual_root=True, **kwargs)


@register_model
def dla60_res2net(pretrained=False, **kwargs):
    return _make("dla60_res2net", (1, 1, 1, 2, 3, 1), (16, 32, 128, 256, 512, 1024),
                 DlaBottle2neck, pretrained, cardinality=1, base_width=28, **kwargs)


@register_model
def dla60_res2next(pretrained=False, **kwargs):
    return _make("dla60_res2next", (1, 1, 1, 2, 3, 1), (16, 32, 128, 256, 512, 1024),
                 DlaBottle2neck, pretrained, cardinality=8, base_width=4, **kwargs)
