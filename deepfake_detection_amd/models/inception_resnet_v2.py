"""Inception-ResNet-V2 — capability parity with reference
dfd/timm/models/inception_resnet_v2.py (355 LoC, 2 entrypoints:
inception_resnet_v2 / ens_adv_inception_resnet_v2 — same arch, different
pretrained weights). Szegedy et al., AAAI 2017.
"""

import torch
import torch.nn as nn
import torch.nn.functional as F

from .layers import SelectAdaptivePool2d
from .registry import register_model

__all__ = ["InceptionResnetV2"]


def _cfg(**kwargs):
    return {
        "url": "", "num_classes": 1000, "input_size": (3, 299, 299),
        "pool_size": (8, 8), "crop_pct": 0.8975, "interpolation": "bicubic",
        "mean": (0.5, 0.5, 0.5), "std": (0.5, 0.5, 0.5),
        "first_conv": "conv2d_1a.conv", "classifier": "classif", **kwargs,
    }


default_cfgs = {
    "inception_resnet_v2": _cfg(),
    "ens_adv_inception_resnet_v2": _cfg(),
}


class BasicConv2d(nn.Module):
    def __init__(self, in_planes, out_planes, kernel_size, stride, padding=0):
        super().__init__()
        self.conv = nn.Conv2d(in_planes, out_planes, kernel_size, stride,
                              padding=padding, bias=False)
        self.bn = nn.BatchNorm2d(out_planes, eps=0.001)
        self.relu = nn.ReLU(inplace=False)

    def forward(self, x):
        return self.relu(self.bn(self.conv(x)))


class Mixed_5b(nn.Module):
    def __init__(self):
        super().__init__()
        self.branch0 = BasicConv2d(192, 96, 1, 1)
        self.branch1 = nn.Sequential(
            BasicConv2d(192, 48, 1, 1), BasicConv2d(48, 64, 5, 1, padding=2))
        self.branch2 = nn.Sequential(
            BasicConv2d(192, 64, 1, 1), BasicConv2d(64, 96, 3, 1, padding=1),
            BasicConv2d(96, 96, 3, 1, padding=1))
        self.branch3 = nn.Sequential(
            nn.AvgPool2d(3, stride=1, padding=1, count_include_pad=False),
            BasicConv2d(192, 64, 1, 1))

    def forward(self, x):
        return torch.cat([self.branch0(x), self.branch1(x), self.branch2(x),
                          self.branch3(x)], 1)


class Block35(nn.Module):
    def __init__(self, scale=1.0):
        super().__init__()
        self.scale = scale
        self.branch0 = BasicConv2d(320, 32, 1, 1)
        self.branch1 = nn.Sequential(
            BasicConv2d(320, 32, 1, 1), BasicConv2d(32, 32, 3, 1, padding=1))
        self.branch2 = nn.Sequential(
            BasicConv2d(320, 32, 1, 1), BasicConv2d(32, 48, 3, 1, padding=1),
            BasicConv2d(48, 64, 3, 1, padding=1))
        self.conv2d = nn.Conv2d(128, 320, 1, 1)
        self.relu = nn.ReLU(inplace=False)

    def forward(self, x):
        out = torch.cat([self.branch0(x), self.branch1(x), self.branch2(x)], 1)
        return self.relu(x + self.scale * self.conv2d(out))


class Mixed_6a(nn.Module):
    def __init__(self):
        super().__init__()
        self.branch0 = BasicConv2d(320, 384, 3, 2)
        self.branch1 = nn.Sequential(
            BasicConv2d(320, 256, 1, 1), BasicConv2d(256, 256, 3, 1, padding=1),
            BasicConv2d(256, 384, 3, 2))
        self.branch2 = nn.MaxPool2d(3, stride=2)

    def forward(self, x):
        return torch.cat([self.branch0(x), self.branch1(x), self.branch2(x)], 1)


class Block17(nn.Module):
    def __init__(self, scale=1.0):
        super().__init__()
        self.scale = scale
        self.branch0 = BasicConv2d(1088, 192, 1, 1)
        self.branch1 = nn.Sequential(
            BasicConv2d(1088, 128, 1, 1),
            BasicConv2d(128, 160, (1, 7), 1, padding=(0, 3)),
            BasicConv2d(160, 192, (7, 1), 1, padding=(3, 0)))
        self.conv2d = nn.Conv2d(384, 1088, 1, 1)
        self.relu = nn.ReLU(inplace=False)

    def forward(self, x):
        out = torch.cat([self.branch0(x), self.branch1(x)], 1)
        return self.relu(x + self.scale * self.conv2d(out))


class Mixed_7a(nn.Module):
    def __init__(self):
        super().__init__()
        self.branch0 = nn.Sequential(
            BasicConv2d(1088, 256, 1, 1), BasicConv2d(256, 384, 3, 2))
        self.branch1 = nn.Sequential(
            BasicConv2d(1088, 256, 1, 1), BasicConv2d(256, 288, 3, 2))
        self.branch2 = nn.Sequential(
            BasicConv2d(1088, 256, 1, 1), BasicConv2d(256, 288, 3, 1, padding=1),
            BasicConv2d(288, 320, 3, 2))
        self.branch3 = nn.MaxPool2d(3, stride=2)

    def forward(self, x):
        return torch.cat([self.branch0(x), self.branch1(x), self.branch2(x),
                          self.branch3(x)], 1)


class Block8(nn.Module):
    def __init__(self, scale=1.0, no_relu=False):
        super().__init__()
        self.scale = scale
        self.branch0 = BasicConv2d(2080, 192, 1, 1)
        self.branch1 = nn.Sequential(
            BasicConv2d(2080, 192, 1, 1),
            BasicConv2d(192, 224, (1, 3), 1, padding=(0, 1)),
            BasicConv2d(224, 256, (3, 1), 1, padding=(1, 0)))
        self.conv2d = nn.Conv2d(448, 2080, 1, 1)
        self.relu = None if no_relu else nn.ReLU(inplace=False)

    def forward(self, x):
        out = torch.cat([self.branch0(x), self.branch1(x)], 1)
        out = x + self.scale * self.conv2d(out)
        return self.relu(out) if self.relu is not None else out


class InceptionResnetV2(nn.Module):
    def __init__(self, num_classes=1000, in_chans=3, drop_rate=0.0, global_pool="avg"):
        super().__init__()
        self.num_classes = num_classes
        self.drop_rate = drop_rate
        self.num_features = 1536

        self.conv2d_1a = BasicConv2d(in_chans, 32, 3, 2)
        self.conv2d_2a = BasicConv2d(32, 32, 3, 1)
        self.conv2d_2b = BasicConv2d(32, 64, 3, 1, padding=1)
        self.maxpool_3a = nn.MaxPool2d(3, stride=2)
        self.conv2d_3b = BasicConv2d(64, 80, 1, 1)
        self.conv2d_4a = BasicConv2d(80, 192, 3, 1)
        self.maxpool_5a = nn.MaxPool2d(3, stride=2)
        self.mixed_5b = Mixed_5b()
        self.repeat = nn.Sequential(*[Block35(scale=0.17) for _ in range(10)])
        self.mixed_6a = Mixed_6a()
        self.repeat_1 = nn.Sequential(*[Block17(scale=0.10) for _ in range(20)])
        self.mixed_7a = Mixed_7a()
        self.repeat_2 = nn.Sequential(*[Block8(scale=0.20) for _ in range(9)])
        self.block8 = Block8(no_relu=True)
        self.conv2d_7b = BasicConv2d(2080, self.num_features, 1, 1)
        self.global_pool = SelectAdaptivePool2d(pool_type=global_pool)
        self.classif = nn.Linear(self.num_features * self.global_pool.feat_mult(), num_classes)

    def get_classifier(self):
        return self.classif

    def reset_classifier(self, num_classes, global_pool="avg"):
        self.global_pool = SelectAdaptivePool2d(pool_type=global_pool)
        self.num_classes = num_classes
        self.classif = nn.Linear(
            self.num_features * self.global_pool.feat_mult(), num_classes) if num_classes else None

    def forward_features(self, x):
        x = self.conv2d_1a(x)
        x = self.conv2d_2a(x)
        x = self.conv2d_2b(x)
        x = self.maxpool_3a(x)
        x = self.conv2d_3b(x)
        x = self.conv2d_4a(x)
        x = self.maxpool_5a(x)
        x = self.mixed_5b(x)
        x = self.repeat(x)
        x = self.mixed_6a(x)
        x = self.repeat_1(x)
        x = self.mixed_7a(x)
        x = self.repeat_2(x)
        x = self.block8(x)
        x = self.conv2d_7b(x)
        return x

    def forward(self, x):
        x = self.forward_features(x)
        x = self.global_pool(x).flatten(1)
        if self.drop_rate > 0:
            x = F.dropout(x, p=self.drop_rate, training=self.training)
        return self.classif(x)


@register_model
def inception_resnet_v2(pretrained=False, **kwargs):
    model = InceptionResnetV2(**kwargs)
    model.default_cfg = default_cfgs["inception_resnet_v2"]
    return model


@register_model
def ens_adv_inception_resnet_v2(pretrained=False, **kwargs):
    """Ensemble-adversarially-trained weights of the same architecture."""
    model = InceptionResnetV2(**kwargs)
    model.default_cfg = default_cfgs["ens_adv_inception_resnet_v2"]
    return model
