"""Inception-V4.

Capability parity with reference dfd/timm/models/inception_v4.py (308 LoC,
1 entrypoint). This implementation is spec-driven: every mixed/reduction
cell is a `_Cat` of named branches built by small helpers, so the whole
architecture reads as data while the state-dict keys (features.N.branchX...)
stay byte-identical to the reference's module layout. Inception-C's
tree-shaped cell (two split tails) keeps its own module.
"""

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import functional as O
from .layers import SelectAdaptivePool2d
from .registry import register_model

__all__ = ["InceptionV4"]

default_cfgs = {
    "inception_v4": {
        "url": "", "num_classes": 1000, "input_size": (3, 299, 299),
        "pool_size": (8, 8), "crop_pct": 0.875, "interpolation": "bicubic",
        "mean": (0.5, 0.5, 0.5), "std": (0.5, 0.5, 0.5),
        "first_conv": "features.0.conv", "classifier": "last_linear",
    }
}


class BasicConv2d(nn.Module):
    def __init__(self, in_planes, out_planes, kernel_size, stride, padding=0):
        super().__init__()
        self.conv = nn.Conv2d(in_planes, out_planes, kernel_size=kernel_size,
                              stride=stride, padding=padding, bias=False)
        self.bn = nn.BatchNorm2d(out_planes, eps=0.001, momentum=0.1)
        self.relu = nn.ReLU(inplace=True)

    def forward(self, x):
        x = self.conv(x)
        return O.bn_act(x, self.bn, "relu")


def _c(cin, cout, k, s=1, p=0):
    return BasicConv2d(cin, cout, kernel_size=k, stride=s, padding=p)


def _chain(*convs):
    return nn.Sequential(*convs)


def _avg_then(conv):
    return nn.Sequential(
        nn.AvgPool2d(3, stride=1, padding=1, count_include_pad=False), conv)


class _Cat(nn.Module):
    """Channel-concatenation of named branches (branch attribute names are
    the state-dict keys; kwargs order is the concat order)."""

    def __init__(self, **branches):
        super().__init__()
        self._order = list(branches)
        for name, mod in branches.items():
            self.add_module(name, mod)

    def forward(self, x):
        return torch.cat([getattr(self, n)(x) for n in self._order], 1)


def _mixed_3a():
    return _Cat(maxpool=nn.MaxPool2d(3, stride=2), conv=_c(64, 96, 3, s=2))


def _mixed_4a():
    return _Cat(
        branch0=_chain(_c(160, 64, 1), _c(64, 96, 3)),
        branch1=_chain(_c(160, 64, 1), _c(64, 64, (1, 7), p=(0, 3)),
                       _c(64, 64, (7, 1), p=(3, 0)), _c(64, 96, (3, 3))))


def _mixed_5a():
    return _Cat(conv=_c(192, 192, 3, s=2), maxpool=nn.MaxPool2d(3, stride=2))


def _inception_a():
    return _Cat(
        branch0=_c(384, 96, 1),
        branch1=_chain(_c(384, 64, 1), _c(64, 96, 3, p=1)),
        branch2=_chain(_c(384, 64, 1), _c(64, 96, 3, p=1), _c(96, 96, 3, p=1)),
        branch3=_avg_then(_c(384, 96, 1)))


def _reduction_a():
    return _Cat(
        branch0=_c(384, 384, 3, s=2),
        branch1=_chain(_c(384, 192, 1), _c(192, 224, 3, p=1), _c(224, 256, 3, s=2)),
        branch2=nn.MaxPool2d(3, stride=2))


def _inception_b():
    return _Cat(
        branch0=_c(1024, 384, 1),
        branch1=_chain(_c(1024, 192, 1), _c(192, 224, (1, 7), p=(0, 3)),
                       _c(224, 256, (7, 1), p=(3, 0))),
        branch2=_chain(_c(1024, 192, 1), _c(192, 192, (7, 1), p=(3, 0)),
                       _c(192, 224, (1, 7), p=(0, 3)), _c(224, 224, (7, 1), p=(3, 0)),
                       _c(224, 256, (1, 7), p=(0, 3))),
        branch3=_avg_then(_c(1024, 128, 1)))


def _reduction_b():
    return _Cat(
        branch0=_chain(_c(1024, 192, 1), _c(192, 192, 3, s=2)),
        branch1=_chain(_c(1024, 256, 1), _c(256, 256, (1, 7), p=(0, 3)),
                       _c(256, 320, (7, 1), p=(3, 0)), _c(320, 320, 3, s=2)),
        branch2=nn.MaxPool2d(3, stride=2))


class InceptionC(nn.Module):
    """The C cell's two middle branches fork at their tails, so it keeps an
    explicit module (attribute names = reference state-dict keys)."""

    def __init__(self):
        super().__init__()
        self.branch0 = _c(1536, 256, 1)
        self.branch1_0 = _c(1536, 384, 1)
        self.branch1_1a = _c(384, 256, (1, 3), p=(0, 1))
        self.branch1_1b = _c(384, 256, (3, 1), p=(1, 0))
        self.branch2_0 = _c(1536, 384, 1)
        self.branch2_1 = _c(384, 448, (3, 1), p=(1, 0))
        self.branch2_2 = _c(448, 512, (1, 3), p=(0, 1))
        self.branch2_3a = _c(512, 256, (1, 3), p=(0, 1))
        self.branch2_3b = _c(512, 256, (3, 1), p=(1, 0))
        self.branch3 = _avg_then(_c(1536, 256, 1))

    def forward(self, x):
        t1 = self.branch1_0(x)
        t2 = self.branch2_2(self.branch2_1(self.branch2_0(x)))
        return torch.cat((
            self.branch0(x),
            torch.cat((self.branch1_1a(t1), self.branch1_1b(t1)), 1),
            torch.cat((self.branch2_3a(t2), self.branch2_3b(t2)), 1),
            self.branch3(x)), 1)


class InceptionV4(nn.Module):
    def __init__(self, num_classes=1000, in_chans=3, drop_rate=0.0, global_pool="avg"):
        super().__init__()
        self.drop_rate = drop_rate
        self.num_classes = num_classes
        self.num_features = 1536

        stages = [
            _c(in_chans, 32, 3, s=2), _c(32, 32, 3), _c(32, 64, 3, p=1),
            _mixed_3a(), _mixed_4a(), _mixed_5a(),
        ]
        stages += [_inception_a() for _ in range(4)]
        stages += [_reduction_a()]
        stages += [_inception_b() for _ in range(7)]
        stages += [_reduction_b()]
        stages += [InceptionC() for _ in range(3)]
        self.features = nn.Sequential(*stages)
        self.global_pool = SelectAdaptivePool2d(pool_type=global_pool)
        self.last_linear = nn.Linear(self.num_features * self.global_pool.feat_mult(),
                                     num_classes)

    def get_classifier(self):
        return self.last_linear

    def reset_classifier(self, num_classes, global_pool="avg"):
        self.num_classes = num_classes
        self.global_pool = SelectAdaptivePool2d(pool_type=global_pool)
        self.last_linear = nn.Linear(
            self.num_features * self.global_pool.feat_mult(), num_classes) if num_classes else None

    def forward_features(self, x):
        return self.features(x)

    def forward(self, x):
        x = self.forward_features(x)
        if self.global_pool.pool_type == "avg":
            x = O.global_avg_pool(x)
        else:
            x = self.global_pool(x).flatten(1)
        if self.drop_rate > 0:
            x = F.dropout(x, p=self.drop_rate, training=self.training)
        return self.last_linear(x)


@register_model
def inception_v4(pretrained=False, **kwargs):
    model = InceptionV4(**kwargs)
    model.default_cfg = default_cfgs["inception_v4"]
    return model
