"""SelecSLS (selective short/long-range skip connections), capability parity
with reference dfd/timm/models/selecsls.py (294 LoC, 5 entrypoints:
selecsls42/42b/60/60b/84). Feature/head tables follow the paper's published
configurations (Mehta et al., "XNect", SIGGRAPH 2020).
"""

import torch
import torch.nn as nn
import torch.nn.functional as F

from .layers import SelectAdaptivePool2d
from .registry import register_model

__all__ = ["SelecSLS"]


def _cfg(**kwargs):
    return {
        "url": "", "num_classes": 1000, "input_size": (3, 224, 224),
        "pool_size": (4, 4), "crop_pct": 0.875, "interpolation": "bicubic",
        "mean": (0.485, 0.456, 0.406), "std": (0.229, 0.224, 0.225),
        "first_conv": "stem", "classifier": "fc", **kwargs,
    }


default_cfgs = {n: _cfg() for n in
                ["selecsls42", "selecsls42b", "selecsls60", "selecsls60b", "selecsls84"]}


def conv_bn(in_chs, out_chs, k=3, stride=1):
    pad = ((stride - 1) + (k - 1)) // 2
    return nn.Sequential(
        nn.Conv2d(in_chs, out_chs, k, stride, padding=pad, bias=False),
        nn.BatchNorm2d(out_chs),
        nn.ReLU(inplace=True),
    )


class SelecSLSBlock(nn.Module):
    """Block with internal 3/1-kernel conv chain; first block of a stage
    drops the cross-stage skip, later blocks concatenate it."""

    def __init__(self, in_chs, skip_chs, mid_chs, out_chs, is_first, stride):
        super().__init__()
        self.stride = stride
        self.is_first = is_first
        assert stride in (1, 2)
        self.conv1 = conv_bn(in_chs, mid_chs, 3, stride)
        self.conv2 = conv_bn(mid_chs, mid_chs, 1)
        self.conv3 = conv_bn(mid_chs, mid_chs // 2, 3)
        self.conv4 = conv_bn(mid_chs // 2, mid_chs, 1)
        self.conv5 = conv_bn(mid_chs, mid_chs // 2, 3)
        self.conv6 = conv_bn(2 * mid_chs + (0 if is_first else skip_chs), out_chs, 1)

    def forward(self, x):
        assert isinstance(x, list)
        assert len(x) in (1, 2)
        d1 = self.conv1(x[0])
        d2 = self.conv3(self.conv2(d1))
        d3 = self.conv5(self.conv4(d2))
        if self.is_first:
            out = self.conv6(torch.cat([d1, d2, d3], 1))
            return [out, out]
        return [self.conv6(torch.cat([d1, d2, d3, x[1]], 1)), x[1]]


class _BlockList(nn.Sequential):
    # nn.Sequential that threads the [features, skip] list through blocks
    def forward(self, x):  # noqa: D102
        for m in self:
            x = m(x)
        return x


# (in_chs, skip_chs, mid_chs, out_chs, is_first, stride) per block;
# head: (in, out, k, stride) conv_bn list; num_features of the head output
_ARCH = {
    "selecsls42": dict(
        features=[
            (32, 0, 64, 64, True, 2), (64, 64, 64, 128, False, 1),
            (128, 0, 144, 144, True, 2), (144, 144, 144, 288, False, 1),
            (288, 0, 304, 304, True, 2), (304, 304, 304, 480, False, 1),
        ],
        head=[(480, 960, 3, 2), (960, 1024, 3, 1), (1024, 1024, 3, 2), (1024, 1280, 1, 1)],
        num_features=1280,
    ),
    "selecsls60": dict(
        features=[
            (32, 0, 64, 64, True, 2), (64, 64, 64, 128, False, 1),
            (128, 0, 128, 128, True, 2), (128, 128, 128, 128, False, 1),
            (128, 128, 128, 288, False, 1),
            (288, 0, 288, 288, True, 2), (288, 288, 288, 288, False, 1),
            (288, 288, 288, 288, False, 1), (288, 288, 288, 416, False, 1),
        ],
        head=[(416, 756, 3, 2), (756, 1024, 3, 1), (1024, 1024, 3, 2), (1024, 1280, 1, 1)],
        num_features=1280,
    ),
    "selecsls84": dict(
        features=[
            (32, 0, 64, 64, True, 2), (64, 64, 64, 144, False, 1),
            (144, 0, 144, 144, True, 2), (144, 144, 144, 144, False, 1),
            (144, 144, 144, 144, False, 1), (144, 144, 144, 144, False, 1),
            (144, 144, 144, 304, False, 1),
            (304, 0, 304, 304, True, 2), (304, 304, 304, 304, False, 1),
            (304, 304, 304, 304, False, 1), (304, 304, 304, 304, False, 1),
            (304, 304, 304, 304, False, 1), (304, 304, 304, 512, False, 1),
        ],
        head=[(512, 960, 3, 2), (960, 1024, 3, 1), (1024, 1024, 3, 2), (1024, 1280, 3, 1)],
        num_features=1280,
    ),
}
# "b" heads swap the last two convs for a wider penultimate stage
_ARCH["selecsls42b"] = dict(
    features=_ARCH["selecsls42"]["features"],
    head=[(480, 960, 3, 2), (960, 1024, 3, 1), (1024, 1280, 3, 2), (1280, 1024, 1, 1)],
    num_features=1024,
)
_ARCH["selecsls60b"] = dict(
    features=_ARCH["selecsls60"]["features"],
    head=[(416, 756, 3, 2), (756, 1024, 3, 1), (1024, 1280, 3, 2), (1280, 1024, 1, 1)],
    num_features=1024,
)


class SelecSLS(nn.Module):
    def __init__(self, variant, num_classes=1000, in_chans=3, drop_rate=0.0,
                 global_pool="avg"):
        super().__init__()
        arch = _ARCH[variant]
        self.num_classes = num_classes
        self.drop_rate = drop_rate
        self.stem = conv_bn(in_chans, 32, stride=2)
        self.features = _BlockList(*[SelecSLSBlock(*a) for a in arch["features"]])
        self.head = nn.Sequential(*[conv_bn(*a) for a in arch["head"]])
        self.num_features = arch["num_features"]
        self.global_pool = SelectAdaptivePool2d(pool_type=global_pool)
        self.fc = nn.Linear(self.num_features * self.global_pool.feat_mult(), num_classes)
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out", nonlinearity="relu")
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.constant_(m.weight, 1.0)
                nn.init.constant_(m.bias, 0.0)

    def get_classifier(self):
        return self.fc

    def reset_classifier(self, num_classes, global_pool="avg"):
        self.global_pool = SelectAdaptivePool2d(pool_type=global_pool)
        self.num_classes = num_classes
        self.fc = nn.Linear(
            self.num_features * self.global_pool.feat_mult(), num_classes) if num_classes else None

    def forward_features(self, x):
        x = self.stem(x)
        x = self.features([x])
        return self.head(x[0])

    def forward(self, x):
        x = self.forward_features(x)
        x = self.global_pool(x).flatten(1)
        if self.drop_rate > 0.0:
            x = F.dropout(x, p=self.drop_rate, training=self.training)
        return self.fc(x)


def _make(variant, pretrained=False, **kwargs):
    model = SelecSLS(variant, **kwargs)
    model.default_cfg = default_cfgs[variant]
    return model


@register_model
def selecsls42(pretrained=False, **kwargs):
    return _make("selecsls42", pretrained, **kwargs)


@register_model
def selecsls42b(pretrained=False, **kwargs):
    return _make("selecsls42b", pretrained, **kwargs)


@register_model
def selecsls60(pretrained=False, **kwargs):
    return _make("selecsls60", pretrained, **kwargs)


@register_model
def selecsls60b(pretrained=False, **kwargs):
    return _make("selecsls60b", pretrained, **kwargs)


@register_model
def selecsls84(pretrained=False, **kwargs):
    return _make("selecsls84", pretrained, **kwargs)
