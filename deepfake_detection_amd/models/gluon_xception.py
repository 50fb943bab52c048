"""Modified Aligned Xception (Gluon) — capability parity with reference
dfd/timm/models/gluon_xception.py (468 LoC, 2 entrypoints:
gluon_xception65 / gluon_xception71). DeepLab-style aligned Xception with
separable-conv blocks, a 16-block middle flow and output-stride control.
"""

from collections import OrderedDict

import torch.nn as nn
import torch.nn.functional as F

from .layers import SelectAdaptivePool2d
from .registry import register_model

__all__ = ["Xception65", "Xception71"]


def _cfg(**kwargs):
    return {
        "url": "", "num_classes": 1000, "input_size": (3, 299, 299),
        "pool_size": (10, 10), "crop_pct": 0.875, "interpolation": "bicubic",
        "mean": (0.5, 0.5, 0.5), "std": (0.5, 0.5, 0.5),
        "first_conv": "conv1", "classifier": "fc", **kwargs,
    }


default_cfgs = {
    "gluon_xception65": _cfg(),
    "gluon_xception71": _cfg(),
}


def _same_pad(k, stride=1, dilation=1):
    return ((stride - 1) + dilation * (k - 1)) // 2


class SeparableConv2d(nn.Module):
    def __init__(self, inplanes, planes, kernel_size=3, stride=1, dilation=1,
                 bias=False, norm_layer=nn.BatchNorm2d):
        super().__init__()
        self.conv_dw = nn.Conv2d(
            inplanes, inplanes, kernel_size, stride=stride,
            padding=_same_pad(kernel_size, stride, dilation), dilation=dilation,
            groups=inplanes, bias=bias)
        self.bn = norm_layer(inplanes)
        self.conv_pw = nn.Conv2d(inplanes, planes, 1, bias=bias)

    def forward(self, x):
        return self.conv_pw(self.bn(self.conv_dw(x)))


class Block(nn.Module):
    def __init__(self, inplanes, planes, num_reps, stride=1, dilation=1,
                 start_with_relu=True, norm_layer=nn.BatchNorm2d):
        super().__init__()
        if planes != inplanes or stride != 1:
            self.skip = nn.Sequential(OrderedDict([
                ("conv1", nn.Conv2d(inplanes, planes, 1, stride=stride, bias=False)),
                ("bn1", norm_layer(planes)),
            ]))
        else:
            self.skip = None
        rep = OrderedDict()
        filters = inplanes
        l = 1
        rep[f"act{l}"] = nn.ReLU(inplace=False if l == 1 and not start_with_relu else True)
        rep[f"conv{l}"] = SeparableConv2d(filters, planes, 3, 1, dilation, norm_layer=norm_layer)
        rep[f"bn{l}"] = norm_layer(planes)
        filters = planes
        for _ in range(num_reps - 1):
            l += 1
            rep[f"act{l}"] = nn.ReLU(inplace=True)
            rep[f"conv{l}"] = SeparableConv2d(filters, filters, 3, 1, dilation,
                                              norm_layer=norm_layer)
            rep[f"bn{l}"] = norm_layer(filters)
        if stride != 1:
            l += 1
            rep[f"act{l}"] = nn.ReLU(inplace=True)
            rep[f"conv{l}"] = SeparableConv2d(planes, planes, 3, stride, norm_layer=norm_layer)
            rep[f"bn{l}"] = norm_layer(planes)
        if not start_with_relu:
            del rep["act1"]
        self.rep = nn.Sequential(rep)

    def forward(self, x):
        skip = self.skip(x) if self.skip is not None else x
        return self.rep(x) + skip


def _os_params(output_stride):
    if output_stride == 32:
        return 2, 2, 1, (1, 1)
    if output_stride == 16:
        return 2, 1, 1, (1, 2)
    assert output_stride == 8
    return 1, 1, 2, (2, 4)


class _XceptionBase(nn.Module):
    num_features = 2048

    def _head(self, exit_dil, norm_layer, num_classes, global_pool):
        self.block20 = Block(728, 1024, num_reps=2, stride=self._exit_stride,
                             dilation=exit_dil[0], norm_layer=norm_layer)
        self.conv3 = SeparableConv2d(1024, 1536, 3, 1, exit_dil[1], norm_layer=norm_layer)
        self.bn3 = norm_layer(1536)
        self.conv4 = SeparableConv2d(1536, 1536, 3, 1, exit_dil[1], norm_layer=norm_layer)
        self.bn4 = norm_layer(1536)
        self.conv5 = SeparableConv2d(1536, self.num_features, 3, 1, exit_dil[1],
                                     norm_layer=norm_layer)
        self.bn5 = norm_layer(self.num_features)
        self.global_pool = SelectAdaptivePool2d(pool_type=global_pool)
        self.fc = nn.Linear(self.num_features * self.global_pool.feat_mult(), num_classes)

    def get_classifier(self):
        return self.fc

    def reset_classifier(self, num_classes, global_pool="avg"):
        self.num_classes = num_classes
        self.global_pool = SelectAdaptivePool2d(pool_type=global_pool)
        self.fc = nn.Linear(
            self.num_features * self.global_pool.feat_mult(), num_classes) if num_classes else None

    def _tail(self, x):
        x = self.mid(x)
        x = self.block20(x)
        x = F.relu(self.bn3(self.conv3(x)), inplace=True)
        x = F.relu(self.bn4(self.conv4(x)), inplace=True)
        x = F.relu(self.bn5(self.conv5(x)), inplace=True)
        return x

    def forward(self, x):
        x = self.forward_features(x)
        x = self.global_pool(x).flatten(1)
        if self.drop_rate > 0.0:
            x = F.dropout(x, p=self.drop_rate, training=self.training)
        return self.fc(x)


class Xception65(_XceptionBase):
    def __init__(self, num_classes=1000, in_chans=3, output_stride=32,
                 norm_layer=nn.BatchNorm2d, drop_rate=0.0, global_pool="avg"):
        super().__init__()
        self.num_classes = num_classes
        self.drop_rate = drop_rate
        eb3_stride, self._exit_stride, mid_dil, exit_dil = _os_params(output_stride)

        self.conv1 = nn.Conv2d(in_chans, 32, 3, stride=2, padding=1, bias=False)
        self.bn1 = norm_layer(32)
        self.conv2 = nn.Conv2d(32, 64, 3, stride=1, padding=1, bias=False)
        self.bn2 = norm_layer(64)
        self.block1 = Block(64, 128, 2, stride=2, start_with_relu=False, norm_layer=norm_layer)
        self.block2 = Block(128, 256, 2, stride=2, norm_layer=norm_layer)
        self.block3 = Block(256, 728, 2, stride=eb3_stride, norm_layer=norm_layer)
        self.mid = nn.Sequential(OrderedDict([
            (f"block{i}", Block(728, 728, 3, stride=1, dilation=mid_dil, norm_layer=norm_layer))
            for i in range(4, 20)]))
        self._head(exit_dil, norm_layer, num_classes, global_pool)

    def forward_features(self, x):
        x = F.relu(self.bn1(self.conv1(x)), inplace=True)
        x = F.relu(self.bn2(self.conv2(x)), inplace=True)
        x = self.block1(x)
        x = F.relu(x, inplace=True)
        x = self.block2(x)
        x = self.block3(x)
        return self._tail(x)


class Xception71(_XceptionBase):
    def __init__(self, num_classes=1000, in_chans=3, output_stride=32,
                 norm_layer=nn.BatchNorm2d, drop_rate=0.0, global_pool="avg"):
        super().__init__()
        self.num_classes = num_classes
        self.drop_rate = drop_rate
        eb3_stride, self._exit_stride, mid_dil, exit_dil = _os_params(output_stride)

        self.conv1 = nn.Conv2d(in_chans, 32, 3, stride=2, padding=1, bias=False)
        self.bn1 = norm_layer(32)
        self.conv2 = nn.Conv2d(32, 64, 3, stride=1, padding=1, bias=False)
        self.bn2 = norm_layer(64)
        self.block1 = Block(64, 128, 2, stride=2, start_with_relu=False, norm_layer=norm_layer)
        self.block2 = nn.Sequential(
            Block(128, 256, 2, stride=1, norm_layer=norm_layer),
            Block(256, 256, 2, stride=2, norm_layer=norm_layer),
            Block(256, 728, 2, stride=2, norm_layer=norm_layer))
        self.block3 = Block(728, 728, 2, stride=eb3_stride, norm_layer=norm_layer)
        self.mid = nn.Sequential(OrderedDict([
            (f"block{i}", Block(728, 728, 3, stride=1, dilation=mid_dil, norm_layer=norm_layer))
            for i in range(4, 20)]))
        self._head(exit_dil, norm_layer, num_classes, global_pool)

    def forward_features(self, x):
        x = F.relu(self.bn1(self.conv1(x)), inplace=True)
        x = F.relu(self.bn2(self.conv2(x)), inplace=True)
        x = self.block1(x)
        x = F.relu(x, inplace=True)
        x = self.block2(x)
        x = self.block3(x)
        return self._tail(x)


@register_model
def gluon_xception65(pretrained=False, **kwargs):
    model = Xception65(**kwargs)
    model.default_cfg = default_cfgs["gluon_xception65"]
    return model


@register_model
def gluon_xception71(pretrained=False, **kwargs):
    model = Xception71(**kwargs)
    model.default_cfg = default_cfgs["gluon_xception71"]
    return model
