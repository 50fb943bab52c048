"""SENet / legacy SE-ResNet / SE-ResNeXt family (reference
dfd/timm/models/senet.py, 511 LoC, 9 entrypoints)."""

import math
from collections import OrderedDict

import torch.nn as nn
import torch.nn.functional as F

from ..ops import functional as O
from .layers import SelectAdaptivePool2d
from .registry import register_model

__all__ = ["SENet"]


def _cfg(**kwargs):
    return {
        "url": "", "num_classes": 1000, "input_size": (3, 224, 224),
        "pool_size": (7, 7), "crop_pct": 0.875, "interpolation": "bilinear",
        "mean": (0.485, 0.456, 0.406), "std": (0.229, 0.224, 0.225),
        "first_conv": "layer0.conv1", "classifier": "last_linear", **kwargs,
    }


default_cfgs = {k: _cfg() for k in [
    "senet154", "seresnet18", "seresnet34", "seresnet50", "seresnet101",
    "seresnet152", "seresnext26_32x4d_legacy", "seresnext50_32x4d_legacy",
    "seresnext101_32x4d_legacy",
]}


class SEModule(nn.Module):
    def __init__(self, channels, reduction):
        super().__init__()
        self.avg_pool = nn.AdaptiveAvgPool2d(1)
        self.fc1 = nn.Conv2d(channels, channels // reduction, kernel_size=1, padding=0)
        self.relu = nn.ReLU(inplace=True)
        self.fc2 = nn.Conv2d(channels // reduction, channels, kernel_size=1, padding=0)
        self.sigmoid = nn.Sigmoid()

    def forward(self, x):
        module_input = x
        x = self.avg_pool(x)
        x = self.fc1(x)
        x = self.relu(x)
        x = self.fc2(x)
        x = self.sigmoid(x)
        return module_input * x


class Bottleneck(nn.Module):
    def forward(self, x):
        residual = x
        out = self.conv1(x)
        out = self.bn1(out)
        out = self.relu(out)
        out = self.conv2(out)
        out = self.bn2(out)
        out = self.relu(out)
        out = self.conv3(out)
        out = self.bn3(out)
        if self.downsample is not None:
            residual = self.downsample(x)
        out = self.se_module(out) + residual
        return self.relu(out)


class SEBottleneck(Bottleneck):
    expansion = 4

    def __init__(self, inplanes, planes, groups, reduction, stride=1, downsample=None):
        super().__init__()
        self.conv1 = nn.Conv2d(inplanes, planes * 2, kernel_size=1, bias=False)
        self.bn1 = nn.BatchNorm2d(planes * 2)
        self.conv2 = nn.Conv2d(planes * 2, planes * 4, kernel_size=3, stride=stride,
                               padding=1, groups=groups, bias=False)
        self.bn2 = nn.BatchNorm2d(planes * 4)
        self.conv3 = nn.Conv2d(planes * 4, planes * 4, kernel_size=1, bias=False)
        self.bn3 = nn.BatchNorm2d(planes * 4)
        self.relu = nn.ReLU(inplace=True)
        self.se_module = SEModule(planes * 4, reduction=reduction)
        self.downsample = downsample
        self.stride = stride


class SEResNetBottleneck(Bottleneck):
    expansion = 4

    def __init__(self, inplanes, planes, groups, reduction, stride=1, downsample=None):
        super().__init__()
        self.conv1 = nn.Conv2d(inplanes, planes, kernel_size=1, bias=False, stride=stride)
        self.bn1 = nn.BatchNorm2d(planes)
        self.conv2 = nn.Conv2d(planes, planes, kernel_size=3, padding=1,
                               groups=groups, bias=False)
        self.bn2 = nn.BatchNorm2d(planes)
        self.conv3 = nn.Conv2d(planes, planes * 4, kernel_size=1, bias=False)
        self.bn3 = nn.BatchNorm2d(planes * 4)
        self.relu = nn.ReLU(inplace=True)
        self.se_module = SEModule(planes * 4, reduction=reduction)
        self.downsample = downsample
        self.stride = stride


class SEResNeXtBottleneck(Bottleneck):
    expansion = 4

    def __init__(self, inplanes, planes, groups, reduction, stride=1,
                 downsample=None, base_width=4):
        super().__init__()
        width = math.floor(planes * (base_width / 64)) * groups
        self.conv1 = nn.Conv2d(inplanes, width, kernel_size=1, bias=False, stride=1)
        self.bn1 = nn.BatchNorm2d(width)
        self.conv2 = nn.Conv2d(width, width, kernel_size=3, stride=stride,
                               padding=1, groups=groups, bias=False)
        self.bn2 = nn.BatchNorm2d(width)
        self.conv3 = nn.Conv2d(width, planes * 4, kernel_size=1, bias=False)
        self.bn3 = nn.BatchNorm2d(planes * 4)
        self.relu = nn.ReLU(inplace=True)
        self.se_module = SEModule(planes * 4, reduction=reduction)
        self.downsample = downsample
        self.stride = stride


class SEResNetBlock(nn.Module):
    expansion = 1

    def __init__(self, inplanes, planes, groups, reduction, stride=1, downsample=None):
        super().__init__()
        self.conv1 = nn.Conv2d(inplanes, planes, kernel_size=3, padding=1,
                               stride=stride, bias=False)
        self.bn1 = nn.BatchNorm2d(planes)
        self.conv2 = nn.Conv2d(planes, planes, kernel_size=3, padding=1,
                               groups=groups, bias=False)
        self.bn2 = nn.BatchNorm2d(planes)
        self.relu = nn.ReLU(inplace=True)
        self.se_module = SEModule(planes, reduction=reduction)
        self.downsample = downsample
        self.stride = stride

    def forward(self, x):
        residual = x
        out = self.conv1(x)
        out = self.bn1(out)
        out = self.relu(out)
        out = self.conv2(out)
        out = self.bn2(out)
        out = self.relu(out)
        if self.downsample is not None:
            residual = self.downsample(x)
        out = self.se_module(out) + residual
        return self.relu(out)


class SENet(nn.Module):
    def __init__(self, block, layers, groups, reduction, drop_rate=0.2,
                 in_chans=3, inplanes=128, input_3x3=True, downsample_kernel_size=3,
                 downsample_padding=1, num_classes=1000, global_pool="avg"):
        super().__init__()
        self.inplanes = inplanes
        self.num_classes = num_classes
        self.drop_rate = drop_rate
        if input_3x3:
            layer0_modules = [
                ("conv1", nn.Conv2d(in_chans, 64, 3, stride=2, padding=1, bias=False)),
                ("bn1", nn.BatchNorm2d(64)),
                ("relu1", nn.ReLU(inplace=True)),
                ("conv2", nn.Conv2d(64, 64, 3, stride=1, padding=1, bias=False)),
                ("bn2", nn.BatchNorm2d(64)),
                ("relu2", nn.ReLU(inplace=True)),
                ("conv3", nn.Conv2d(64, inplanes, 3, stride=1, padding=1, bias=False)),
                ("bn3", nn.BatchNorm2d(inplanes)),
                ("relu3", nn.ReLU(inplace=True)),
            ]
        else:
            layer0_modules = [
                ("conv1", nn.Conv2d(in_chans, inplanes, kernel_size=7, stride=2,
                                    padding=3, bias=False)),
                ("bn1", nn.BatchNorm2d(inplanes)),
                ("relu1", nn.ReLU(inplace=True)),
            ]
        self.layer0 = nn.Sequential(OrderedDict(layer0_modules))
        self.pool0 = nn.MaxPool2d(3, stride=2, ceil_mode=True)
        self.layer1 = self._make_layer(
            block, planes=64, blocks=layers[0], groups=groups, reduction=reduction,
            downsample_kernel_size=1, downsample_padding=0)
        self.layer2 = self._make_layer(
            block, planes=128, blocks=layers[1], stride=2, groups=groups,
            reduction=reduction, downsample_kernel_size=downsample_kernel_size,
            downsample_padding=downsample_padding)
        self.layer3 = self._make_layer(
            block, planes=256, blocks=layers[2], stride=2, groups=groups,
            reduction=reduction, downsample_kernel_size=downsample_kernel_size,
            downsample_padding=downsample_padding)
        self.layer4 = self._make_layer(
            block, planes=512, blocks=layers[3], stride=2, groups=groups,
            reduction=reduction, downsample_kernel_size=downsample_kernel_size,
            downsample_padding=downsample_padding)
        self.avg_pool = SelectAdaptivePool2d(pool_type=global_pool)
        self.num_features = 512 * block.expansion
        self.last_linear = nn.Linear(self.num_features, num_classes)

        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out", nonlinearity="relu")
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.constant_(m.weight, 1.0)
                nn.init.constant_(m.bias, 0.0)

    def _make_layer(self, block, planes, blocks, groups, reduction, stride=1,
                    downsample_kernel_size=1, downsample_padding=0):
        downsample = None
        if stride != 1 or self.inplanes != planes * block.expansion:
            downsample = nn.Sequential(
                nn.Conv2d(self.inplanes, planes * block.expansion,
                          kernel_size=downsample_kernel_size, stride=stride,
                          padding=downsample_padding, bias=False),
                nn.BatchNorm2d(planes * block.expansion),
            )
        layers = [block(self.inplanes, planes, groups, reduction, stride, downsample)]
        self.inplanes = planes * block.expansion
        for _ in range(1, blocks):
            layers.append(block(self.inplanes, planes, groups, reduction))
        return nn.Sequential(*layers)

    def get_classifier(self):
        return self.last_linear

    def reset_classifier(self, num_classes, global_pool="avg"):
        self.num_classes = num_classes
        self.avg_pool = SelectAdaptivePool2d(pool_type=global_pool)
        self.last_linear = nn.Linear(self.num_features, num_classes) if num_classes else None

    def forward_features(self, x):
        x = self.layer0(x)
        x = self.pool0(x)
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        x = self.layer4(x)
        return x

    def forward(self, x):
        x = self.forward_features(x)
        if self.avg_pool.pool_type == "avg":
            x = O.global_avg_pool(x)
        else:
            x = self.avg_pool(x).flatten(1)
        if self.drop_rate > 0.0:
            x = F.dropout(x, p=self.drop_rate, training=self.training)
        return self.last_linear(x)


def _make(variant, block, layers, groups, reduction, pretrained=False, **kwargs):
    model = SENet(block, layers, groups=groups, reduction=reduction, **kwargs)
    model.default_cfg = default_cfgs[variant]
    return model


@register_model
def senet154(pretrained=False, **kwargs):
    return _make("senet154", SEBottleneck, [3, 8, 36, 3], 64, 16, pretrained, **kwargs)


@register_model
def seresnet18(pretrained=False, **kwargs):
    return _make("seresnet18", SEResNetBlock, [2, 2, 2, 2], 1, 16, pretrained,
                 inplanes=64, input_3x3=False, downsample_kernel_size=1,
                 downsample_padding=0, drop_rate=0.0, **kwargs)


@register_model
def seresnet34(pretrained=False, **kwargs):
    return _make("seresnet34", SEResNetBlock, [3, 4, 6, 3], 1, 16, pretrained,
                 inplanes=64, input_3x3=False, downsample_kernel_size=1,
                 downsample_padding=0, drop_rate=0.0, **kwargs)


@register_model
def seresnet50(pretrained=False, **kwargs):
    return _make("seresnet50", SEResNetBottleneck, [3, 4, 6, 3], 1, 16, pretrained,
                 inplanes=64, input_3x3=False, downsample_kernel_size=1,
                 downsample_padding=0, drop_rate=0.0, **kwargs)


@register_model
def seresnet101(pretrained=False, **kwargs):
    return _make("seresnet101", SEResNetBottleneck, [3, 4, 23, 3], 1, 16, pretrained,
                 inplanes=64, input_3x3=False, downsample_kernel_size=1,
                 downsample_padding=0, drop_rate=0.0, **kwargs)


@register_model
def seresnet152(pretrained=False, **kwargs):
    return _make("seresnet152", SEResNetBottleneck, [3, 8, 36, 3], 1, 16, pretrained,
                 inplanes=64, input_3x3=False, downsample_kernel_size=1,
                 downsample_padding=0, drop_rate=0.0, **kwargs)


@register_model
def seresnext26_32x4d_legacy(pretrained=False, **kwargs):
    return _make("seresnext26_32x4d_legacy", SEResNeXtBottleneck, [2, 2, 2, 2], 32, 16,
                 pretrained, inplanes=64, input_3x3=False, downsample_kernel_size=1,
                 downsample_padding=0, drop_rate=0.0, **kwargs)


@register_model
def seresnext50_32x4d_legacy(pretrained=False, **kwargs):
    return _make("seresnext50_32x4d_legacy", SEResNeXtBottleneck, [3, 4, 6, 3], 32, 16,
                 pretrained, inplanes=64, input_3x3=False, downsample_kernel_size=1,
                 downsample_padding=0, drop_rate=0.0, **kwargs)


@register_model
def seresnext101_32x4d_legacy(pretrained=False, **kwargs):
    return _make("seresnext101_32x4d_legacy", SEResNeXtBottleneck, [3, 4, 23, 3], 32, 16,
                 pretrained, inplanes=64, input_3x3=False, downsample_kernel_size=1,
                 downsample_padding=0, drop_rate=0.0, **kwargs)


# reference senet.py registers these without the _legacy suffix — keep both
# names (the *_legacy aliases predate the reference-name audit)
default_cfgs["seresnext50_32x4d"] = _cfg()
default_cfgs["seresnext101_32x4d"] = _cfg()


@register_model
def seresnext50_32x4d(pretrained=False, **kwargs):
    return _make("seresnext50_32x4d", SEResNeXtBottleneck, [3, 4, 6, 3], 32, 16,
                 **kwargs)


@register_model
def seresnext101_32x4d(pretrained=False, **kwargs):
    return _make("seresnext101_32x4d", SEResNeXtBottleneck, [3, 4, 23, 3], 32, 16,
                 **kwargs)
