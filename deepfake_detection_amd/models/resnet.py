"""ResNet / ResNeXt / SE-ResNeXt family (reference dfd/timm/models/resnet.py,
1024 LoC, 40 entrypoints — the registry-filler architectures selectable via
--model).
"""

import math

import torch.nn as nn
import torch.nn.functional as F

from ..ops import functional as O
from .layers import SelectAdaptivePool2d
from .layers_extra import EcaModule
from .registry import register_model

__all__ = ["ResNet", "BasicBlock", "Bottleneck"]


def _cfg(**kwargs):
    return {
        "url": "", "num_classes": 1000, "input_size": (3, 224, 224),
        "pool_size": (7, 7), "crop_pct": 0.875, "interpolation": "bilinear",
        "mean": (0.485, 0.456, 0.406), "std": (0.229, 0.224, 0.225),
        "first_conv": "conv1", "classifier": "fc", **kwargs,
    }


default_cfgs = {k: _cfg() for k in [
    "resnet18", "resnet34", "resnet26", "resnet26d", "resnet50", "resnet50d",
    "resnet101", "resnet152", "wide_resnet50_2", "wide_resnet101_2",
    "resnext50_32x4d", "resnext50d_32x4d", "resnext101_32x4d", "resnext101_32x8d",
    "resnext101_64x4d", "seresnext26_32x4d", "seresnext26d_32x4d",
    "seresnext26t_32x4d",
]}


def get_padding(kernel_size, stride, dilation=1):
    return ((stride - 1) + dilation * (kernel_size - 1)) // 2


class SEModule(nn.Module):
    """Channel SE used by seresnext variants (reference layers/se.py:4)."""

    def __init__(self, channels, reduction_channels):
        super().__init__()
        self.fc1 = nn.Conv2d(channels, reduction_channels, kernel_size=1, bias=True)
        self.relu = nn.ReLU(inplace=True)
        self.fc2 = nn.Conv2d(reduction_channels, channels, kernel_size=1, bias=True)

    def forward(self, x):
        x_se = x.mean(dim=(2, 3), keepdim=True)
        x_se = self.fc1(x_se)
        x_se = self.relu(x_se)
        x_se = self.fc2(x_se)
        return x * x_se.sigmoid()


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, inplanes, planes, stride=1, downsample=None, cardinality=1,
                 base_width=64, use_se=False, use_eca=False, reduce_first=1, dilation=1,
                 first_dilation=None, act_layer=nn.ReLU, norm_layer=nn.BatchNorm2d):
        super().__init__()
        assert cardinality == 1 and base_width == 64
        first_planes = planes // reduce_first
        outplanes = planes * self.expansion
        first_dilation = first_dilation or dilation

        self.conv1 = nn.Conv2d(
            inplanes, first_planes, kernel_size=3, stride=stride,
            padding=first_dilation, dilation=first_dilation, bias=False)
        self.bn1 = norm_layer(first_planes)
        self.act1 = act_layer(inplace=True)
        self.conv2 = nn.Conv2d(
            first_planes, outplanes, kernel_size=3, padding=dilation,
            dilation=dilation, bias=False)
        self.bn2 = norm_layer(outplanes)
        self.se = SEModule(outplanes, planes // 4) if use_se else (
            EcaModule(outplanes) if use_eca else None)
        self.act2 = act_layer(inplace=True)
        self.downsample = downsample
        self.stride = stride
        self.dilation = dilation

    def forward(self, x):
        residual = x
        out = self.conv1(x)
        out = O.bn_act(out, self.bn1, "relu") if O.fusable_bn(self.bn1) \
            else self.act1(self.bn1(out))
        out = self.conv2(out)
        out = O.bn_act(out, self.bn2, "none") if O.fusable_bn(self.bn2) \
            else self.bn2(out)
        if self.se is not None:
            out = self.se(out)
        if self.downsample is not None:
            residual = self.downsample(x)
        out = out + residual
        return self.act2(out)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, inplanes, planes, stride=1, downsample=None, cardinality=1,
                 base_width=64, use_se=False, use_eca=False, reduce_first=1, dilation=1,
                 first_dilation=None, act_layer=nn.ReLU, norm_layer=nn.BatchNorm2d):
        super().__init__()
        width = int(math.floor(planes * (base_width / 64)) * cardinality)
        first_planes = width // reduce_first
        outplanes = planes * self.expansion
        first_dilation = first_dilation or dilation

        self.conv1 = nn.Conv2d(inplanes, first_planes, kernel_size=1, bias=False)
        self.bn1 = norm_layer(first_planes)
        self.act1 = act_layer(inplace=True)
        self.conv2 = nn.Conv2d(
            first_planes, width, kernel_size=3, stride=stride,
            padding=first_dilation, dilation=first_dilation, groups=cardinality, bias=False)
        self.bn2 = norm_layer(width)
        self.act2 = act_layer(inplace=True)
        self.conv3 = nn.Conv2d(width, outplanes, kernel_size=1, bias=False)
        self.bn3 = norm_layer(outplanes)
        self.se = SEModule(outplanes, planes // 4) if use_se else (
            EcaModule(outplanes) if use_eca else None)
        self.act3 = act_layer(inplace=True)
        self.downsample = downsample
        self.stride = stride
        self.dilation = dilation

    def forward(self, x):
        residual = x
        out = self.conv1(x)
        out = O.bn_act(out, self.bn1, "relu") if O.fusable_bn(self.bn1) \
            else self.act1(self.bn1(out))
        out = self.conv2(out)
        out = O.bn_act(out, self.bn2, "relu") if O.fusable_bn(self.bn2) \
            else self.act2(self.bn2(out))
        out = self.conv3(out)
        out = O.bn_act(out, self.bn3, "none") if O.fusable_bn(self.bn3) \
            else self.bn3(out)
        if self.se is not None:
            out = self.se(out)
        if self.downsample is not None:
            residual = self.downsample(x)
        out = out + residual
        return self.act3(out)


class ResNet(nn.Module):
    """ResNet / ResNeXt / SE-ResNeXt with stem variants (deep 3x3x3 stem,
    stem width, avg-pool downsample)."""

    def __init__(self, block, layers, num_classes=1000, in_chans=3, use_se=False,
                 use_eca=False, stem_type="",
                 cardinality=1, base_width=64, stem_width=64, deep_stem=False,
                 block_reduce_first=1, down_kernel_size=1, avg_down=False,
                 output_stride=32, act_layer=nn.ReLU, norm_layer=nn.BatchNorm2d,
                 drop_rate=0.0, global_pool="avg"):
        super().__init__()
        deep_stem = deep_stem or stem_type.startswith("deep")
        self.num_classes = num_classes
        self.inplanes = stem_width * 2 if deep_stem else 64
        self.cardinality = cardinality
        self.base_width = base_width
        self.drop_rate = drop_rate
        self.expansion = block.expansion

        if deep_stem:
            # tiered stems ramp the 3x3x3 stem widths (reference resnet.py
            # seresnext26t/tn variants): tiered (3c/4, c) narrow (c/2, c/2)
            if stem_type == "deep_tiered":
                stem_chs = (3 * stem_width // 4, stem_width)
            elif stem_type == "deep_tiered_narrow":
                stem_chs = (stem_width // 2, stem_width // 2)
            else:
                stem_chs = (stem_width, stem_width)
            self.conv1 = nn.Sequential(
                nn.Conv2d(in_chans, stem_chs[0], 3, stride=2, padding=1, bias=False),
                norm_layer(stem_chs[0]),
                act_layer(inplace=True),
                nn.Conv2d(stem_chs[0], stem_chs[1], 3, stride=1, padding=1, bias=False),
                norm_layer(stem_chs[1]),
                act_layer(inplace=True),
                nn.Conv2d(stem_chs[1], self.inplanes, 3, stride=1, padding=1, bias=False))
        else:
            self.conv1 = nn.Conv2d(in_chans, self.inplanes, kernel_size=7, stride=2,
                                   padding=3, bias=False)
        self.bn1 = norm_layer(self.inplanes)
        self.act1 = act_layer(inplace=True)
        self.maxpool = nn.MaxPool2d(kernel_size=3, stride=2, padding=1)

        if output_stride == 32:
            strides, dilations = [1, 2, 2, 2], [1] * 4
        elif output_stride == 16:
            strides, dilations = [1, 2, 2, 1], [1, 1, 1, 2]
        else:
            strides, dilations = [1, 2, 1, 1], [1, 1, 2, 4]

        la = dict(use_se=use_se, use_eca=use_eca, reduce_first=block_reduce_first,
                  act_layer=act_layer, norm_layer=norm_layer,
                  avg_down=avg_down, down_kernel_size=down_kernel_size)
        self.layer1 = self._make_layer(block, 64, layers[0], strides[0], dilations[0], **la)
        self.layer2 = self._make_layer(block, 128, layers[1], strides[1], dilations[1], **la)
        self.layer3 = self._make_layer(block, 256, layers[2], strides[2], dilations[2], **la)
        self.layer4 = self._make_layer(block, 512, layers[3], strides[3], dilations[3], **la)

        self.global_pool = SelectAdaptivePool2d(pool_type=global_pool)
        self.num_features = 512 * block.expansion
        self.fc = nn.Linear(self.num_features * self.global_pool.feat_mult(), num_classes)

        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out", nonlinearity="relu")
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.constant_(m.weight, 1.0)
                nn.init.constant_(m.bias, 0.0)

    def _make_layer(self, block, planes, blocks, stride=1, dilation=1, use_se=False,
                    use_eca=False,
                    reduce_first=1, act_layer=nn.ReLU, norm_layer=nn.BatchNorm2d,
                    avg_down=False, down_kernel_size=1):
        downsample = None
        down_kernel_size = 1 if stride == 1 and dilation == 1 else down_kernel_size
        if stride != 1 or self.inplanes != planes * block.expansion:
            downsample_padding = get_padding(down_kernel_size, stride)
            downsample_layers = []
            conv_stride = stride
            if avg_down:
                avg_stride = stride if dilation == 1 else 1
                conv_stride = 1
                avg_pool = nn.AvgPool2d(avg_stride, avg_stride, ceil_mode=True,
                                        count_include_pad=False) if avg_stride > 1 \
                    else nn.Identity()
                downsample_layers = [avg_pool]
            downsample_layers += [
                nn.Conv2d(self.inplanes, planes * block.expansion, down_kernel_size,
                          stride=conv_stride, padding=downsample_padding, bias=False),
                norm_layer(planes * block.expansion)]
            downsample = nn.Sequential(*downsample_layers)

        first_dilation = 1 if dilation in (1, 2) else 2
        layers = [block(
            self.inplanes, planes, stride, downsample, cardinality=self.cardinality,
            base_width=self.base_width, use_se=use_se, use_eca=use_eca, reduce_first=reduce_first,
            dilation=dilation, first_dilation=first_dilation, act_layer=act_layer,
            norm_layer=norm_layer)]
        self.inplanes = planes * block.expansion
        for _ in range(1, blocks):
            layers.append(block(
                self.inplanes, planes, use_se=use_se, use_eca=use_eca, reduce_first=reduce_first,
                cardinality=self.cardinality, base_width=self.base_width,
                dilation=dilation, act_layer=act_layer, norm_layer=norm_layer))
        return nn.Sequential(*layers)

    def get_classifier(self):
        return self.fc

    def reset_classifier(self, num_classes, global_pool="avg"):
        self.num_classes = num_classes
        self.global_pool = SelectAdaptivePool2d(pool_type=global_pool)
        self.fc = nn.Linear(
            self.num_features * self.global_pool.feat_mult(), num_classes) if num_classes else None

    def forward_features(self, x):
        x = self.conv1(x)
        x = self.bn1(x)
        x = self.act1(x)
        x = self.maxpool(x)
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        x = self.layer4(x)
        return x

    def forward(self, x):
        x = self.forward_features(x)
        if self.global_pool.pool_type == "avg":
            x = O.global_avg_pool(x)
        else:
            x = self.global_pool(x).flatten(1)
        if self.drop_rate:
            x = F.dropout(x, p=self.drop_rate, training=self.training)
        return self.fc(x)


def _make(variant, block, layers, pretrained=False, **kwargs):
    model = ResNet(block, layers, **kwargs)
    model.default_cfg = default_cfgs[variant]
    return model


@register_model
def resnet18(pretrained=False, **kwargs):
    return _make("resnet18", BasicBlock, [2, 2, 2, 2], pretrained, **kwargs)


@register_model
def resnet34(pretrained=False, **kwargs):
    return _make("resnet34", BasicBlock, [3, 4, 6, 3], pretrained, **kwargs)


@register_model
def resnet26(pretrained=False, **kwargs):
    return _make("resnet26", Bottleneck, [2, 2, 2, 2], pretrained, **kwargs)


@register_model
def resnet26d(pretrained=False, **kwargs):
    return _make("resnet26d", Bottleneck, [2, 2, 2, 2], pretrained,
                 stem_width=32, deep_stem=True, avg_down=True, **kwargs)


@register_model
def resnet50(pretrained=False, **kwargs):
    return _make("resnet50", Bottleneck, [3, 4, 6, 3], pretrained, **kwargs)


@register_model
def resnet50d(pretrained=False, **kwargs):
    return _make("resnet50d", Bottleneck, [3, 4, 6, 3], pretrained,
                 stem_width=32, deep_stem=True, avg_down=True, **kwargs)


@register_model
def resnet101(pretrained=False, **kwargs):
    return _make("resnet101", Bottleneck, [3, 4, 23, 3], pretrained, **kwargs)


@register_model
def resnet152(pretrained=False, **kwargs):
    return _make("resnet152", Bottleneck, [3, 8, 36, 3], pretrained, **kwargs)


@register_model
def wide_resnet50_2(pretrained=False, **kwargs):
    return _make("wide_resnet50_2", Bottleneck, [3, 4, 6, 3], pretrained,
                 base_width=128, **kwargs)


@register_model
def wide_resnet101_2(pretrained=False, **kwargs):
    return _make("wide_resnet101_2", Bottleneck, [3, 4, 23, 3], pretrained,
                 base_width=128, **kwargs)


@register_model
def resnext50_32x4d(pretrained=False, **kwargs):
    return _make("resnext50_32x4d", Bottleneck, [3, 4, 6, 3], pretrained,
                 cardinality=32, base_width=4, **kwargs)


@register_model
def resnext50d_32x4d(pretrained=False, **kwargs):
    return _make("resnext50d_32x4d", Bottleneck, [3, 4, 6, 3], pretrained,
                 cardinality=32, base_width=4, stem_width=32, deep_stem=True,
                 avg_down=True, **kwargs)


@register_model
def resnext101_32x4d(pretrained=False, **kwargs):
    return _make("resnext101_32x4d", Bottleneck, [3, 4, 23, 3], pretrained,
                 cardinality=32, base_width=4, **kwargs)


@register_model
def resnext101_32x8d(pretrained=False, **kwargs):
    return _make("resnext101_32x8d", Bottleneck, [3, 4, 23, 3], pretrained,
                 cardinality=32, base_width=8, **kwargs)


@register_model
def resnext101_64x4d(pretrained=False, **kwargs):
    return _make("resnext101_64x4d", Bottleneck, [3, 4, 23, 3], pretrained,
                 cardinality=64, base_width=4, **kwargs)


@register_model
def seresnext26_32x4d(pretrained=False, **kwargs):
    return _make("seresnext26_32x4d", Bottleneck, [2, 2, 2, 2], pretrained,
                 cardinality=32, base_width=4, use_se=True, **kwargs)


@register_model
def seresnext26d_32x4d(pretrained=False, **kwargs):
    return _make("seresnext26d_32x4d", Bottleneck, [2, 2, 2, 2], pretrained,
                 cardinality=32, base_width=4, use_se=True, stem_width=32,
                 deep_stem=True, avg_down=True, **kwargs)


# ---------------------------------------------------------------------------
# Weight-variant aliases (torchvision / Instagram / semi- and
# semi-weakly-supervised pretrain sets) and ECA / tiered-stem variants
# (reference resnet.py:480-1024) — same architectures, distinct names and
# pretrained cfgs.
# ---------------------------------------------------------------------------

for _n in ["tv_resnet34", "tv_resnet50", "tv_resnext50_32x4d",
           "ig_resnext101_32x8d", "ig_resnext101_32x16d", "ig_resnext101_32x32d",
           "ig_resnext101_32x48d", "ssl_resnet18", "ssl_resnet50",
           "ssl_resnext50_32x4d", "ssl_resnext101_32x4d", "ssl_resnext101_32x8d",
           "ssl_resnext101_32x16d", "swsl_resnet18", "swsl_resnet50",
           "swsl_resnext50_32x4d", "swsl_resnext101_32x4d", "swsl_resnext101_32x8d",
           "swsl_resnext101_32x16d", "seresnext26t_32x4d", "seresnext26tn_32x4d",
           "ecaresnext26tn_32x4d", "ecaresnet18", "ecaresnet50"]:
    default_cfgs.setdefault(_n, _cfg())


@register_model
def tv_resnet34(pretrained=False, **kwargs):
    return _make("tv_resnet34", BasicBlock, [3, 4, 6, 3], pretrained, **kwargs)


@register_model
def tv_resnet50(pretrained=False, **kwargs):
    return _make("tv_resnet50", Bottleneck, [3, 4, 6, 3], pretrained, **kwargs)


@register_model
def tv_resnext50_32x4d(pretrained=False, **kwargs):
    return _make("tv_resnext50_32x4d", Bottleneck, [3, 4, 6, 3], pretrained,
                 cardinality=32, base_width=4, **kwargs)


def _resnext101(variant, width, pretrained=False, **kwargs):
    return _make(variant, Bottleneck, [3, 4, 23, 3], pretrained,
                 cardinality=32, base_width=width, **kwargs)


@register_model
def ig_resnext101_32x8d(pretrained=False, **kwargs):
    return _resnext101("ig_resnext101_32x8d", 8, pretrained, **kwargs)


@register_model
def ig_resnext101_32x16d(pretrained=False, **kwargs):
    return _resnext101("ig_resnext101_32x16d", 16, pretrained, **kwargs)


@register_model
def ig_resnext101_32x32d(pretrained=False, **kwargs):
    return _resnext101("ig_resnext101_32x32d", 32, pretrained, **kwargs)


@register_model
def ig_resnext101_32x48d(pretrained=False, **kwargs):
    return _resnext101("ig_resnext101_32x48d", 48, pretrained, **kwargs)


@register_model
def ssl_resnet18(pretrained=False, **kwargs):
    return _make("ssl_resnet18", BasicBlock, [2, 2, 2, 2], pretrained, **kwargs)


@register_model
def ssl_resnet50(pretrained=False, **kwargs):
    return _make("ssl_resnet50", Bottleneck, [3, 4, 6, 3], pretrained, **kwargs)


@register_model
def ssl_resnext50_32x4d(pretrained=False, **kwargs):
    return _make("ssl_resnext50_32x4d", Bottleneck, [3, 4, 6, 3], pretrained,
                 cardinality=32, base_width=4, **kwargs)


@register_model
def ssl_resnext101_32x4d(pretrained=False, **kwargs):
    return _resnext101("ssl_resnext101_32x4d", 4, pretrained, **kwargs)


@register_model
def ssl_resnext101_32x8d(pretrained=False, **kwargs):
    return _resnext101("ssl_resnext101_32x8d", 8, pretrained, **kwargs)


@register_model
def ssl_resnext101_32x16d(pretrained=False, **kwargs):
    return _resnext101("ssl_resnext101_32x16d", 16, pretrained, **kwargs)


@register_model
def swsl_resnet18(pretrained=False, **kwargs):
    return _make("swsl_resnet18", BasicBlock, [2, 2, 2, 2], pretrained, **kwargs)


@register_model
def swsl_resnet50(pretrained=False, **kwargs):
    return _make("swsl_resnet50", Bottleneck, [3, 4, 6, 3], pretrained, **kwargs)


@register_model
def swsl_resnext50_32x4d(pretrained=False, **kwargs):
    return _make("swsl_resnext50_32x4d", Bottleneck, [3, 4, 6, 3], pretrained,
                 cardinality=32, base_width=4, **kwargs)


@register_model
def swsl_resnext101_32x4d(pretrained=False, **kwargs):
    return _resnext101("swsl_resnext101_32x4d", 4, pretrained, **kwargs)


@register_model
def swsl_resnext101_32x8d(pretrained=False, **kwargs):
    return _resnext101("swsl_resnext101_32x8d", 8, pretrained, **kwargs)


@register_model
def swsl_resnext101_32x16d(pretrained=False, **kwargs):
    return _resnext101("swsl_resnext101_32x16d", 16, pretrained, **kwargs)


@register_model
def seresnext26t_32x4d(pretrained=False, **kwargs):
    """SE-ResNeXt-26 with deep tiered stem (reference resnet.py)."""
    return _make("seresnext26t_32x4d", Bottleneck, [2, 2, 2, 2], pretrained,
                 cardinality=32, base_width=4, use_se=True, stem_width=64,
                 stem_type="deep_tiered", avg_down=True, **kwargs)


@register_model
def seresnext26tn_32x4d(pretrained=False, **kwargs):
    """SE-ResNeXt-26 with narrow tiered stem."""
    return _make("seresnext26tn_32x4d", Bottleneck, [2, 2, 2, 2], pretrained,
                 cardinality=32, base_width=4, use_se=True, stem_width=64,
                 stem_type="deep_tiered_narrow", avg_down=True, **kwargs)


@register_model
def ecaresnext26tn_32x4d(pretrained=False, **kwargs):
    """ECA-ResNeXt-26 with narrow tiered stem (ECA instead of SE)."""
    return _make("ecaresnext26tn_32x4d", Bottleneck, [2, 2, 2, 2], pretrained,
                 cardinality=32, base_width=4, use_eca=True, stem_width=64,
                 stem_type="deep_tiered_narrow", avg_down=True, **kwargs)


@register_model
def ecaresnet18(pretrained=False, **kwargs):
    return _make("ecaresnet18", BasicBlock, [2, 2, 2, 2], pretrained,
                 use_eca=True, **kwargs)


@register_model
def ecaresnet50(pretrained=False, **kwargs):
    return _make("ecaresnet50", Bottleneck, [3, 4, 6, 3], pretrained,
                 use_eca=True, **kwargs)
