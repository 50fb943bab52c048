"""PNASNet-5-Large — capability parity with reference
dfd/timm/models/pnasnet.py (398 LoC, 1 entrypoint). Liu et al., ECCV 2018
(progressive NAS); the single discovered cell, stacked 12 times with two
stem cells. Module names match the published checkpoint.
"""

from collections import OrderedDict

import torch
import torch.nn as nn
import torch.nn.functional as F

from .layers import SelectAdaptivePool2d
from .registry import register_model

__all__ = ["PNASNet5Large"]


def _cfg(**kwargs):
    return {
        "url": "", "num_classes": 1000, "input_size": (3, 331, 331),
        "pool_size": (11, 11), "crop_pct": 0.875, "interpolation": "bicubic",
        "mean": (0.5, 0.5, 0.5), "std": (0.5, 0.5, 0.5),
        "first_conv": "conv_0.conv", "classifier": "last_linear",
        "label_offset": 1, **kwargs,
    }


default_cfgs = {"pnasnet5large": _cfg()}


class MaxPool(nn.Module):
    """3x3 max pool; optional TF-style asymmetric zero pad (pad top-left,
    crop after) used in the zero_pad reduction cell."""

    def __init__(self, kernel_size, stride=1, padding=1, zero_pad=False):
        super().__init__()
        self.zero_pad = nn.ZeroPad2d((1, 0, 1, 0)) if zero_pad else None
        self.pool = nn.MaxPool2d(kernel_size, stride=stride, padding=padding)

    def forward(self, x):
        if self.zero_pad:
            return self.pool(self.zero_pad(x))[:, :, 1:, 1:]
        return self.pool(x)


class SeparableConv2d(nn.Module):
    def __init__(self, in_channels, out_channels, dw_kernel_size, dw_stride, dw_padding):
        super().__init__()
        self.depthwise_conv2d = nn.Conv2d(
            in_channels, in_channels, dw_kernel_size, stride=dw_stride,
            padding=dw_padding, groups=in_channels, bias=False)
        self.pointwise_conv2d = nn.Conv2d(in_channels, out_channels, 1, bias=False)

    def forward(self, x):
        return self.pointwise_conv2d(self.depthwise_conv2d(x))


class BranchSeparables(nn.Module):
    """relu -> sep(k, stride) -> bn -> relu -> sep(k, 1) -> bn."""

    def __init__(self, in_channels, out_channels, kernel_size, stride=1,
                 stem_cell=False, zero_pad=False):
        super().__init__()
        padding = kernel_size // 2
        middle = out_channels if stem_cell else in_channels
        self.zero_pad = nn.ZeroPad2d((1, 0, 1, 0)) if zero_pad else None
        self.relu_1 = nn.ReLU()
        self.separable_1 = SeparableConv2d(in_channels, middle, kernel_size,
                                           dw_stride=stride, dw_padding=padding)
        self.bn_sep_1 = nn.BatchNorm2d(middle, eps=0.001)
        self.relu_2 = nn.ReLU()
        self.separable_2 = SeparableConv2d(middle, out_channels, kernel_size,
                                           dw_stride=1, dw_padding=padding)
        self.bn_sep_2 = nn.BatchNorm2d(out_channels, eps=0.001)

    def forward(self, x):
        x = self.relu_1(x)
        if self.zero_pad:
            x = self.separable_1(self.zero_pad(x))[:, :, 1:, 1:].contiguous()
        else:
            x = self.separable_1(x)
        x = self.bn_sep_1(x)
        x = self.relu_2(x)
        return self.bn_sep_2(self.separable_2(x))


class ReluConvBn(nn.Module):
    def __init__(self, in_channels, out_channels, kernel_size, stride=1):
        super().__init__()
        self.relu = nn.ReLU()
        self.conv = nn.Conv2d(in_channels, out_channels, kernel_size,
                              stride=stride, bias=False)
        self.bn = nn.BatchNorm2d(out_channels, eps=0.001)

    def forward(self, x):
        return self.bn(self.conv(self.relu(x)))


class FactorizedReduction(nn.Module):
    """Halve spatial size with two shifted stride-2 1x1 paths."""

    def __init__(self, in_channels, out_channels):
        super().__init__()
        self.relu = nn.ReLU()
        self.path_1 = nn.Sequential(OrderedDict([
            ("avgpool", nn.AvgPool2d(1, stride=2, count_include_pad=False)),
            ("conv", nn.Conv2d(in_channels, out_channels // 2, 1, bias=False)),
        ]))
        self.path_2 = nn.Sequential(OrderedDict([
            ("pad", nn.ZeroPad2d((0, 1, 0, 1))),
            ("avgpool", nn.AvgPool2d(1, stride=2, count_include_pad=False)),
            ("conv", nn.Conv2d(in_channels, out_channels // 2, 1, bias=False)),
        ]))
        self.final_path_bn = nn.BatchNorm2d(out_channels, eps=0.001)

    def forward(self, x):
        x = self.relu(x)
        p1 = self.path_1(x)
        p2 = self.path_2.pad(x)[:, :, 1:, 1:]
        p2 = self.path_2.conv(self.path_2.avgpool(p2))
        return self.final_path_bn(torch.cat([p1, p2], 1))


class CellBase(nn.Module):
    """The PNAS cell: 5 combination iterations, outputs concatenated."""

    def cell_forward(self, x_left, x_right):
        c0 = self.comb_iter_0_left(x_left) + self.comb_iter_0_right(x_left)
        c1 = self.comb_iter_1_left(x_right) + self.comb_iter_1_right(x_right)
        c2 = self.comb_iter_2_left(x_right) + self.comb_iter_2_right(x_right)
        c3 = self.comb_iter_3_left(c2) + self.comb_iter_3_right(x_right)
        c4 = self.comb_iter_4_left(x_left) + (
            self.comb_iter_4_right(x_right) if self.comb_iter_4_right else x_right)
        return torch.cat([c0, c1, c2, c3, c4], 1)


class CellStem0(CellBase):
    def __init__(self, in_chs_left, out_chs_left, in_chs_right, out_chs_right):
        super().__init__()
        self.conv_1x1 = ReluConvBn(in_chs_right, out_chs_right, 1)
        self.comb_iter_0_left = BranchSeparables(in_chs_left, out_chs_left, 5,
                                                 stride=2, stem_cell=True)
        self.comb_iter_0_right = nn.Sequential(OrderedDict([
            ("max_pool", MaxPool(3, stride=2)),
            ("conv", nn.Conv2d(in_chs_left, out_chs_left, 1, bias=False)),
            ("bn", nn.BatchNorm2d(out_chs_left, eps=0.001)),
        ]))
        self.comb_iter_1_left = BranchSeparables(out_chs_right, out_chs_right, 7, stride=2)
        self.comb_iter_1_right = MaxPool(3, stride=2)
        self.comb_iter_2_left = BranchSeparables(out_chs_right, out_chs_right, 5, stride=2)
        self.comb_iter_2_right = BranchSeparables(out_chs_right, out_chs_right, 3, stride=2)
        self.comb_iter_3_left = BranchSeparables(out_chs_right, out_chs_right, 3)
        self.comb_iter_3_right = MaxPool(3, stride=2)
        self.comb_iter_4_left = BranchSeparables(in_chs_right, out_chs_right, 3,
                                                 stride=2, stem_cell=True)
        self.comb_iter_4_right = ReluConvBn(out_chs_right, out_chs_right, 1, stride=2)

    def forward(self, x_left):
        x_right = self.conv_1x1(x_left)
        return self.cell_forward(x_left, x_right)


class Cell(CellBase):
    def __init__(self, in_chs_left, out_chs_left, in_chs_right, out_chs_right,
                 is_reduction=False, zero_pad=False, match_prev_layer_dims=False):
        super().__init__()
        stride = 2 if is_reduction else 1
        if match_prev_layer_dims:
            self.conv_prev_1x1 = FactorizedReduction(in_chs_left, out_chs_left)
        else:
            self.conv_prev_1x1 = ReluConvBn(in_chs_left, out_chs_left, 1)
        self.conv_1x1 = ReluConvBn(in_chs_right, out_chs_right, 1)
        self.comb_iter_0_left = BranchSeparables(out_chs_left, out_chs_left, 5,
                                                 stride=stride, zero_pad=zero_pad)
        self.comb_iter_0_right = MaxPool(3, stride=stride, zero_pad=zero_pad)
        self.comb_iter_1_left = BranchSeparables(out_chs_right, out_chs_right, 7,
                                                 stride=stride, zero_pad=zero_pad)
        self.comb_iter_1_right = MaxPool(3, stride=stride, zero_pad=zero_pad)
        self.comb_iter_2_left = BranchSeparables(out_chs_right, out_chs_right, 5,
                                                 stride=stride, zero_pad=zero_pad)
        self.comb_iter_2_right = BranchSeparables(out_chs_right, out_chs_right, 3,
                                                  stride=stride, zero_pad=zero_pad)
        self.comb_iter_3_left = BranchSeparables(out_chs_right, out_chs_right, 3)
        self.comb_iter_3_right = MaxPool(3, stride=stride, zero_pad=zero_pad)
        self.comb_iter_4_left = BranchSeparables(out_chs_left, out_chs_left, 3,
                                                 stride=stride, zero_pad=zero_pad)
        self.comb_iter_4_right = ReluConvBn(
            out_chs_right, out_chs_right, 1, stride=stride) if is_reduction else None

    def forward(self, x_left, x_right):
        return self.cell_forward(self.conv_prev_1x1(x_left), self.conv_1x1(x_right))


class PNASNet5Large(nn.Module):
    def __init__(self, num_classes=1000, in_chans=3, drop_rate=0.5, global_pool="avg"):
        super().__init__()
        self.num_classes = num_classes
        self.num_features = 4320
        self.drop_rate = drop_rate

        self.conv_0 = nn.Sequential(OrderedDict([
            ("conv", nn.Conv2d(in_chans, 96, 3, stride=2, bias=False)),
            ("bn", nn.BatchNorm2d(96, eps=0.001)),
        ]))
        self.cell_stem_0 = CellStem0(96, 54, 96, 54)
        self.cell_stem_1 = Cell(96, 108, 270, 108, match_prev_layer_dims=True,
                                is_reduction=True)
        self.cell_0 = Cell(270, 216, 540, 216, match_prev_layer_dims=True)
        self.cell_1 = Cell(540, 216, 1080, 216)
        self.cell_2 = Cell(1080, 216, 1080, 216)
        self.cell_3 = Cell(1080, 216, 1080, 216)
        self.cell_4 = Cell(1080, 432, 1080, 432, is_reduction=True, zero_pad=True)
        self.cell_5 = Cell(1080, 432, 2160, 432, match_prev_layer_dims=True)
        self.cell_6 = Cell(2160, 432, 2160, 432)
        self.cell_7 = Cell(2160, 432, 2160, 432)
        self.cell_8 = Cell(2160, 864, 2160, 864, is_reduction=True)
        self.cell_9 = Cell(2160, 864, 4320, 864, match_prev_layer_dims=True)
        self.cell_10 = Cell(4320, 864, 4320, 864)
        self.cell_11 = Cell(4320, 864, 4320, 864)
        self.relu = nn.ReLU()
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
        x0 = self.conv_0(x)
        s0 = self.cell_stem_0(x0)
        s1 = self.cell_stem_1(x0, s0)
        prev, cur = s0, s1
        for i in range(12):
            nxt = getattr(self, f"cell_{i}")(prev, cur)
            prev, cur = cur, nxt
        return self.relu(cur)

    def forward(self, x):
        x = self.forward_features(x)
        x = self.global_pool(x).flatten(1)
        if self.drop_rate > 0:
            x = F.dropout(x, p=self.drop_rate, training=self.training)
        return self.last_linear(x)


@register_model
def pnasnet5large(pretrained=False, **kwargs):
    model = PNASNet5Large(**kwargs)
    model.default_cfg = default_cfgs["pnasnet5large"]
    return model
