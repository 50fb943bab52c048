"""NASNet-A-Large (6 @ 4032) — capability parity with reference
dfd/timm/models/nasnet.py (620 LoC, 1 entrypoint). Zoph et al., CVPR 2018.
Normal/reduction cells discovered by NAS, stacked 6-per-stage with two stem
cells. Module names match the published checkpoint.
"""

from collections import OrderedDict

import torch
import torch.nn as nn
import torch.nn.functional as F

from .layers import SelectAdaptivePool2d
from .registry import register_model

__all__ = ["NASNetALarge"]


def _cfg(**kwargs):
    return {
        "url": "", "num_classes": 1000, "input_size": (3, 331, 331),
        "pool_size": (11, 11), "crop_pct": 0.875, "interpolation": "bicubic",
        "mean": (0.5, 0.5, 0.5), "std": (0.5, 0.5, 0.5),
        "first_conv": "conv0.conv", "classifier": "last_linear",
        "label_offset": 1, **kwargs,
    }


default_cfgs = {"nasnetalarge": _cfg()}


def _relu_conv_bn(in_ch, out_ch):
    return nn.Sequential(OrderedDict([
        ("relu", nn.ReLU()),
        ("conv", nn.Conv2d(in_ch, out_ch, 1, stride=1, bias=False)),
        ("bn", nn.BatchNorm2d(out_ch, eps=0.001, momentum=0.1)),
    ]))


class MaxPoolPad(nn.Module):
    """TF-compatible stride-2 max pool: pad top-left, pool, crop."""

    def __init__(self):
        super().__init__()
        self.pad = nn.ZeroPad2d((1, 0, 1, 0))
        self.pool = nn.MaxPool2d(3, stride=2, padding=1)

    def forward(self, x):
        return self.pool(self.pad(x))[:, :, 1:, 1:].contiguous()


class AvgPoolPad(nn.Module):
    def __init__(self, stride=2, padding=1):
        super().__init__()
        self.pad = nn.ZeroPad2d((1, 0, 1, 0))
        self.pool = nn.AvgPool2d(3, stride=stride, padding=padding, count_include_pad=False)

    def forward(self, x):
        return self.pool(self.pad(x))[:, :, 1:, 1:].contiguous()


class SeparableConv2d(nn.Module):
    def __init__(self, in_channels, out_channels, dw_kernel, dw_stride, dw_padding,
                 bias=False):
        super().__init__()
        self.depthwise_conv2d = nn.Conv2d(
            in_channels, in_channels, dw_kernel, stride=dw_stride,
            padding=dw_padding, bias=bias, groups=in_channels)
        self.pointwise_conv2d = nn.Conv2d(in_channels, out_channels, 1, stride=1, bias=bias)

    def forward(self, x):
        return self.pointwise_conv2d(self.depthwise_conv2d(x))


class BranchSeparables(nn.Module):
    """relu -> sep(k, stride) -> bn -> relu -> sep(k, 1) -> bn (C -> C -> out)."""

    def __init__(self, in_channels, out_channels, kernel_size, stride, padding,
                 bias=False):
        super().__init__()
        self.relu = nn.ReLU()
        self.separable_1 = SeparableConv2d(in_channels, in_channels, kernel_size,
                                           stride, padding, bias=bias)
        self.bn_sep_1 = nn.BatchNorm2d(in_channels, eps=0.001, momentum=0.1)
        self.relu1 = nn.ReLU()
        self.separable_2 = SeparableConv2d(in_channels, out_channels, kernel_size,
                                           1, padding, bias=bias)
        self.bn_sep_2 = nn.BatchNorm2d(out_channels, eps=0.001, momentum=0.1)

    def forward(self, x):
        x = self.relu(x)
        x = self.bn_sep_1(self.separable_1(x))
        x = self.relu1(x)
        return self.bn_sep_2(self.separable_2(x))


class BranchSeparablesStem(nn.Module):
    """Stem variant: first separable maps stem_size -> out directly."""

    def __init__(self, in_channels, out_channels, kernel_size, stride, padding,
                 bias=False):
        super().__init__()
        self.relu = nn.ReLU()
        self.separable_1 = SeparableConv2d(in_channels, out_channels, kernel_size,
                                           stride, padding, bias=bias)
        self.bn_sep_1 = nn.BatchNorm2d(out_channels, eps=0.001, momentum=0.1)
        self.relu1 = nn.ReLU()
        self.separable_2 = SeparableConv2d(out_channels, out_channels, kernel_size,
                                           1, padding, bias=bias)
        self.bn_sep_2 = nn.BatchNorm2d(out_channels, eps=0.001, momentum=0.1)

    def forward(self, x):
        x = self.relu(x)
        x = self.bn_sep_1(self.separable_1(x))
        x = self.relu1(x)
        return self.bn_sep_2(self.separable_2(x))


class BranchSeparablesReduction(BranchSeparables):
    """Reduction variant: asymmetric zero-pad before the strided separable."""

    def __init__(self, in_channels, out_channels, kernel_size, stride, padding,
                 z_padding=1, bias=False):
        super().__init__(in_channels, out_channels, kernel_size, stride, padding, bias)
        self.padding = nn.ZeroPad2d((z_padding, 0, z_padding, 0))

    def forward(self, x):
        x = self.relu(x)
        x = self.padding(x)
        x = self.separable_1(x)[:, :, 1:, 1:].contiguous()
        x = self.bn_sep_1(x)
        x = self.relu1(x)
        return self.bn_sep_2(self.separable_2(x))


class _FactorizedPath(nn.Module):
    """relu -> two shifted stride-2 1x1 paths -> concat -> bn (module names
    path_1/path_2/final_path_bn for checkpoint parity)."""

    def __init__(self, in_ch, out_half):
        super().__init__()
        self.relu = nn.ReLU()
        self.path_1 = nn.Sequential(OrderedDict([
            ("avgpool", nn.AvgPool2d(1, stride=2, count_include_pad=False)),
            ("conv", nn.Conv2d(in_ch, out_half, 1, stride=1, bias=False)),
        ]))
        self.path_2 = nn.Sequential(OrderedDict([
            ("pad", nn.ZeroPad2d((0, 1, 0, 1))),
            ("avgpool", nn.AvgPool2d(1, stride=2, count_include_pad=False)),
            ("conv", nn.Conv2d(in_ch, out_half, 1, stride=1, bias=False)),
        ]))
        self.final_path_bn = nn.BatchNorm2d(out_half * 2, eps=0.001, momentum=0.1)

    def forward(self, x):
        x = self.relu(x)
        p1 = self.path_1(x)
        p2 = self.path_2.pad(x)[:, :, 1:, 1:]
        p2 = self.path_2.conv(self.path_2.avgpool(p2))
        return self.final_path_bn(torch.cat([p1, p2], 1))


class CellStem0(nn.Module):
    def __init__(self, stem_size, num_channels=42):
        super().__init__()
        c = num_channels
        self.conv_1x1 = _relu_conv_bn(stem_size, c)
        self.comb_iter_0_left = BranchSeparables(c, c, 5, 2, 2)
        self.comb_iter_0_right = BranchSeparablesStem(stem_size, c, 7, 2, 3)
        self.comb_iter_1_left = nn.MaxPool2d(3, stride=2, padding=1)
        self.comb_iter_1_right = BranchSeparablesStem(stem_size, c, 7, 2, 3)
        self.comb_iter_2_left = nn.AvgPool2d(3, stride=2, padding=1, count_include_pad=False)
        self.comb_iter_2_right = BranchSeparablesStem(stem_size, c, 5, 2, 2)
        self.comb_iter_3_right = nn.AvgPool2d(3, stride=1, padding=1, count_include_pad=False)
        self.comb_iter_4_left = BranchSeparables(c, c, 3, 1, 1)
        self.comb_iter_4_right = nn.MaxPool2d(3, stride=2, padding=1)

    def forward(self, x):
        x1 = self.conv_1x1(x)
        c0 = self.comb_iter_0_left(x1) + self.comb_iter_0_right(x)
        c1 = self.comb_iter_1_left(x1) + self.comb_iter_1_right(x)
        c2 = self.comb_iter_2_left(x1) + self.comb_iter_2_right(x)
        c3 = self.comb_iter_3_right(c0) + c1
        c4 = self.comb_iter_4_left(c0) + self.comb_iter_4_right(x1)
        return torch.cat([c1, c2, c3, c4], 1)


class CellStem1(nn.Module):
    def __init__(self, stem_size, num_channels):
        super().__init__()
        c = num_channels
        self.conv_1x1 = _relu_conv_bn(2 * c, c)
        self._fp = _FactorizedPath(stem_size, c // 2)
        self.comb_iter_0_left = BranchSeparables(c, c, 5, 2, 2)
        self.comb_iter_0_right = BranchSeparables(c, c, 7, 2, 3)
        self.comb_iter_1_left = nn.MaxPool2d(3, stride=2, padding=1)
        self.comb_iter_1_right = BranchSeparables(c, c, 7, 2, 3)
        self.comb_iter_2_left = nn.AvgPool2d(3, stride=2, padding=1, count_include_pad=False)
        self.comb_iter_2_right = BranchSeparables(c, c, 5, 2, 2)
        self.comb_iter_3_right = nn.AvgPool2d(3, stride=1, padding=1, count_include_pad=False)
        self.comb_iter_4_left = BranchSeparables(c, c, 3, 1, 1)
        self.comb_iter_4_right = nn.MaxPool2d(3, stride=2, padding=1)

    def forward(self, x_conv0, x_stem_0):
        x_left = self.conv_1x1(x_stem_0)
        x_right = self._fp(x_conv0)
        c0 = self.comb_iter_0_left(x_left) + self.comb_iter_0_right(x_right)
        c1 = self.comb_iter_1_left(x_left) + self.comb_iter_1_right(x_right)
        c2 = self.comb_iter_2_left(x_left) + self.comb_iter_2_right(x_right)
        c3 = self.comb_iter_3_right(c0) + c1
        c4 = self.comb_iter_4_left(c0) + self.comb_iter_4_right(x_left)
        return torch.cat([c1, c2, c3, c4], 1)


class FirstCell(nn.Module):
    """Normal cell whose previous-layer input needs a factorized reduction."""

    def __init__(self, in_chs_left, out_chs_left, in_chs_right, out_chs_right):
        super().__init__()
        self.conv_1x1 = _relu_conv_bn(in_chs_right, out_chs_right)
        self._fp = _FactorizedPath(in_chs_left, out_chs_left)
        r = out_chs_right
        self.comb_iter_0_left = BranchSeparables(r, r, 5, 1, 2)
        self.comb_iter_0_right = BranchSeparables(r, r, 3, 1, 1)
        self.comb_iter_1_left = BranchSeparables(r, r, 5, 1, 2)
        self.comb_iter_1_right = BranchSeparables(r, r, 3, 1, 1)
        self.comb_iter_2_left = nn.AvgPool2d(3, stride=1, padding=1, count_include_pad=False)
        self.comb_iter_3_left = nn.AvgPool2d(3, stride=1, padding=1, count_include_pad=False)
        self.comb_iter_3_right = nn.AvgPool2d(3, stride=1, padding=1, count_include_pad=False)
        self.comb_iter_4_left = BranchSeparables(r, r, 3, 1, 1)

    def forward(self, x, x_prev):
        x_left = self._fp(x_prev)
        x_right = self.conv_1x1(x)
        c0 = self.comb_iter_0_left(x_right) + self.comb_iter_0_right(x_left)
        c1 = self.comb_iter_1_left(x_left) + self.comb_iter_1_right(x_left)
        c2 = self.comb_iter_2_left(x_right) + x_left
        c3 = self.comb_iter_3_left(x_left) + self.comb_iter_3_right(x_left)
        c4 = self.comb_iter_4_left(x_right) + x_right
        return torch.cat([x_left, c0, c1, c2, c3, c4], 1)


class NormalCell(nn.Module):
    def __init__(self, in_chs_left, out_chs_left, in_chs_right, out_chs_right):
        super().__init__()
        self.conv_prev_1x1 = _relu_conv_bn(in_chs_left, out_chs_left)
        self.conv_1x1 = _relu_conv_bn(in_chs_right, out_chs_right)
        l, r = out_chs_left, out_chs_right
        self.comb_iter_0_left = BranchSeparables(r, r, 5, 1, 2)
        self.comb_iter_0_right = BranchSeparables(l, l, 3, 1, 1)
        self.comb_iter_1_left = BranchSeparables(l, l, 5, 1, 2)
        self.comb_iter_1_right = BranchSeparables(l, l, 3, 1, 1)
        self.comb_iter_2_left = nn.AvgPool2d(3, stride=1, padding=1, count_include_pad=False)
        self.comb_iter_3_left = nn.AvgPool2d(3, stride=1, padding=1, count_include_pad=False)
        self.comb_iter_3_right = nn.AvgPool2d(3, stride=1, padding=1, count_include_pad=False)
        self.comb_iter_4_left = BranchSeparables(r, r, 3, 1, 1)

    def forward(self, x, x_prev):
        x_left = self.conv_prev_1x1(x_prev)
        x_right = self.conv_1x1(x)
        c0 = self.comb_iter_0_left(x_right) + self.comb_iter_0_right(x_left)
        c1 = self.comb_iter_1_left(x_left) + self.comb_iter_1_right(x_left)
        c2 = self.comb_iter_2_left(x_right) + x_left
        c3 = self.comb_iter_3_left(x_left) + self.comb_iter_3_right(x_left)
        c4 = self.comb_iter_4_left(x_right) + x_right
        return torch.cat([x_left, c0, c1, c2, c3, c4], 1)


class _ReductionCellBase(nn.Module):
    def forward(self, x, x_prev):
        x_left = self.conv_prev_1x1(x_prev)
        x_right = self.conv_1x1(x)
        c0 = self.comb_iter_0_left(x_right) + self.comb_iter_0_right(x_left)
        c1 = self.comb_iter_1_left(x_right) + self.comb_iter_1_right(x_left)
        c2 = self.comb_iter_2_left(x_right) + self.comb_iter_2_right(x_left)
        c3 = self.comb_iter_3_right(c0) + c1
        c4 = self.comb_iter_4_left(c0) + self.comb_iter_4_right(x_right)
        return torch.cat([c1, c2, c3, c4], 1)


class ReductionCell0(_ReductionCellBase):
    """First reduction cell: zero-padded strided branches (TF parity)."""

    def __init__(self, in_chs_left, out_chs_left, in_chs_right, out_chs_right):
        super().__init__()
        self.conv_prev_1x1 = _relu_conv_bn(in_chs_left, out_chs_left)
        self.conv_1x1 = _relu_conv_bn(in_chs_right, out_chs_right)
        r = out_chs_right
        self.comb_iter_0_left = BranchSeparablesReduction(r, r, 5, 2, 2)
        self.comb_iter_0_right = BranchSeparablesReduction(r, r, 7, 2, 3)
        self.comb_iter_1_left = MaxPoolPad()
        self.comb_iter_1_right = BranchSeparablesReduction(r, r, 7, 2, 3)
        self.comb_iter_2_left = AvgPoolPad()
        self.comb_iter_2_right = BranchSeparablesReduction(r, r, 5, 2, 2)
        self.comb_iter_3_right = nn.AvgPool2d(3, stride=1, padding=1, count_include_pad=False)
        self.comb_iter_4_left = BranchSeparablesReduction(r, r, 3, 1, 1)
        self.comb_iter_4_right = MaxPoolPad()


class ReductionCell1(_ReductionCellBase):
    def __init__(self, in_chs_left, out_chs_left, in_chs_right, out_chs_right):
        super().__init__()
        self.conv_prev_1x1 = _relu_conv_bn(in_chs_left, out_chs_left)
        self.conv_1x1 = _relu_conv_bn(in_chs_right, out_chs_right)
        r = out_chs_right
        self.comb_iter_0_left = BranchSeparables(r, r, 5, 2, 2)
        self.comb_iter_0_right = BranchSeparables(r, r, 7, 2, 3)
        self.comb_iter_1_left = nn.MaxPool2d(3, stride=2, padding=1)
        self.comb_iter_1_right = BranchSeparables(r, r, 7, 2, 3)
        self.comb_iter_2_left = nn.AvgPool2d(3, stride=2, padding=1, count_include_pad=False)
        self.comb_iter_2_right = BranchSeparables(r, r, 5, 2, 2)
        self.comb_iter_3_right = nn.AvgPool2d(3, stride=1, padding=1, count_include_pad=False)
        self.comb_iter_4_left = BranchSeparables(r, r, 3, 1, 1)
        self.comb_iter_4_right = nn.MaxPool2d(3, stride=2, padding=1)


class NASNetALarge(nn.Module):
    """NASNet-A-Large (6 @ 4032)."""

    def __init__(self, num_classes=1000, in_chans=3, stem_size=96, num_features=4032,
                 channel_multiplier=2, drop_rate=0.0, global_pool="avg"):
        super().__init__()
        self.num_classes = num_classes
        self.stem_size = stem_size
        self.num_features = num_features
        self.drop_rate = drop_rate
        c = num_features // 24

        self.conv0 = nn.Sequential(OrderedDict([
            ("conv", nn.Conv2d(in_chans, stem_size, 3, padding=0, stride=2, bias=False)),
            ("bn", nn.BatchNorm2d(stem_size, eps=0.001, momentum=0.1)),
        ]))
        self.cell_stem_0 = CellStem0(stem_size, num_channels=c // (channel_multiplier ** 2))
        self.cell_stem_1 = CellStem1(stem_size, num_channels=c // channel_multiplier)

        self.cell_0 = FirstCell(c, c // 2, 2 * c, c)
        for i in range(1, 6):
            setattr(self, f"cell_{i}", NormalCell(
                2 * c if i == 1 else 6 * c, c, 6 * c, c))
        self.reduction_cell_0 = ReductionCell0(6 * c, 2 * c, 6 * c, 2 * c)
        self.cell_6 = FirstCell(6 * c, c, 8 * c, 2 * c)
        for i in range(7, 12):
            setattr(self, f"cell_{i}", NormalCell(
                8 * c if i == 7 else 12 * c, 2 * c, 12 * c, 2 * c))
        self.reduction_cell_1 = ReductionCell1(12 * c, 4 * c, 12 * c, 4 * c)
        self.cell_12 = FirstCell(12 * c, 2 * c, 16 * c, 4 * c)
        for i in range(13, 18):
            setattr(self, f"cell_{i}", NormalCell(
                16 * c if i == 13 else 24 * c, 4 * c, 24 * c, 4 * c))

        self.relu = nn.ReLU()
        self.global_pool = SelectAdaptivePool2d(pool_type=global_pool)
        self.last_linear = nn.Linear(num_features * self.global_pool.feat_mult(),
                                     num_classes)

    def get_classifier(self):
        return self.last_linear

    def reset_classifier(self, num_classes, global_pool="avg"):
        self.num_classes = num_classes
        self.global_pool = SelectAdaptivePool2d(pool_type=global_pool)
        self.last_linear = nn.Linear(
            self.num_features * self.global_pool.feat_mult(), num_classes) if num_classes else None

    def forward_features(self, x):
        x_conv0 = self.conv0(x)
        x_stem_0 = self.cell_stem_0(x_conv0)
        x_stem_1 = self.cell_stem_1(x_conv0, x_stem_0)
        prev, cur = x_stem_0, x_stem_1
        for i in range(6):
            nxt = getattr(self, f"cell_{i}")(cur, prev)
            prev, cur = cur, nxt
        # reduction output feeds the next cell together with the cell TWO
        # steps back (reference nasnet.py forward: cell_6(x_red_0, x_cell_4))
        cur = self.reduction_cell_0(cur, prev)
        for i in range(6, 12):
            nxt = getattr(self, f"cell_{i}")(cur, prev)
            prev, cur = cur, nxt
        cur = self.reduction_cell_1(cur, prev)
        for i in range(12, 18):
            nxt = getattr(self, f"cell_{i}")(cur, prev)
            prev, cur = cur, nxt
        return self.relu(cur)

    def forward(self, x):
        x = self.forward_features(x)
        x = self.global_pool(x).flatten(1)
        if self.drop_rate > 0:
            x = F.dropout(x, p=self.drop_rate, training=self.training)
        return self.last_linear(x)


@register_model
def nasnetalarge(pretrained=False, **kwargs):
    model = NASNetALarge(**kwargs)
    model.default_cfg = default_cfgs["nasnetalarge"]
    return model
