"""Auxiliary layers: SplitBatchNorm (AdvProp aux-BN), attention modules
(SE/ECA/CBAM/SelectiveKernel), TF-SAME avg pool, median pool, test-time
pooling head, feature hooks.

Capability parity with reference dfd/timm/models/layers/:
split_batchnorm.py:18-75, se.py:4, eca.py:41,75, cbam.py:78,
selective_kernel.py:51, avg_pool2d_same, median_pool, test_time_pool.py:12-33
and feature_hooks.py:5-30.
"""

import math
from collections import OrderedDict, defaultdict
from functools import partial

import torch
import torch.nn as nn
import torch.nn.functional as F

from .layers import get_padding, pad_same

__all__ = [
    "SplitBatchNorm2d",
    "convert_splitbn_model",
    "SEModule",
    "EcaModule",
    "CecaModule",
    "CbamModule",
    "LightCbamModule",
    "SelectiveKernelConv",
    "AvgPool2dSame",
    "MedianPool2d",
    "TestTimePoolHead",
    "apply_test_time_pool",
    "FeatureHooks",
]


class SplitBatchNorm2d(nn.BatchNorm2d):
    """AdvProp-style split BN: the batch is chunked into num_splits groups,
    the first uses the main BN stats, the rest use aux BNs
    (reference split_batchnorm.py:18-50)."""

    def __init__(self, num_features, eps=1e-5, momentum=0.1, affine=True,
                 track_running_stats=True, num_splits=2):
        super().__init__(num_features, eps, momentum, affine, track_running_stats)
        assert num_splits > 1, "Should have at least one aux BN layer (num_splits at least 2)"
        self.num_splits = num_splits
        self.aux_bn = nn.ModuleList([
            nn.BatchNorm2d(num_features, eps, momentum, affine, track_running_stats)
            for _ in range(num_splits - 1)
        ])

    def forward(self, input):
        if self.training:
            split_size = input.shape[0] // self.num_splits
            assert input.shape[0] == split_size * self.num_splits, \
                "batch size must be evenly divisible by num_splits"
            split_input = input.split(split_size)
            x = [super().forward(split_input[0])]
            for i, a in enumerate(self.aux_bn):
                x.append(a(split_input[i + 1]))
            return torch.cat(x, dim=0)
        return super().forward(input)


def convert_splitbn_model(module, num_splits=2):
    """Recursively convert BatchNorm2d modules to SplitBatchNorm2d
    (reference split_batchnorm.py:53-75)."""
    mod = module
    if isinstance(module, torch.nn.modules.instancenorm._InstanceNorm):
        return module
    if isinstance(module, torch.nn.modules.batchnorm._BatchNorm):
        mod = SplitBatchNorm2d(
            module.num_features, module.eps, module.momentum, module.affine,
            module.track_running_stats, num_splits=num_splits)
        mod.running_mean = module.running_mean
        mod.running_var = module.running_var
        mod.num_batches_tracked = module.num_batches_tracked
        if module.affine:
            mod.weight.data = module.weight.data.clone().detach()
            mod.bias.data = module.bias.data.clone().detach()
        for aux in mod.aux_bn:
            aux.running_mean = module.running_mean.clone()
            aux.running_var = module.running_var.clone()
            aux.num_batches_tracked = module.num_batches_tracked.clone()
            if module.affine:
                aux.weight.data = module.weight.data.clone().detach()
                aux.bias.data = module.bias.data.clone().detach()
    for name, child in module.named_children():
        mod.add_module(name, convert_splitbn_model(child, num_splits=num_splits))
    del module
    return mod


class SEModule(nn.Module):
    """Classic channel SE (reference se.py:4)."""

    def __init__(self, channels, reduction=16, act_layer=nn.ReLU):
        super().__init__()
        self.avg_pool = nn.AdaptiveAvgPool2d(1)
        reduction_channels = max(channels // reduction, 8)
        self.fc1 = nn.Conv2d(channels, reduction_channels, kernel_size=1, padding=0, bias=True)
        self.act = act_layer(inplace=True)
        self.fc2 = nn.Conv2d(reduction_channels, channels, kernel_size=1, padding=0, bias=True)

    def forward(self, x):
        x_se = self.avg_pool(x)
        x_se = self.fc1(x_se)
        x_se = self.act(x_se)
        x_se = self.fc2(x_se)
        return x * x_se.sigmoid()


class EcaModule(nn.Module):
    """Efficient Channel Attention (reference eca.py:41): 1-D conv over
    pooled channels, kernel adapted to channel count."""

    def __init__(self, channels=None, kernel_size=3, gamma=2, beta=1):
        super().__init__()
        assert kernel_size % 2 == 1
        if channels is not None:
            t = int(abs(math.log(channels, 2) + beta) / gamma)
            kernel_size = max(t if t % 2 else t + 1, 3)
        self.conv = nn.Conv1d(1, 1, kernel_size=kernel_size,
                              padding=(kernel_size - 1) // 2, bias=False)

    def forward(self, x):
        y = x.mean((2, 3)).view(x.shape[0], 1, -1)
        y = self.conv(y)
        y = y.view(x.shape[0], -1, 1, 1).sigmoid()
        return x * y.expand_as(x)


class CecaModule(nn.Module):
    """ECA with circular padding (reference eca.py:75)."""

    def __init__(self, channels=None, kernel_size=3, gamma=2, beta=1):
        super().__init__()
        assert kernel_size % 2 == 1
        if channels is not None:
            t = int(abs(math.log(channels, 2) + beta) / gamma)
            kernel_size = max(t if t % 2 else t + 1, 3)
        self.padding = (kernel_size - 1) // 2
        self.conv = nn.Conv1d(1, 1, kernel_size=kernel_size, padding=0, bias=False)

    def forward(self, x):
        y = x.mean((2, 3)).view(x.shape[0], 1, -1)
        y = F.pad(y, (self.padding, self.padding), mode="circular")
        y = self.conv(y)
        y = y.view(x.shape[0], -1, 1, 1).sigmoid()
        return x * y.expand_as(x)


class ChannelAttn(nn.Module):
    def __init__(self, channels, reduction=16, act_layer=nn.ReLU):
        super().__init__()
        self.fc1 = nn.Conv2d(channels, channels // reduction, 1, bias=False)
        self.act = act_layer(inplace=True)
        self.fc2 = nn.Conv2d(channels // reduction, channels, 1, bias=False)

    def forward(self, x):
        x_avg = self.fc2(self.act(self.fc1(x.mean((2, 3), keepdim=True))))
        x_max = self.fc2(self.act(self.fc1(F.adaptive_max_pool2d(x, 1))))
        return x * (x_avg + x_max).sigmoid()


class SpatialAttn(nn.Module):
    def __init__(self, kernel_size=7):
        super().__init__()
        self.conv = nn.Conv2d(2, 1, kernel_size, padding=kernel_size // 2, bias=False)

    def forward(self, x):
        x_attn = torch.cat(
            [x.mean(dim=1, keepdim=True), x.amax(dim=1, keepdim=True)], dim=1)
        return x * self.conv(x_attn).sigmoid()


class CbamModule(nn.Module):
    """Convolutional Block Attention (reference cbam.py:78)."""

    def __init__(self, channels, spatial_kernel_size=7):
        super().__init__()
        self.channel = ChannelAttn(channels)
        self.spatial = SpatialAttn(spatial_kernel_size)

    def forward(self, x):
        return self.spatial(self.channel(x))


class LightCbamModule(nn.Module):
    def __init__(self, channels, spatial_kernel_size=7):
        super().__init__()
        self.channel = ChannelAttn(channels)
        self.spatial = SpatialAttn(spatial_kernel_size)

    def forward(self, x):
        return self.spatial(self.channel(x))


class SelectiveKernelConv(nn.Module):
    """Selective-kernel conv (reference selective_kernel.py:51): parallel
    paths with different kernel sizes, softmax attention over paths."""

    def __init__(self, in_channels, out_channels, kernel_size=None, stride=1,
                 dilation=1, groups=1, attn_reduction=16, min_attn_channels=32,
                 keep_3x3=True, split_input=False, act_layer=nn.ReLU,
                 norm_layer=nn.BatchNorm2d):
        super().__init__()
        kernel_size = kernel_size or [3, 5]
        if not isinstance(kernel_size, list):
            kernel_size = [kernel_size] * 2
        if keep_3x3:
            dilation = [dilation * (k - 1) // 2 for k in kernel_size]
            kernel_size = [3] * len(kernel_size)
        else:
            dilation = [dilation] * len(kernel_size)
        self.num_paths = len(kernel_size)
        self.in_channels = in_channels
        self.out_channels = out_channels
        self.split_input = split_input
        if self.split_input:
            assert in_channels % self.num_paths == 0
            in_channels = in_channels // self.num_paths
        groups = min(out_channels, groups)

        self.paths = nn.ModuleList()
        for k, d in zip(kernel_size, dilation):
            p = get_padding(k, stride, d)
            self.paths.append(nn.Sequential(OrderedDict([
                ("conv", nn.Conv2d(in_channels, out_channels, kernel_size=k,
                                   stride=stride, padding=p, dilation=d,
                                   groups=groups, bias=False)),
                ("bn", norm_layer(out_channels)),
                ("act", act_layer(inplace=True)),
            ])))

        attn_channels = max(int(out_channels / attn_reduction), min_attn_channels)
        self.fc_reduce = nn.Conv2d(out_channels, attn_channels, kernel_size=1, bias=False)
        self.bn = nn.BatchNorm2d(attn_channels)
        self.act = act_layer(inplace=True)
        self.fc_select = nn.Conv2d(attn_channels, out_channels * self.num_paths,
                                   kernel_size=1, bias=False)

    def forward(self, x):
        if self.split_input:
            x_split = torch.split(x, self.in_channels // self.num_paths, 1)
            x_paths = [op(x_split[i]) for i, op in enumerate(self.paths)]
        else:
            x_paths = [op(x) for op in self.paths]
        x_stack = torch.stack(x_paths, dim=1)  # (B, P, C, H, W)
        x_sum = x_stack.sum(dim=1)
        attn = x_sum.mean((2, 3), keepdim=True)
        attn = self.fc_reduce(attn)
        attn = self.bn(attn)
        attn = self.act(attn)
        attn = self.fc_select(attn)
        B, C = attn.shape[:2]
        attn = attn.view(B, self.num_paths, C // self.num_paths, 1, 1)
        attn = torch.softmax(attn, dim=1)
        return (x_stack * attn).sum(dim=1)


class AvgPool2dSame(nn.AvgPool2d):
    """AvgPool2d with TF-SAME dynamic padding."""

    def __init__(self, kernel_size, stride=None, padding=0, ceil_mode=False,
                 count_include_pad=True):
        kernel_size = (kernel_size, kernel_size) if isinstance(kernel_size, int) else kernel_size
        stride = (stride, stride) if isinstance(stride, int) else (stride or kernel_size)
        super().__init__(kernel_size, stride, (0, 0), ceil_mode, count_include_pad)

    def forward(self, x):
        x = pad_same(x, self.kernel_size, self.stride)
        return F.avg_pool2d(x, self.kernel_size, self.stride, self.padding,
                            self.ceil_mode, self.count_include_pad)


class MedianPool2d(nn.Module):
    """Median pooling, usable as a robust blur (reference median_pool.py)."""

    def __init__(self, kernel_size=3, stride=1, padding=0, same=False):
        super().__init__()
        self.k = (kernel_size, kernel_size) if isinstance(kernel_size, int) else kernel_size
        self.stride = (stride, stride) if isinstance(stride, int) else stride
        if isinstance(padding, int):
            padding = (padding,) * 4
        self.padding = padding
        self.same = same

    def _padding(self, x):
        if self.same:
            ih, iw = x.size()[2:]
            if ih % self.stride[0] == 0:
                ph = max(self.k[0] - self.stride[0], 0)
            else:
                ph = max(self.k[0] - (ih % self.stride[0]), 0)
            if iw % self.stride[1] == 0:
                pw = max(self.k[1] - self.stride[1], 0)
            else:
                pw = max(self.k[1] - (iw % self.stride[1]), 0)
            return (pw // 2, pw - pw // 2, ph // 2, ph - ph // 2)
        return self.padding

    def forward(self, x):
        x = F.pad(x, self._padding(x), mode="reflect")
        x = x.unfold(2, self.k[0], self.stride[0]).unfold(3, self.k[1], self.stride[1])
        return x.contiguous().view(x.size()[:4] + (-1,)).median(dim=-1)[0]


class TestTimePoolHead(nn.Module):
    """Test-time pooling head: run the classifier as a 1x1 conv over the
    unpooled feature map, then average (reference test_time_pool.py:12-33)."""

    __test__ = False  # "Test" prefix is the reference's name, not a pytest test

    def __init__(self, base, original_pool=7):
        super().__init__()
        self.base = base
        self.original_pool = original_pool
        base_fc = self.base.get_classifier()
        if isinstance(base_fc, nn.Conv2d):
            self.fc = base_fc
        else:
            self.fc = nn.Conv2d(
                self.base.num_features, self.base.num_classes, kernel_size=1, bias=True)
            self.fc.weight.data.copy_(base_fc.weight.data.view(self.fc.weight.size()))
            self.fc.bias.data.copy_(base_fc.bias.data.view(self.fc.bias.size()))
        self.base.reset_classifier(0)

    def forward(self, x):
        x = self.base.forward_features(x)
        x = F.avg_pool2d(x, kernel_size=self.original_pool, stride=1)
        x = self.fc(x)
        x = adaptive_avgmax_pool2d_compat(x, 1)
        return x.view(x.size(0), -1)


def adaptive_avgmax_pool2d_compat(x, output_size=1):
    x_avg = F.adaptive_avg_pool2d(x, output_size)
    x_max = F.adaptive_max_pool2d(x, output_size)
    return 0.5 * (x_avg + x_max)


def apply_test_time_pool(model, config, args=None):
    test_time_pool = False
    input_size = config["input_size"]
    if input_size[-1] > model.default_cfg["input_size"][-1] and \
            input_size[-2] > model.default_cfg["input_size"][-2]:
        default_pool = model.default_cfg.get("pool_size", (7, 7))
        model = TestTimePoolHead(model, original_pool=default_pool[-1])
        test_time_pool = True
    return model, test_time_pool


class FeatureHooks:
    """Forward-hook feature collector (reference feature_hooks.py:5-30)."""

    def __init__(self, hooks, named_modules):
        modules = {k: v for k, v in named_modules}
        for h in hooks:
            hook_name = h["name"]
            m = modules[hook_name]
            hook_fn = partial(self._collect_output_hook, hook_name)
            if h.get("type", "forward") == "forward_pre":
                m.register_forward_pre_hook(hook_fn)
            else:
                m.register_forward_hook(hook_fn)
        self._feature_outputs = defaultdict(OrderedDict)

    def _collect_output_hook(self, name, *args):
        x = args[-1]
        if isinstance(x, tuple):
            x = x[0]
        self._feature_outputs[x.device][name] = x

    def get_output(self, device):
        output = tuple(self._feature_outputs[device].values())
        self._feature_outputs[device] = OrderedDict()
        return output
