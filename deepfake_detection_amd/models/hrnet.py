"""HRNet (High-Resolution Network) classification models — capability parity
with reference dfd/timm/models/hrnet.py (804 LoC, 9 entrypoints:
hrnet_w18_small, hrnet_w18_small_v2, hrnet_w18/30/32/40/44/48/64).
Wang et al., TPAMI 2020. Parallel multi-resolution branches with repeated
cross-resolution fusion; classification head incrementally downsamples the
four branches into a 2048-wide feature.
"""

import torch.nn as nn
import torch.nn.functional as F

from .layers import SelectAdaptivePool2d
from .registry import register_model

__all__ = ["HighResolutionNet"]

_BN_MOMENTUM = 0.1


def _cfg(**kwargs):
    return {
        "url": "", "num_classes": 1000, "input_size": (3, 224, 224),
        "pool_size": (7, 7), "crop_pct": 0.875, "interpolation": "bilinear",
        "mean": (0.485, 0.456, 0.406), "std": (0.229, 0.224, 0.225),
        "first_conv": "conv1", "classifier": "classifier", **kwargs,
    }


def _stage(num_modules, num_branches, block, num_blocks, num_channels):
    return dict(NUM_MODULES=num_modules, NUM_BRANCHES=num_branches, BLOCK=block,
                NUM_BLOCKS=num_blocks, NUM_CHANNELS=num_channels, FUSE_METHOD="SUM")


def _full_cfg(w):
    return dict(
        STEM_WIDTH=64,
        STAGE1=_stage(1, 1, "BOTTLENECK", (4,), (64,)),
        STAGE2=_stage(1, 2, "BASIC", (4, 4), (w, 2 * w)),
        STAGE3=_stage(4, 3, "BASIC", (4, 4, 4), (w, 2 * w, 4 * w)),
        STAGE4=_stage(3, 4, "BASIC", (4, 4, 4, 4), (w, 2 * w, 4 * w, 8 * w)),
    )


cfg_cls = {
    "hrnet_w18_small": dict(
        STEM_WIDTH=64,
        STAGE1=_stage(1, 1, "BOTTLENECK", (1,), (32,)),
        STAGE2=_stage(1, 2, "BASIC", (2, 2), (16, 32)),
        STAGE3=_stage(1, 3, "BASIC", (2, 2, 2), (16, 32, 64)),
        STAGE4=_stage(1, 4, "BASIC", (2, 2, 2, 2), (16, 32, 64, 128)),
    ),
    "hrnet_w18_small_v2": dict(
        STEM_WIDTH=64,
        STAGE1=_stage(1, 1, "BOTTLENECK", (2,), (64,)),
        STAGE2=_stage(1, 2, "BASIC", (2, 2), (18, 36)),
        STAGE3=_stage(3, 3, "BASIC", (2, 2, 2), (18, 36, 72)),
        STAGE4=_stage(2, 4, "BASIC", (2, 2, 2, 2), (18, 36, 72, 144)),
    ),
}
for _w in (18, 30, 32, 40, 44, 48, 64):
    cfg_cls[f"hrnet_w{_w}"] = _full_cfg(_w)

default_cfgs = {n: _cfg() for n in cfg_cls}


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, inplanes, planes, stride=1, downsample=None):
        super().__init__()
        self.conv1 = nn.Conv2d(inplanes, planes, 3, stride, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(planes, momentum=_BN_MOMENTUM)
        self.relu = nn.ReLU(inplace=True)
        self.conv2 = nn.Conv2d(planes, planes, 3, 1, 1, bias=False)
        self.bn2 = nn.BatchNorm2d(planes, momentum=_BN_MOMENTUM)
        self.downsample = downsample
        self.stride = stride

    def forward(self, x):
        residual = x
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.bn2(self.conv2(out))
        if self.downsample is not None:
            residual = self.downsample(x)
        return self.relu(out + residual)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, inplanes, planes, stride=1, downsample=None):
        super().__init__()
        self.conv1 = nn.Conv2d(inplanes, planes, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(planes, momentum=_BN_MOMENTUM)
        self.conv2 = nn.Conv2d(planes, planes, 3, stride, 1, bias=False)
        self.bn2 = nn.BatchNorm2d(planes, momentum=_BN_MOMENTUM)
        self.conv3 = nn.Conv2d(planes, planes * self.expansion, 1, bias=False)
        self.bn3 = nn.BatchNorm2d(planes * self.expansion, momentum=_BN_MOMENTUM)
        self.relu = nn.ReLU(inplace=True)
        self.downsample = downsample
        self.stride = stride

    def forward(self, x):
        residual = x
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        if self.downsample is not None:
            residual = self.downsample(x)
        return self.relu(out + residual)


_BLOCKS = {"BASIC": BasicBlock, "BOTTLENECK": Bottleneck}


class HighResolutionModule(nn.Module):
    """num_branches parallel block stacks + full cross-resolution fusion."""

    def __init__(self, num_branches, block, num_blocks, num_inchannels,
                 num_channels, multi_scale_output=True):
        super().__init__()
        self.num_inchannels = num_inchannels
        self.num_branches = num_branches
        self.multi_scale_output = multi_scale_output

        self.branches = nn.ModuleList([
            self._make_branch(i, block, num_blocks, num_channels)
            for i in range(num_branches)])
        self.fuse_layers = self._make_fuse_layers()
        self.relu = nn.ReLU(inplace=True)

    def _make_branch(self, idx, block, num_blocks, num_channels):
        downsample = None
        in_ch = self.num_inchannels[idx]
        out_ch = num_channels[idx] * block.expansion
        if in_ch != out_ch:
            downsample = nn.Sequential(
                nn.Conv2d(in_ch, out_ch, 1, bias=False),
                nn.BatchNorm2d(out_ch, momentum=_BN_MOMENTUM))
        layers = [block(in_ch, num_channels[idx], 1, downsample)]
        self.num_inchannels[idx] = out_ch
        for _ in range(1, num_blocks[idx]):
            layers.append(block(out_ch, num_channels[idx]))
        return nn.Sequential(*layers)

    def _make_fuse_layers(self):
        if self.num_branches == 1:
            return nn.ModuleList()
        nb = self.num_branches
        chs = self.num_inchannels
        fuse_layers = []
        for i in range(nb if self.multi_scale_output else 1):
            layer = []
            for j in range(nb):
                if j > i:
                    layer.append(nn.Sequential(
                        nn.Conv2d(chs[j], chs[i], 1, bias=False),
                        nn.BatchNorm2d(chs[i], momentum=_BN_MOMENTUM),
                        nn.Upsample(scale_factor=2 ** (j - i), mode="nearest")))
                elif j == i:
                    layer.append(nn.Identity())
                else:
                    convs = []
                    for k in range(i - j):
                        out_ch = chs[i] if k == i - j - 1 else chs[j]
                        convs.append(nn.Sequential(
                            nn.Conv2d(chs[j], out_ch, 3, 2, 1, bias=False),
                            nn.BatchNorm2d(out_ch, momentum=_BN_MOMENTUM),
                            nn.Identity() if k == i - j - 1 else nn.ReLU(inplace=True)))
                    layer.append(nn.Sequential(*convs))
            fuse_layers.append(nn.ModuleList(layer))
        return nn.ModuleList(fuse_layers)

    def forward(self, x):
        if self.num_branches == 1:
            return [self.branches[0](x[0])]
        x = [branch(xi) for branch, xi in zip(self.branches, x)]
        out = []
        for i, fuse in enumerate(self.fuse_layers):
            y = x[0] if i == 0 else fuse[0](x[0])
            for j in range(1, self.num_branches):
                y = y + (x[j] if i == j else fuse[j](x[j]))
            out.append(self.relu(y))
        return out


class HighResolutionNet(nn.Module):
    def __init__(self, cfg, num_classes=1000, in_chans=3, drop_rate=0.0,
                 global_pool="avg"):
        super().__init__()
        self.num_classes = num_classes
        self.drop_rate = drop_rate
        stem_width = cfg["STEM_WIDTH"]

        self.conv1 = nn.Conv2d(in_chans, stem_width, 3, 2, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(stem_width, momentum=_BN_MOMENTUM)
        self.conv2 = nn.Conv2d(stem_width, 64, 3, 2, 1, bias=False)
        self.bn2 = nn.BatchNorm2d(64, momentum=_BN_MOMENTUM)
        self.relu = nn.ReLU(inplace=True)

        s1 = cfg["STAGE1"]
        block = _BLOCKS[s1["BLOCK"]]
        self.layer1 = self._make_layer(block, 64, s1["NUM_CHANNELS"][0], s1["NUM_BLOCKS"][0])
        pre_chs = [s1["NUM_CHANNELS"][0] * block.expansion]

        for si, name in ((2, "STAGE2"), (3, "STAGE3"), (4, "STAGE4")):
            sc = cfg[name]
            block = _BLOCKS[sc["BLOCK"]]
            chs = [c * block.expansion for c in sc["NUM_CHANNELS"]]
            setattr(self, f"transition{si - 1}", self._make_transition(pre_chs, chs))
            stage, pre_chs = self._make_stage(sc, chs)
            setattr(self, f"stage{si}", stage)
        self.stage4_cfg = cfg["STAGE4"]

        # classification head: incre (Bottleneck widen) -> downsample chain ->
        # final 1x1 to 2048
        head_block = Bottleneck
        head_chs = (32, 64, 128, 256)
        self.incre_modules = nn.ModuleList([
            self._make_layer(head_block, pre_chs[i], head_chs[i], 1)
            for i in range(len(pre_chs))])
        downs = []
        for i in range(len(pre_chs) - 1):
            in_ch = head_chs[i] * head_block.expansion
            out_ch = head_chs[i + 1] * head_block.expansion
            downs.append(nn.Sequential(
                nn.Conv2d(in_ch, out_ch, 3, 2, 1),
                nn.BatchNorm2d(out_ch, momentum=_BN_MOMENTUM),
                nn.ReLU(inplace=True)))
        self.downsamp_modules = nn.ModuleList(downs)
        self.final_layer = nn.Sequential(
            nn.Conv2d(head_chs[-1] * head_block.expansion, 2048, 1),
            nn.BatchNorm2d(2048, momentum=_BN_MOMENTUM),
            nn.ReLU(inplace=True))
        self.num_features = 2048

        self.global_pool = SelectAdaptivePool2d(pool_type=global_pool)
        self.classifier = nn.Linear(2048 * self.global_pool.feat_mult(), num_classes)

    @staticmethod
    def _make_layer(block, inplanes, planes, blocks, stride=1):
        downsample = None
        if stride != 1 or inplanes != planes * block.expansion:
            downsample = nn.Sequential(
                nn.Conv2d(inplanes, planes * block.expansion, 1, stride, bias=False),
                nn.BatchNorm2d(planes * block.expansion, momentum=_BN_MOMENTUM))
        layers = [block(inplanes, planes, stride, downsample)]
        inplanes = planes * block.expansion
        for _ in range(1, blocks):
            layers.append(block(inplanes, planes))
        return nn.Sequential(*layers)

    @staticmethod
    def _make_transition(prev_chs, cur_chs):
        layers = []
        for i, cur in enumerate(cur_chs):
            if i < len(prev_chs):
                if cur != prev_chs[i]:
                    layers.append(nn.Sequential(
                        nn.Conv2d(prev_chs[i], cur, 3, 1, 1, bias=False),
                        nn.BatchNorm2d(cur, momentum=_BN_MOMENTUM),
                        nn.ReLU(inplace=True)))
                else:
                    layers.append(nn.Identity())
            else:
                convs = []
                in_ch = prev_chs[-1]
                for j in range(i + 1 - len(prev_chs)):
                    out_ch = cur if j == i - len(prev_chs) else in_ch
                    convs.append(nn.Sequential(
                        nn.Conv2d(in_ch, out_ch, 3, 2, 1, bias=False),
                        nn.BatchNorm2d(out_ch, momentum=_BN_MOMENTUM),
                        nn.ReLU(inplace=True)))
                layers.append(nn.Sequential(*convs))
        return nn.ModuleList(layers)

    def _make_stage(self, cfg, in_chs, multi_scale_output=True):
        modules = []
        block = _BLOCKS[cfg["BLOCK"]]
        for m in range(cfg["NUM_MODULES"]):
            mso = multi_scale_output or m < cfg["NUM_MODULES"] - 1
            modules.append(HighResolutionModule(
                cfg["NUM_BRANCHES"], block, cfg["NUM_BLOCKS"], list(in_chs),
                cfg["NUM_CHANNELS"], mso))
            in_chs = modules[-1].num_inchannels
        return nn.Sequential(*modules), in_chs

    def get_classifier(self):
        return self.classifier

    def reset_classifier(self, num_classes, global_pool="avg"):
        self.num_classes = num_classes
        self.global_pool = SelectAdaptivePool2d(pool_type=global_pool)
        self.classifier = nn.Linear(
            2048 * self.global_pool.feat_mult(), num_classes) if num_classes else None

    def _stages(self, x):
        x = self.relu(self.bn1(self.conv1(x)))
        x = self.relu(self.bn2(self.conv2(x)))
        x = self.layer1(x)
        xs = [t(x) if not isinstance(t, nn.Identity) else x for t in self.transition1]
        xs = self.stage2(xs)
        xs = [t(xs[-1]) if not isinstance(t, nn.Identity) else xs[i]
              for i, t in enumerate(self.transition2)]
        xs = self.stage3(xs)
        xs = [t(xs[-1]) if not isinstance(t, nn.Identity) else xs[i]
              for i, t in enumerate(self.transition3)]
        return self.stage4(xs)

    def forward_features(self, x):
        xs = self._stages(x)
        y = self.incre_modules[0](xs[0])
        for i, down in enumerate(self.downsamp_modules):
            y = self.incre_modules[i + 1](xs[i + 1]) + down(y)
        return self.final_layer(y)

    def forward(self, x):
        x = self.forward_features(x)
        x = self.global_pool(x).flatten(1)
        if self.drop_rate > 0.0:
            x = F.dropout(x, p=self.drop_rate, training=self.training)
        return self.classifier(x)


def _make(variant, pretrained=False, **kwargs):
    model = HighResolutionNet(cfg_cls[variant], **kwargs)
    model.default_cfg = default_cfgs[variant]
    return model


def _entry(variant):
    def fn(pretrained=False, **kwargs):
        return _make(variant, pretrained, **kwargs)

    fn.__name__ = variant
    return fn


for _n in cfg_cls:
    register_model(_entry(_n))
