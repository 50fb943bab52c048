"""MobileNetV3 + MNASNet/FBNet/SPNASNet family (reference
dfd/timm/models/mobilenetv3.py, 454 LoC, 11 entrypoints) — built on the
same arch-string decoder and EfficientNetBuilder as the EfficientNets.
"""

import torch.nn as nn
import torch.nn.functional as F

from .blocks import resolve_bn_args, round_channels
from .builder import EfficientNetBuilder, decode_arch_def, efficientnet_init_weights
from .layers import HardSwish, SelectAdaptivePool2d, create_conv2d, hard_sigmoid
from .registry import register_model

__all__ = ["MobileNetV3"]


def _cfg(**kwargs):
    return {
        "url": "", "num_classes": 1000, "input_size": (3, 224, 224),
        "pool_size": (7, 7), "crop_pct": 0.875, "interpolation": "bilinear",
        "mean": (0.485, 0.456, 0.406), "std": (0.229, 0.224, 0.225),
        "first_conv": "conv_stem", "classifier": "classifier", **kwargs,
    }


default_cfgs = {
    "mobilenetv3_large_075": _cfg(),
    "mobilenetv3_large_100": _cfg(),
    "mobilenetv3_small_075": _cfg(),
    "mobilenetv3_small_100": _cfg(),
    "mobilenetv3_rw": _cfg(),
    "mnasnet_050": _cfg(),
    "mnasnet_075": _cfg(),
    "mnasnet_100": _cfg(),
    "mnasnet_a1": _cfg(),
    "mnasnet_b1": _cfg(),
    "fbnetc_100": _cfg(),
    "spnasnet_100": _cfg(),
}


class MobileNetV3(nn.Module):
    """MobileNetV3: stem -> blocks -> pool -> conv_head (post-pool) ->
    classifier. Differs from EfficientNet in head ordering."""

    def __init__(self, block_args, num_classes=1000, in_chans=3, stem_size=16,
                 num_features=1280, head_bias=True, channel_multiplier=1.0,
                 pad_type="", act_layer=nn.ReLU, drop_rate=0.0, drop_path_rate=0.0,
                 se_kwargs=None, norm_layer=nn.BatchNorm2d, norm_kwargs=None,
                 global_pool="avg"):
        super().__init__()
        norm_kwargs = norm_kwargs or {}
        self.num_classes = num_classes
        self.num_features = num_features
        self.drop_rate = drop_rate
        self._in_chs = in_chans

        stem_size = round_channels(stem_size, channel_multiplier)
        self.conv_stem = create_conv2d(self._in_chs, stem_size, 3, stride=2, padding=pad_type)
        self.bn1 = norm_layer(stem_size, **norm_kwargs)
        self.act1 = act_layer(inplace=True)
        self._in_chs = stem_size

        builder = EfficientNetBuilder(
            channel_multiplier, 8, None, 32, pad_type, act_layer, se_kwargs,
            norm_layer, norm_kwargs, drop_path_rate)
        self.blocks = nn.Sequential(*builder(self._in_chs, block_args))
        self.feature_info = builder.features
        self._in_chs = builder.in_chs

        self.global_pool = SelectAdaptivePool2d(pool_type=global_pool)
        self.conv_head = create_conv2d(
            self._in_chs, self.num_features, 1, padding=pad_type, bias=head_bias)
        self.act2 = act_layer(inplace=True)
        self.classifier = nn.Linear(
            self.num_features * self.global_pool.feat_mult(), self.num_classes)

        efficientnet_init_weights(self)

    def as_sequential(self):
        layers = [self.conv_stem, self.bn1, self.act1]
        layers.extend(self.blocks)
        layers.extend([self.global_pool, self.conv_head, self.act2])
        layers.extend([nn.Flatten(), nn.Dropout(self.drop_rate), self.classifier])
        return nn.Sequential(*layers)

    def get_classifier(self):
        return self.classifier

    def reset_classifier(self, num_classes, global_pool="avg"):
        self.num_classes = num_classes
        self.global_pool = SelectAdaptivePool2d(pool_type=global_pool)
        self.classifier = nn.Linear(
            self.num_features * self.global_pool.feat_mult(), num_classes) if num_classes else None

    def forward_features(self, x):
        x = self.conv_stem(x)
        x = self.bn1(x)
        x = self.act1(x)
        x = self.blocks(x)
        x = self.global_pool(x)
        x = self.conv_head(x)
        x = self.act2(x)
        return x

    def forward(self, x):
        x = self.forward_features(x)
        x = x.flatten(1)
        if self.drop_rate > 0.0:
            x = F.dropout(x, p=self.drop_rate, training=self.training)
        return self.classifier(x)


def _create_model(model_kwargs, default_cfg, pretrained=False):
    model = MobileNetV3(**model_kwargs)
    model.default_cfg = default_cfg
    return model


def _gen_mobilenet_v3(variant, channel_multiplier=1.0, pretrained=False, **kwargs):
    if "minimal" in variant:
        # "minimal" variants: ReLU everywhere, k3 only, no SE
        act_layer = nn.ReLU
        if "small" in variant:
            num_features = 1024
            arch_def = [
                ["ds_r1_k3_s2_e1_c16"],
                ["ir_r1_k3_s2_e4.5_c24", "ir_r1_k3_s1_e3.67_c24"],
                ["ir_r1_k3_s2_e4_c40", "ir_r2_k3_s1_e6_c40"],
                ["ir_r2_k3_s1_e3_c48"],
                ["ir_r3_k3_s2_e6_c96"],
                ["cn_r1_k1_s1_c576"],
            ]
        else:
            num_features = 1280
            arch_def = [
                ["ds_r1_k3_s1_e1_c16"],
                ["ir_r1_k3_s2_e4_c24", "ir_r1_k3_s1_e3_c24"],
                ["ir_r3_k3_s2_e3_c40"],
                ["ir_r1_k3_s2_e6_c80", "ir_r1_k3_s1_e2.5_c80", "ir_r2_k3_s1_e2.3_c80"],
                ["ir_r2_k3_s1_e6_c112"],
                ["ir_r3_k3_s2_e6_c160"],
                ["cn_r1_k1_s1_c960"],
            ]
        model_kwargs = dict(
            block_args=decode_arch_def(arch_def),
            num_features=num_features,
            stem_size=16,
            channel_multiplier=channel_multiplier,
            act_layer=act_layer,
            norm_kwargs=resolve_bn_args(kwargs),
            head_bias=True,
            **kwargs,
        )
        return _create_model(model_kwargs, default_cfgs[variant], pretrained)
    if "small" in variant:
        num_features = 1024
        act_layer = HardSwish
        arch_def = [
            ["ds_r1_k3_s2_e1_c16_se0.25_nre"],
            ["ir_r1_k3_s2_e4.5_c24_nre", "ir_r1_k3_s1_e3.67_c24_nre"],
            ["ir_r1_k5_s2_e4_c40_se0.25", "ir_r2_k5_s1_e6_c40_se0.25"],
            ["ir_r2_k5_s1_e3_c48_se0.25"],
            ["ir_r3_k5_s2_e6_c96_se0.25"],
            ["cn_r1_k1_s1_c576"],
        ]
    else:
        num_features = 1280
        act_layer = HardSwish
        arch_def = [
            ["ds_r1_k3_s1_e1_c16_nre"],
            ["ir_r1_k3_s2_e4_c24_nre", "ir_r1_k3_s1_e3_c24_nre"],
            ["ir_r3_k5_s2_e3_c40_se0.25_nre"],
            ["ir_r1_k3_s2_e6_c80", "ir_r1_k3_s1_e2.5_c80", "ir_r2_k3_s1_e2.3_c80"],
            ["ir_r2_k3_s1_e6_c112_se0.25"],
            ["ir_r3_k5_s2_e6_c160_se0.25"],
            ["cn_r1_k1_s1_c960"],
        ]
    se_kwargs = dict(gate_fn=hard_sigmoid, act_layer=nn.ReLU, reduce_mid=True, divisor=8)
    model_kwargs = dict(
        block_args=decode_arch_def(arch_def),
        num_features=num_features,
        stem_size=16,
        channel_multiplier=channel_multiplier,
        act_layer=act_layer,
        se_kwargs=se_kwargs,
        norm_kwargs=resolve_bn_args(kwargs),
        head_bias=True,
        **kwargs,
    )
    return _create_model(model_kwargs, default_cfgs[variant], pretrained)


def _gen_mnasnet_a1(variant, channel_multiplier=1.0, pretrained=False, **kwargs):
    arch_def = [
        ["ds_r1_k3_s1_e1_c16_noskip"],
        ["ir_r2_k3_s2_e6_c24"],
        ["ir_r3_k5_s2_e3_c40_se0.25"],
        ["ir_r4_k3_s2_e6_c80"],
        ["ir_r2_k3_s1_e6_c112_se0.25"],
        ["ir_r3_k5_s2_e6_c160_se0.25"],
        ["ir_r1_k3_s1_e6_c320"],
    ]
    model_kwargs = dict(
        block_args=decode_arch_def(arch_def),
        stem_size=32,
        channel_multiplier=channel_multiplier,
        act_layer=nn.ReLU,
        norm_kwargs=resolve_bn_args(kwargs),
        **kwargs,
    )
    return _create_model(model_kwargs, default_cfgs[variant], pretrained)


def _gen_mnasnet_b1(variant, channel_multiplier=1.0, pretrained=False, **kwargs):
    arch_def = [
        ["ds_r1_k3_s1_c16_noskip"],
        ["ir_r3_k3_s2_e3_c24"],
        ["ir_r3_k5_s2_e3_c40"],
        ["ir_r3_k5_s2_e6_c80"],
        ["ir_r2_k3_s1_e6_c96"],
        ["ir_r4_k5_s2_e6_c192"],
        ["ir_r1_k3_s1_e6_c320_noskip"],
    ]
    model_kwargs = dict(
        block_args=decode_arch_def(arch_def),
        stem_size=32,
        channel_multiplier=channel_multiplier,
        act_layer=nn.ReLU,
        norm_kwargs=resolve_bn_args(kwargs),
        **kwargs,
    )
    return _create_model(model_kwargs, default_cfgs[variant], pretrained)


def _gen_fbnetc(variant, channel_multiplier=1.0, pretrained=False, **kwargs):
    arch_def = [
        ["ir_r1_k3_s1_e1_c16"],
        ["ir_r1_k3_s2_e6_c24", "ir_r2_k3_s1_e1_c24"],
        ["ir_r1_k5_s2_e6_c32", "ir_r1_k5_s1_e3_c32", "ir_r1_k5_s1_e6_c32", "ir_r1_k3_s1_e6_c32"],
        ["ir_r1_k5_s2_e6_c64", "ir_r1_k5_s1_e3_c64", "ir_r2_k5_s1_e6_c64"],
        ["ir_r3_k5_s1_e6_c112", "ir_r1_k5_s1_e3_c112"],
        ["ir_r4_k5_s2_e6_c184"],
        ["ir_r1_k3_s1_e6_c352"],
    ]
    model_kwargs = dict(
        block_args=decode_arch_def(arch_def),
        stem_size=16,
        num_features=1984,
        channel_multiplier=channel_multiplier,
        act_layer=nn.ReLU,
        norm_kwargs=resolve_bn_args(kwargs),
        **kwargs,
    )
    return _create_model(model_kwargs, default_cfgs[variant], pretrained)


def _gen_spnasnet(variant, channel_multiplier=1.0, pretrained=False, **kwargs):
    arch_def = [
        ["ds_r1_k3_s1_c16_noskip"],
        ["ir_r3_k3_s2_e3_c24"],
        ["ir_r1_k5_s2_e6_c40", "ir_r3_k3_s1_e3_c40"],
        ["ir_r1_k5_s2_e6_c80", "ir_r3_k3_s1_e3_c80"],
        ["ir_r1_k5_s1_e6_c96", "ir_r3_k5_s1_e3_c96"],
        ["ir_r4_k5_s2_e6_c192"],
        ["ir_r1_k3_s1_e6_c320_noskip"],
    ]
    model_kwargs = dict(
        block_args=decode_arch_def(arch_def),
        stem_size=32,
        channel_multiplier=channel_multiplier,
        act_layer=nn.ReLU,
        norm_kwargs=resolve_bn_args(kwargs),
        **kwargs,
    )
    return _create_model(model_kwargs, default_cfgs[variant], pretrained)


@register_model
def mobilenetv3_large_075(pretrained=False, **kwargs):
    return _gen_mobilenet_v3("mobilenetv3_large_075", 0.75, pretrained, **kwargs)


@register_model
def mobilenetv3_large_100(pretrained=False, **kwargs):
    return _gen_mobilenet_v3("mobilenetv3_large_100", 1.0, pretrained, **kwargs)


@register_model
def mobilenetv3_small_075(pretrained=False, **kwargs):
    return _gen_mobilenet_v3("mobilenetv3_small_075", 0.75, pretrained, **kwargs)


@register_model
def mobilenetv3_small_100(pretrained=False, **kwargs):
    return _gen_mobilenet_v3("mobilenetv3_small_100", 1.0, pretrained, **kwargs)


@register_model
def mobilenetv3_rw(pretrained=False, **kwargs):
    return _gen_mobilenet_v3("mobilenetv3_rw", 1.0, pretrained, **kwargs)


@register_model
def mnasnet_050(pretrained=False, **kwargs):
    return _gen_mnasnet_b1("mnasnet_050", 0.5, pretrained, **kwargs)


@register_model
def mnasnet_075(pretrained=False, **kwargs):
    return _gen_mnasnet_b1("mnasnet_075", 0.75, pretrained, **kwargs)


@register_model
def mnasnet_100(pretrained=False, **kwargs):
    return _gen_mnasnet_b1("mnasnet_100", 1.0, pretrained, **kwargs)


@register_model
def mnasnet_b1(pretrained=False, **kwargs):
    return mnasnet_100(pretrained, **kwargs)


@register_model
def mnasnet_a1(pretrained=False, **kwargs):
    return _gen_mnasnet_a1("mnasnet_a1", 1.0, pretrained, **kwargs)


@register_model
def fbnetc_100(pretrained=False, **kwargs):
    return _gen_fbnetc("fbnetc_100", 1.0, pretrained, **kwargs)


@register_model
def spnasnet_100(pretrained=False, **kwargs):
    return _gen_spnasnet("spnasnet_100", 1.0, pretrained, **kwargs)


# ---------------------------------------------------------------------------
# tf_ MobileNetV3 variants (SAME padding, TF bn eps) and the remaining
# MNASNet / MobileNetV2 entrypoints hosted by the reference in
# efficientnet.py (reference efficientnet.py:487-700, mobilenetv3.py:352-430)
# ---------------------------------------------------------------------------

for _n in ["tf_mobilenetv3_large_075", "tf_mobilenetv3_large_100",
           "tf_mobilenetv3_large_minimal_100", "tf_mobilenetv3_small_075",
           "tf_mobilenetv3_small_100", "tf_mobilenetv3_small_minimal_100",
           "mnasnet_140", "semnasnet_050", "semnasnet_075", "semnasnet_100",
           "semnasnet_140", "mnasnet_small", "mobilenetv2_100"]:
    default_cfgs.setdefault(_n, _cfg())


def _tf_mv3(variant, cm):
    def fn(pretrained=False, **kwargs):
        kwargs.setdefault("bn_eps", 1e-3)
        kwargs.setdefault("pad_type", "same")
        return _gen_mobilenet_v3(variant, cm, pretrained, **kwargs)

    fn.__name__ = variant
    return fn


for _name, _cm in [("tf_mobilenetv3_large_075", 0.75), ("tf_mobilenetv3_large_100", 1.0),
                   ("tf_mobilenetv3_large_minimal_100", 1.0), ("tf_mobilenetv3_small_075", 0.75),
                   ("tf_mobilenetv3_small_100", 1.0), ("tf_mobilenetv3_small_minimal_100", 1.0)]:
    register_model(_tf_mv3(_name, _cm))


@register_model
def mnasnet_140(pretrained=False, **kwargs):
    return _gen_mnasnet_b1("mnasnet_140", 1.4, pretrained, **kwargs)


@register_model
def semnasnet_050(pretrained=False, **kwargs):
    return _gen_mnasnet_a1("semnasnet_050", 0.5, pretrained, **kwargs)


@register_model
def semnasnet_075(pretrained=False, **kwargs):
    return _gen_mnasnet_a1("semnasnet_075", 0.75, pretrained, **kwargs)


@register_model
def semnasnet_100(pretrained=False, **kwargs):
    return _gen_mnasnet_a1("semnasnet_100", 1.0, pretrained, **kwargs)


@register_model
def semnasnet_140(pretrained=False, **kwargs):
    return _gen_mnasnet_a1("semnasnet_140", 1.4, pretrained, **kwargs)


def _gen_mnasnet_small(variant, channel_multiplier=1.0, pretrained=False, **kwargs):
    arch_def = [
        ["ds_r1_k3_s1_c8"],
        ["ir_r1_k3_s2_e3_c16"],
        ["ir_r2_k3_s2_e6_c16"],
        ["ir_r4_k5_s2_e6_c32_se0.25"],
        ["ir_r3_k3_s1_e6_c32_se0.25"],
        ["ir_r3_k5_s2_e6_c88_se0.25"],
        ["ir_r1_k3_s1_e6_c144"],
    ]
    model_kwargs = dict(
        block_args=decode_arch_def(arch_def),
        stem_size=8,
        channel_multiplier=channel_multiplier,
        norm_kwargs=resolve_bn_args(kwargs),
        **kwargs,
    )
    return _create_model(model_kwargs, default_cfgs[variant], pretrained)


@register_model
def mnasnet_small(pretrained=False, **kwargs):
    return _gen_mnasnet_small("mnasnet_small", 1.0, pretrained, **kwargs)


def _gen_mobilenet_v2(variant, channel_multiplier=1.0, pretrained=False, **kwargs):
    arch_def = [
        ["ds_r1_k3_s1_c16"],
        ["ir_r2_k3_s2_e6_c24"],
        ["ir_r3_k3_s2_e6_c32"],
        ["ir_r4_k3_s2_e6_c64"],
        ["ir_r3_k3_s1_e6_c96"],
        ["ir_r3_k3_s2_e6_c160"],
        ["ir_r1_k3_s1_e6_c320"],
    ]
    model_kwargs = dict(
        block_args=decode_arch_def(arch_def),
        num_features=1280,
        stem_size=32,
        channel_multiplier=channel_multiplier,
        norm_kwargs=resolve_bn_args(kwargs),
        act_layer=nn.ReLU6,
        **kwargs,
    )
    return _create_model(model_kwargs, default_cfgs[variant], pretrained)


@register_model
def mobilenetv2_100(pretrained=False, **kwargs):
    return _gen_mobilenet_v2("mobilenetv2_100", 1.0, pretrained, **kwargs)
