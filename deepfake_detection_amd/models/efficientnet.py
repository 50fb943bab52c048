"""EfficientNet family (generic impl + entrypoints), MI355X-native.

Capability parity with reference dfd/timm/models/efficientnet.py:
`EfficientNet` module (:246-351), `EfficientNetFeatures` (:458-518),
`_gen_efficientnet` (:760), `_gen_efficientnet_deepfake` (:806-851, stem 128
-> re-rounded 256, num_features=round_channels(128*cm)=256),
`efficientnet_deepfake_v3/_v4` entrypoints (:1178-1192, B7 multipliers
cm=2.0 dm=3.1), edge/lite/condconv generators, and the default_cfg scheme
(:95-98, deepfake input (12,600,600)).

State-dict key layout is byte-identical to the reference vendored-timm
(conv_stem.weight, bn1.*, blocks.{s}.{b}.*, conv_head.*, bn2.*,
classifier.*) so `.pth.tar` checkpoints interchange; verified shape for
efficientnet_deepfake_v4: 62,373,826 params, blocks/stage [4,7,7,10,10,13,4]
(SURVEY.md §2.3).
"""

import torch.nn as nn
import torch.nn.functional as F

from ..ops import functional as O
from .blocks import resolve_bn_args, round_channels
from .builder import EfficientNetBuilder, decode_arch_def, efficientnet_init_weights
from .helpers import load_pretrained
from .layers import SelectAdaptivePool2d, Swish, create_conv2d
from .registry import register_model

__all__ = ["EfficientNet", "EfficientNetFeatures"]


def _cfg(url="", **kwargs):
    return {
        "url": url,
        "num_classes": 1000,
        "input_size": (3, 224, 224),
        "pool_size": (7, 7),
        "crop_pct": 0.875,
        "interpolation": "bicubic",
        "mean": (0.485, 0.456, 0.406),
        "std": (0.229, 0.224, 0.225),
        "first_conv": "conv_stem",
        "classifier": "classifier",
        **kwargs,
    }


default_cfgs = {
    "efficientnet_b0": _cfg(),
    "efficientnet_b1": _cfg(input_size=(3, 240, 240), pool_size=(8, 8)),
    "efficientnet_b2": _cfg(input_size=(3, 260, 260), pool_size=(9, 9)),
    "efficientnet_b3": _cfg(input_size=(3, 300, 300), pool_size=(10, 10)),
    "efficientnet_b4": _cfg(input_size=(3, 380, 380), pool_size=(12, 12)),
    "efficientnet_b5": _cfg(input_size=(3, 456, 456), pool_size=(15, 15), crop_pct=0.934),
    "efficientnet_b6": _cfg(input_size=(3, 528, 528), pool_size=(17, 17), crop_pct=0.942),
    "efficientnet_b7": _cfg(input_size=(3, 600, 600), pool_size=(19, 19), crop_pct=0.949),
    "efficientnet_b8": _cfg(input_size=(3, 672, 672), pool_size=(21, 21), crop_pct=0.954),
    "efficientnet_l2": _cfg(input_size=(3, 800, 800), pool_size=(25, 25), crop_pct=0.961),
    # deepfake variants: 4 frames x 3 chans folded into 12 input channels
    # (reference efficientnet.py:95-98)
    "efficientnet_deepfake_v3": _cfg(input_size=(12, 600, 600), pool_size=(19, 19), num_classes=2),
    "efficientnet_deepfake_v4": _cfg(input_size=(12, 600, 600), pool_size=(19, 19), num_classes=2),
    "efficientnet_es": _cfg(),
    "efficientnet_em": _cfg(input_size=(3, 240, 240), pool_size=(8, 8)),
    "efficientnet_el": _cfg(input_size=(3, 300, 300), pool_size=(10, 10)),
    "efficientnet_cc_b0_4e": _cfg(),
    "efficientnet_cc_b0_8e": _cfg(),
    "efficientnet_cc_b1_8e": _cfg(input_size=(3, 240, 240), pool_size=(8, 8)),
    "efficientnet_lite0": _cfg(),
    "efficientnet_lite1": _cfg(input_size=(3, 240, 240), pool_size=(8, 8)),
    "efficientnet_lite2": _cfg(input_size=(3, 260, 260), pool_size=(9, 9)),
    "efficientnet_lite3": _cfg(input_size=(3, 300, 300), pool_size=(10, 10)),
    "efficientnet_lite4": _cfg(input_size=(3, 380, 380), pool_size=(12, 12)),
    "tf_efficientnet_b0": _cfg(),
    "tf_efficientnet_b1": _cfg(input_size=(3, 240, 240), pool_size=(8, 8)),
    "tf_efficientnet_b2": _cfg(input_size=(3, 260, 260), pool_size=(9, 9)),
    "tf_efficientnet_b3": _cfg(input_size=(3, 300, 300), pool_size=(10, 10)),
    "tf_efficientnet_b4": _cfg(input_size=(3, 380, 380), pool_size=(12, 12)),
    "tf_efficientnet_b5": _cfg(input_size=(3, 456, 456), pool_size=(15, 15), crop_pct=0.934),
    "tf_efficientnet_b6": _cfg(input_size=(3, 528, 528), pool_size=(17, 17), crop_pct=0.942),
    "tf_efficientnet_b7": _cfg(input_size=(3, 600, 600), pool_size=(19, 19), crop_pct=0.949),
    "tf_efficientnet_b8": _cfg(input_size=(3, 672, 672), pool_size=(21, 21), crop_pct=0.954),
}


class EfficientNet(nn.Module):
    """Generic EfficientNet: conv_stem 3x3 s2 -> bn1 -> act -> blocks ->
    conv_head 1x1 -> bn2 -> act -> global pool -> dropout -> classifier.

    MI355X execution notes: the module graph runs channels_last (NHWC); BN+act
    pairs and SE chains dispatch through ops.functional to fused HIP kernels
    on ROCm devices; the head global-avg-pool uses the fused pool kernel.
    """

    def __init__(self, block_args, num_classes=1000, num_features=1280, in_chans=3,
                 stem_size=32, channel_multiplier=1.0, channel_divisor=8, channel_min=None,
                 output_stride=32, pad_type="", act_layer=nn.ReLU, drop_rate=0.0,
                 drop_path_rate=0.0, se_kwargs=None, norm_layer=nn.BatchNorm2d,
                 norm_kwargs=None, global_pool="avg"):
        super().__init__()
        norm_kwargs = norm_kwargs or {}
        self.num_classes = num_classes
        self.num_features = num_features
        self.drop_rate = drop_rate
        self._in_chs = in_chans

        # Stem — stem_size is re-rounded by the channel multiplier
        # (reference efficientnet.py:274-275: deepfake 128 -> 256 at cm=2.0)
        stem_size = round_channels(stem_size, channel_multiplier, channel_divisor, channel_min)
        self.conv_stem = create_conv2d(self._in_chs, stem_size, 3, stride=2, padding=pad_type)
        self.bn1 = norm_layer(stem_size, **norm_kwargs)
        self.act1 = act_layer(inplace=True)
        self._in_chs = stem_size
        self._act_name = O.act_name_of(self.act1)

        # Middle stages
        builder = EfficientNetBuilder(
            channel_multiplier, channel_divisor, channel_min, output_stride, pad_type,
            act_layer, se_kwargs, norm_layer, norm_kwargs, drop_path_rate)
        self.blocks = nn.Sequential(*builder(self._in_chs, block_args))
        self.feature_info = builder.features
        self._in_chs = builder.in_chs

        # Head
        self.conv_head = create_conv2d(self._in_chs, self.num_features, 1, padding=pad_type)
        from .blocks import _mark_bn_producer as _mark
        _mark(self.conv_head)
        _mark(self.conv_stem)
        self.bn2 = norm_layer(self.num_features, **norm_kwargs)
        self.act2 = act_layer(inplace=True)
        self.global_pool = SelectAdaptivePool2d(pool_type=global_pool)

        self.classifier = nn.Linear(self.num_features * self.global_pool.feat_mult(), self.num_classes)

        efficientnet_init_weights(self)

    def as_sequential(self):
        layers = [self.conv_stem, self.bn1, self.act1]
        layers.extend(self.blocks)
        layers.extend([self.conv_head, self.bn2, self.act2, self.global_pool])
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
        if self._act_name != "other" and O.fusable_bn(self.bn1):
            x = O.bn_act(x, self.bn1, self._act_name)
        else:
            x = self.act1(self.bn1(x))
        x = self.blocks(x)
        x = self.conv_head(x)
        if self._act_name != "other" and O.fusable_bn(self.bn2):
            x = O.bn_act(x, self.bn2, self._act_name)
        else:
            x = self.act2(self.bn2(x))
        return x

    def forward(self, x):
        x = self.forward_features(x)
        if self.global_pool.pool_type == "avg":
            x = O.global_avg_pool(x)
        else:
            x = self.global_pool(x).flatten(1)
        if self.drop_rate > 0.0:
            x = F.dropout(x, p=self.drop_rate, training=self.training)
        return self.classifier(x)

    def get_classifier_params(self):
        return self.classifier.parameters()


class EfficientNetFeatures(nn.Module):
    """Backbone feature extractor returning intermediate feature maps
    (reference efficientnet.py:458-518, feature_hooks.py:5-30 — implemented
    here with direct stage taps instead of forward hooks)."""

    def __init__(self, block_args, out_indices=(0, 1, 2, 3, 4), feature_location="pre_pwl",
                 in_chans=3, stem_size=32, channel_multiplier=1.0, channel_divisor=8,
                 channel_min=None, output_stride=32, pad_type="", act_layer=nn.ReLU,
                 drop_rate=0.0, drop_path_rate=0.0, se_kwargs=None,
                 norm_layer=nn.BatchNorm2d, norm_kwargs=None):
        super().__init__()
        norm_kwargs = norm_kwargs or {}
        self.out_indices = out_indices
        self.drop_rate = drop_rate
        self._in_chs = in_chans

        stem_size = round_channels(stem_size, channel_multiplier, channel_divisor, channel_min)
        self.conv_stem = create_conv2d(self._in_chs, stem_size, 3, stride=2, padding=pad_type)
        self.bn1 = norm_layer(stem_size, **norm_kwargs)
        self.act1 = act_layer(inplace=True)
        self._in_chs = stem_size

        builder = EfficientNetBuilder(
            channel_multiplier, channel_divisor, channel_min, output_stride, pad_type,
            act_layer, se_kwargs, norm_layer, norm_kwargs, drop_path_rate,
            feature_location=feature_location)
        self.blocks = nn.ModuleList(builder(self._in_chs, block_args))
        self.feature_info = builder.features
        self._in_chs = builder.in_chs

        efficientnet_init_weights(self)

    def feature_channels(self, idx=None):
        if idx is None:
            return [self.feature_info[i]["num_chs"] for i in self.out_indices]
        return self.feature_info[idx]["num_chs"]

    def forward(self, x):
        x = self.conv_stem(x)
        x = self.bn1(x)
        x = self.act1(x)
        features = []
        for i, stage in enumerate(self.blocks):
            x = stage(x)
            if i in self.out_indices:
                features.append(x)
        return features


def _create_model(model_kwargs, default_cfg, pretrained=False):
    if model_kwargs.pop("features_only", False):
        load_strict = False
        model_kwargs.pop("num_classes", 0)
        model_kwargs.pop("num_features", 0)
        model_kwargs.pop("head_conv", None)
        model_class = EfficientNetFeatures
    else:
        load_strict = True
        model_class = EfficientNet

    model = model_class(**model_kwargs)
    model.default_cfg = default_cfg
    if pretrained:
        load_pretrained(
            model, default_cfg,
            num_classes=model_kwargs.get("num_classes", 0),
            in_chans=model_kwargs.get("in_chans", 3),
            strict=load_strict)
    return model


def _gen_efficientnet(variant, channel_multiplier=1.0, depth_multiplier=1.0,
                      pretrained=False, **kwargs):
    """Standard EfficientNet scaling over the B0 arch skeleton.

    name: (channel_multiplier, depth_multiplier, resolution)
    b0 (1.0,1.0,224) b1 (1.0,1.1,240) b2 (1.1,1.2,260) b3 (1.2,1.4,300)
    b4 (1.4,1.8,380) b5 (1.6,2.2,456) b6 (1.8,2.6,528) b7 (2.0,3.1,600)
    b8 (2.2,3.6,672) l2 (4.3,5.3,800)
    """
    arch_def = [
        ["ds_r1_k3_s1_e1_c16_se0.25"],
        ["ir_r2_k3_s2_e6_c24_se0.25"],
        ["ir_r2_k5_s2_e6_c40_se0.25"],
        ["ir_r3_k3_s2_e6_c80_se0.25"],
        ["ir_r3_k5_s1_e6_c112_se0.25"],
        ["ir_r4_k5_s2_e6_c192_se0.25"],
        ["ir_r1_k3_s1_e6_c320_se0.25"],
    ]
    model_kwargs = dict(
        block_args=decode_arch_def(arch_def, depth_multiplier),
        num_features=round_channels(1280, channel_multiplier, 8, None),
        stem_size=32,
        channel_multiplier=channel_multiplier,
        act_layer=Swish,
        norm_kwargs=resolve_bn_args(kwargs),
        **kwargs,
    )
    return _create_model(model_kwargs, default_cfgs[variant], pretrained)


def _gen_efficientnet_deepfake(variant, channel_multiplier=1.0, depth_multiplier=1.0,
                               pretrained=False, **kwargs):
    """Deepfake variant: B0 skeleton with slim head — num_features =
    round_channels(128*cm) (=256 at cm 2.0) and stem_size=128 (re-rounded to
    256 inside EfficientNet.__init__) (reference efficientnet.py:806-851)."""
    arch_def = [
        ["ds_r1_k3_s1_e1_c16_se0.25"],
        ["ir_r2_k3_s2_e6_c24_se0.25"],
        ["ir_r2_k5_s2_e6_c40_se0.25"],
        ["ir_r3_k3_s2_e6_c80_se0.25"],
        ["ir_r3_k5_s1_e6_c112_se0.25"],
        ["ir_r4_k5_s2_e6_c192_se0.25"],
        ["ir_r1_k3_s1_e6_c320_se0.25"],
    ]
    model_kwargs = dict(
        block_args=decode_arch_def(arch_def, depth_multiplier),
        num_features=round_channels(128, channel_multiplier, 8, None),
        stem_size=128,
        channel_multiplier=channel_multiplier,
        act_layer=Swish,
        norm_kwargs=resolve_bn_args(kwargs),
        **kwargs,
    )
    return _create_model(model_kwargs, default_cfgs[variant], pretrained)


def _gen_efficientnet_edge(variant, channel_multiplier=1.0, depth_multiplier=1.0,
                           pretrained=False, **kwargs):
    """EfficientNet-EdgeTPU (er blocks, ReLU)."""
    arch_def = [
        ["cn_r1_k3_s1_c24"],
        ["er_r2_k3_s2_e8_c32"],
        ["er_r4_k3_s2_e8_c48"],
        ["ir_r5_k5_s2_e8_c96"],
        ["ir_r4_k5_s1_e8_c144"],
        ["ir_r2_k5_s2_e8_c192"],
    ]
    model_kwargs = dict(
        block_args=decode_arch_def(arch_def, depth_multiplier),
        num_features=round_channels(1280, channel_multiplier, 8, None),
        stem_size=32,
        channel_multiplier=channel_multiplier,
        act_layer=nn.ReLU,
        norm_kwargs=resolve_bn_args(kwargs),
        **kwargs,
    )
    return _create_model(model_kwargs, default_cfgs[variant], pretrained)


def _gen_efficientnet_condconv(variant, channel_multiplier=1.0, depth_multiplier=1.0,
                               experts_multiplier=1, pretrained=False, **kwargs):
    """EfficientNet-CondConv (experts in the last three stages)."""
    arch_def = [
        ["ds_r1_k3_s1_e1_c16_se0.25"],
        ["ir_r2_k3_s2_e6_c24_se0.25"],
        ["ir_r2_k5_s2_e6_c40_se0.25"],
        ["ir_r3_k3_s2_e6_c80_se0.25"],
        ["ir_r3_k5_s1_e6_c112_se0.25_cc4"],
        ["ir_r4_k5_s2_e6_c192_se0.25_cc4"],
        ["ir_r1_k3_s1_e6_c320_se0.25_cc4"],
    ]
    model_kwargs = dict(
        block_args=decode_arch_def(arch_def, depth_multiplier, experts_multiplier=experts_multiplier),
        num_features=round_channels(1280, channel_multiplier, 8, None),
        stem_size=32,
        channel_multiplier=channel_multiplier,
        act_layer=Swish,
        norm_kwargs=resolve_bn_args(kwargs),
        **kwargs,
    )
    return _create_model(model_kwargs, default_cfgs[variant], pretrained)


def _gen_efficientnet_lite(variant, channel_multiplier=1.0, depth_multiplier=1.0,
                           pretrained=False, **kwargs):
    """EfficientNet-Lite: no SE, ReLU6, fixed stem/head widths."""
    arch_def = [
        ["ds_r1_k3_s1_e1_c16"],
        ["ir_r2_k3_s2_e6_c24"],
        ["ir_r2_k5_s2_e6_c40"],
        ["ir_r3_k3_s2_e6_c80"],
        ["ir_r3_k5_s1_e6_c112"],
        ["ir_r4_k5_s2_e6_c192"],
        ["ir_r1_k3_s1_e6_c320"],
    ]
    model_kwargs = dict(
        block_args=decode_arch_def(arch_def, depth_multiplier),
        num_features=1280,
        stem_size=32,
        channel_multiplier=channel_multiplier,
        act_layer=nn.ReLU6,
        norm_kwargs=resolve_bn_args(kwargs),
        **kwargs,
    )
    return _create_model(model_kwargs, default_cfgs[variant], pretrained)


@register_model
def efficientnet_b0(pretrained=False, **kwargs):
    return _gen_efficientnet("efficientnet_b0", 1.0, 1.0, pretrained, **kwargs)


@register_model
def efficientnet_b1(pretrained=False, **kwargs):
    return _gen_efficientnet("efficientnet_b1", 1.0, 1.1, pretrained, **kwargs)


@register_model
def efficientnet_b2(pretrained=False, **kwargs):
    return _gen_efficientnet("efficientnet_b2", 1.1, 1.2, pretrained, **kwargs)


@register_model
def efficientnet_b3(pretrained=False, **kwargs):
    return _gen_efficientnet("efficientnet_b3", 1.2, 1.4, pretrained, **kwargs)


@register_model
def efficientnet_b4(pretrained=False, **kwargs):
    return _gen_efficientnet("efficientnet_b4", 1.4, 1.8, pretrained, **kwargs)


@register_model
def efficientnet_b5(pretrained=False, **kwargs):
    return _gen_efficientnet("efficientnet_b5", 1.6, 2.2, pretrained, **kwargs)


@register_model
def efficientnet_b6(pretrained=False, **kwargs):
    return _gen_efficientnet("efficientnet_b6", 1.8, 2.6, pretrained, **kwargs)


@register_model
def efficientnet_b7(pretrained=False, **kwargs):
    return _gen_efficientnet("efficientnet_b7", 2.0, 3.1, pretrained, **kwargs)


@register_model
def efficientnet_b8(pretrained=False, **kwargs):
    return _gen_efficientnet("efficientnet_b8", 2.2, 3.6, pretrained, **kwargs)


@register_model
def efficientnet_l2(pretrained=False, **kwargs):
    return _gen_efficientnet("efficientnet_l2", 4.3, 5.3, pretrained, **kwargs)


@register_model
def efficientnet_deepfake_v3(pretrained=False, **kwargs):
    """Deepfake detector, B7 multipliers, slim 256-wide head
    (reference efficientnet.py:1178-1185)."""
    return _gen_efficientnet_deepfake("efficientnet_deepfake_v3", 2.0, 3.1, pretrained, **kwargs)


@register_model
def efficientnet_deepfake_v4(pretrained=False, **kwargs):
    """THE production model: 12-channel (4-frame) input, B7 multipliers,
    256-wide stem and head, 2 classes (reference efficientnet.py:1187-1192);
    62,373,826 params at in_chans=12 num_classes=2."""
    return _gen_efficientnet_deepfake("efficientnet_deepfake_v4", 2.0, 3.1, pretrained, **kwargs)


@register_model
def efficientnet_es(pretrained=False, **kwargs):
    return _gen_efficientnet_edge("efficientnet_es", 1.0, 1.0, pretrained, **kwargs)


@register_model
def efficientnet_em(pretrained=False, **kwargs):
    return _gen_efficientnet_edge("efficientnet_em", 1.0, 1.1, pretrained, **kwargs)


@register_model
def efficientnet_el(pretrained=False, **kwargs):
    return _gen_efficientnet_edge("efficientnet_el", 1.2, 1.4, pretrained, **kwargs)


@register_model
def efficientnet_cc_b0_4e(pretrained=False, **kwargs):
    return _gen_efficientnet_condconv("efficientnet_cc_b0_4e", 1.0, 1.0, 1, pretrained, **kwargs)


@register_model
def efficientnet_cc_b0_8e(pretrained=False, **kwargs):
    return _gen_efficientnet_condconv("efficientnet_cc_b0_8e", 1.0, 1.0, 2, pretrained, **kwargs)


@register_model
def efficientnet_cc_b1_8e(pretrained=False, **kwargs):
    return _gen_efficientnet_condconv("efficientnet_cc_b1_8e", 1.0, 1.1, 2, pretrained, **kwargs)


@register_model
def efficientnet_lite0(pretrained=False, **kwargs):
    return _gen_efficientnet_lite("efficientnet_lite0", 1.0, 1.0, pretrained, **kwargs)


@register_model
def efficientnet_lite1(pretrained=False, **kwargs):
    return _gen_efficientnet_lite("efficientnet_lite1", 1.0, 1.1, pretrained, **kwargs)


@register_model
def efficientnet_lite2(pretrained=False, **kwargs):
    return _gen_efficientnet_lite("efficientnet_lite2", 1.1, 1.2, pretrained, **kwargs)


@register_model
def efficientnet_lite3(pretrained=False, **kwargs):
    return _gen_efficientnet_lite("efficientnet_lite3", 1.2, 1.4, pretrained, **kwargs)


@register_model
def efficientnet_lite4(pretrained=False, **kwargs):
    return _gen_efficientnet_lite("efficientnet_lite4", 1.4, 1.8, pretrained, **kwargs)


def _tf(variant, cm, dm):
    def fn(pretrained=False, **kwargs):
        kwargs["bn_eps"] = kwargs.get("bn_eps", 1e-3)
        kwargs["pad_type"] = kwargs.get("pad_type", "same")
        return _gen_efficientnet(variant, cm, dm, pretrained, **kwargs)

    fn.__name__ = variant
    return fn


# tf_ variants: SAME padding + TF bn eps (registered programmatically)
for _name, _cm, _dm in [
    ("tf_efficientnet_b0", 1.0, 1.0), ("tf_efficientnet_b1", 1.0, 1.1),
    ("tf_efficientnet_b2", 1.1, 1.2), ("tf_efficientnet_b3", 1.2, 1.4),
    ("tf_efficientnet_b4", 1.4, 1.8), ("tf_efficientnet_b5", 1.6, 2.2),
    ("tf_efficientnet_b6", 1.8, 2.6), ("tf_efficientnet_b7", 2.0, 3.1),
    ("tf_efficientnet_b8", 2.2, 3.6),
]:
    register_model(_tf(_name, _cm, _dm))


# ---------------------------------------------------------------------------
# AdvProp (_ap) and NoisyStudent (_ns) tf variants — same arch, different
# pretrained weights / input res (reference efficientnet.py:1270-1500)
# ---------------------------------------------------------------------------
for _name, _cm, _dm, _res, _cp in [
    ("tf_efficientnet_b0_ap", 1.0, 1.0, 224, 0.875), ("tf_efficientnet_b1_ap", 1.0, 1.1, 240, 0.882),
    ("tf_efficientnet_b2_ap", 1.1, 1.2, 260, 0.890), ("tf_efficientnet_b3_ap", 1.2, 1.4, 300, 0.904),
    ("tf_efficientnet_b4_ap", 1.4, 1.8, 380, 0.922), ("tf_efficientnet_b5_ap", 1.6, 2.2, 456, 0.934),
    ("tf_efficientnet_b6_ap", 1.8, 2.6, 528, 0.942), ("tf_efficientnet_b7_ap", 2.0, 3.1, 600, 0.949),
    ("tf_efficientnet_b8_ap", 2.2, 3.6, 672, 0.954),
    ("tf_efficientnet_b0_ns", 1.0, 1.0, 224, 0.875), ("tf_efficientnet_b1_ns", 1.0, 1.1, 240, 0.882),
    ("tf_efficientnet_b2_ns", 1.1, 1.2, 260, 0.890), ("tf_efficientnet_b3_ns", 1.2, 1.4, 300, 0.904),
    ("tf_efficientnet_b4_ns", 1.4, 1.8, 380, 0.922), ("tf_efficientnet_b5_ns", 1.6, 2.2, 456, 0.934),
    ("tf_efficientnet_b6_ns", 1.8, 2.6, 528, 0.942), ("tf_efficientnet_b7_ns", 2.0, 3.1, 600, 0.949),
    ("tf_efficientnet_l2_ns_475", 4.3, 5.3, 475, 0.936), ("tf_efficientnet_l2_ns", 4.3, 5.3, 800, 0.961),
]:
    _p = _res // 32
    default_cfgs[_name] = _cfg(input_size=(3, _res, _res), pool_size=(_p, _p), crop_pct=_cp)
    register_model(_tf(_name, _cm, _dm))


def _tf_edge(variant, cm, dm):
    def fn(pretrained=False, **kwargs):
        kwargs["bn_eps"] = kwargs.get("bn_eps", 1e-3)
        kwargs["pad_type"] = kwargs.get("pad_type", "same")
        return _gen_efficientnet_edge(variant, cm, dm, pretrained, **kwargs)

    fn.__name__ = variant
    return fn


def _tf_cc(variant, cm, dm, ei):
    def fn(pretrained=False, **kwargs):
        kwargs["bn_eps"] = kwargs.get("bn_eps", 1e-3)
        kwargs["pad_type"] = kwargs.get("pad_type", "same")
        return _gen_efficientnet_condconv(variant, cm, dm, ei, pretrained, **kwargs)

    fn.__name__ = variant
    return fn


for _name, _cm, _dm in [("tf_efficientnet_es", 1.0, 1.0), ("tf_efficientnet_em", 1.0, 1.1),
                        ("tf_efficientnet_el", 1.2, 1.4)]:
    _res = {"tf_efficientnet_es": 224, "tf_efficientnet_em": 240, "tf_efficientnet_el": 300}[_name]
    default_cfgs[_name] = _cfg(input_size=(3, _res, _res), pool_size=(_res // 32, _res // 32))
    register_model(_tf_edge(_name, _cm, _dm))

for _name, _cm, _dm, _ei in [("tf_efficientnet_cc_b0_4e", 1.0, 1.0, 1),
                             ("tf_efficientnet_cc_b0_8e", 1.0, 1.0, 2),
                             ("tf_efficientnet_cc_b1_8e", 1.0, 1.1, 2)]:
    _res = 240 if _name.endswith("b1_8e") else 224
    default_cfgs[_name] = _cfg(input_size=(3, _res, _res), pool_size=(_res // 32, _res // 32))
    register_model(_tf_cc(_name, _cm, _dm, _ei))


default_cfgs["efficientnet_b2a"] = _cfg(input_size=(3, 288, 288), pool_size=(9, 9), crop_pct=1.0)
default_cfgs["efficientnet_b3a"] = _cfg(input_size=(3, 320, 320), pool_size=(10, 10), crop_pct=1.0)
default_cfgs["efficientnet_b7_deepfake"] = _cfg(input_size=(3, 600, 600), pool_size=(19, 19), crop_pct=0.949)


@register_model
def efficientnet_b2a(pretrained=False, **kwargs):
    """B2 arch at 288x288, crop_pct 1.0 (reference efficientnet.py:1106-1111)."""
    return _gen_efficientnet("efficientnet_b2a", 1.1, 1.2, pretrained, **kwargs)


@register_model
def efficientnet_b3a(pretrained=False, **kwargs):
    """B3 arch at 320x320, crop_pct 1.0 (reference efficientnet.py:1124-1129)."""
    return _gen_efficientnet("efficientnet_b3a", 1.2, 1.4, pretrained, **kwargs)


@register_model
def efficientnet_b7_deepfake(pretrained=False, **kwargs):
    """Stock B7 (full 2560 head) under the deepfake name
    (reference efficientnet.py:1166-1174)."""
    return _gen_efficientnet("efficientnet_b7_deepfake", 2.0, 3.1, pretrained, **kwargs)


# ---------------------------------------------------------------------------
# MixNet (MixedConv kernels; reference efficientnet.py:913-971,1602-1692)
# ---------------------------------------------------------------------------

def _gen_mixnet_s(variant, channel_multiplier=1.0, pretrained=False, **kwargs):
    arch_def = [
        ["ds_r1_k3_s1_e1_c16"],
        ["ir_r1_k3_a1.1_p1.1_s2_e6_c24", "ir_r1_k3_a1.1_p1.1_s1_e3_c24"],
        ["ir_r1_k3.5.7_s2_e6_c40_se0.5_nsw", "ir_r3_k3.5_a1.1_p1.1_s1_e6_c40_se0.5_nsw"],
        ["ir_r1_k3.5.7_p1.1_s2_e6_c80_se0.25_nsw", "ir_r2_k3.5_p1.1_s1_e6_c80_se0.25_nsw"],
        ["ir_r1_k3.5.7_a1.1_p1.1_s1_e6_c120_se0.5_nsw", "ir_r2_k3.5.7.9_a1.1_p1.1_s1_e3_c120_se0.5_nsw"],
        ["ir_r1_k3.5.7.9.11_s2_e6_c200_se0.5_nsw", "ir_r2_k3.5.7.9_p1.1_s1_e6_c200_se0.5_nsw"],
    ]
    model_kwargs = dict(
        block_args=decode_arch_def(arch_def),
        num_features=1536,
        stem_size=16,
        channel_multiplier=channel_multiplier,
        act_layer=nn.ReLU,
        norm_kwargs=resolve_bn_args(kwargs),
        **kwargs,
    )
    return _create_model(model_kwargs, default_cfgs[variant], pretrained)


def _gen_mixnet_m(variant, channel_multiplier=1.0, depth_multiplier=1.0, pretrained=False, **kwargs):
    arch_def = [
        ["ds_r1_k3_s1_e1_c24"],
        ["ir_r1_k3.5.7_a1.1_p1.1_s2_e6_c32", "ir_r1_k3_a1.1_p1.1_s1_e3_c32"],
        ["ir_r1_k3.5.7.9_s2_e6_c40_se0.5_nsw", "ir_r3_k3.5_a1.1_p1.1_s1_e6_c40_se0.5_nsw"],
        ["ir_r1_k3.5.7_s2_e6_c80_se0.25_nsw", "ir_r3_k3.5.7.9_a1.1_p1.1_s1_e6_c80_se0.25_nsw"],
        ["ir_r1_k3_s1_e6_c120_se0.5_nsw", "ir_r3_k3.5.7.9_a1.1_p1.1_s1_e3_c120_se0.5_nsw"],
        ["ir_r1_k3.5.7.9_s2_e6_c200_se0.5_nsw", "ir_r3_k3.5.7.9_p1.1_s1_e6_c200_se0.5_nsw"],
    ]
    model_kwargs = dict(
        block_args=decode_arch_def(arch_def, depth_multiplier, depth_trunc="round"),
        num_features=1536,
        stem_size=24,
        channel_multiplier=channel_multiplier,
        act_layer=nn.ReLU,
        norm_kwargs=resolve_bn_args(kwargs),
        **kwargs,
    )
    return _create_model(model_kwargs, default_cfgs[variant], pretrained)


for _n in ["mixnet_s", "mixnet_m", "mixnet_l", "mixnet_xl", "mixnet_xxl",
           "tf_mixnet_s", "tf_mixnet_m", "tf_mixnet_l"]:
    default_cfgs[_n] = _cfg()


@register_model
def mixnet_s(pretrained=False, **kwargs):
    return _gen_mixnet_s("mixnet_s", 1.0, pretrained, **kwargs)


@register_model
def mixnet_m(pretrained=False, **kwargs):
    return _gen_mixnet_m("mixnet_m", 1.0, 1.0, pretrained, **kwargs)


@register_model
def mixnet_l(pretrained=False, **kwargs):
    return _gen_mixnet_m("mixnet_l", 1.3, 1.0, pretrained, **kwargs)


@register_model
def mixnet_xl(pretrained=False, **kwargs):
    return _gen_mixnet_m("mixnet_xl", 1.6, 1.2, pretrained, **kwargs)


@register_model
def mixnet_xxl(pretrained=False, **kwargs):
    return _gen_mixnet_m("mixnet_xxl", 2.4, 1.3, pretrained, **kwargs)


@register_model
def tf_mixnet_s(pretrained=False, **kwargs):
    kwargs.setdefault("bn_eps", 1e-3)
    kwargs.setdefault("pad_type", "same")
    return _gen_mixnet_s("tf_mixnet_s", 1.0, pretrained, **kwargs)


@register_model
def tf_mixnet_m(pretrained=False, **kwargs):
    kwargs.setdefault("bn_eps", 1e-3)
    kwargs.setdefault("pad_type", "same")
    return _gen_mixnet_m("tf_mixnet_m", 1.0, 1.0, pretrained, **kwargs)


@register_model
def tf_mixnet_l(pretrained=False, **kwargs):
    kwargs.setdefault("bn_eps", 1e-3)
    kwargs.setdefault("pad_type", "same")
    return _gen_mixnet_m("tf_mixnet_l", 1.3, 1.0, pretrained, **kwargs)
