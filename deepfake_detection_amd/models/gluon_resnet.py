"""Gluon-pretrained ResNet variants (reference dfd/timm/models/gluon_resnet.py,
373 LoC, 24 entrypoints). Same ResNet trunk as models/resnet.py with the MXNet
Gluon stem conventions:

  v1b — stock 7x7 stem                  v1c — deep 3x3x3 stem, width 32
  v1d — v1c + avg-pool downsample       v1e — deep stem width 64 + avg-down
  v1s — deep stem width 64
"""

from .registry import register_model
from .resnet import BasicBlock, Bottleneck, ResNet, _cfg, default_cfgs

_GLUON = [
    "gluon_resnet18_v1b", "gluon_resnet34_v1b", "gluon_resnet50_v1b",
    "gluon_resnet101_v1b", "gluon_resnet152_v1b", "gluon_resnet50_v1c",
    "gluon_resnet101_v1c", "gluon_resnet152_v1c", "gluon_resnet50_v1d",
    "gluon_resnet101_v1d", "gluon_resnet152_v1d", "gluon_resnet50_v1e",
    "gluon_resnet101_v1e", "gluon_resnet152_v1e", "gluon_resnet50_v1s",
    "gluon_resnet101_v1s", "gluon_resnet152_v1s", "gluon_resnext50_32x4d",
    "gluon_resnext101_32x4d", "gluon_resnext101_64x4d", "gluon_seresnext50_32x4d",
    "gluon_seresnext101_32x4d", "gluon_seresnext101_64x4d", "gluon_senet154",
]
for _n in _GLUON:
    default_cfgs.setdefault(_n, _cfg(first_conv="conv1", classifier="fc"))

_LAYERS = {"18": [2, 2, 2, 2], "34": [3, 4, 6, 3], "50": [3, 4, 6, 3],
           "101": [3, 4, 23, 3], "152": [3, 8, 36, 3]}


def _gluon(variant, depth, stem=None, pretrained=False, **kwargs):
    block = BasicBlock if depth in ("18", "34") else Bottleneck
    if stem == "c":
        kwargs.update(deep_stem=True, stem_width=32)
    elif stem == "d":
        kwargs.update(deep_stem=True, stem_width=32, avg_down=True)
    elif stem == "e":
        kwargs.update(deep_stem=True, stem_width=64, avg_down=True)
    elif stem == "s":
        kwargs.update(deep_stem=True, stem_width=64)
    model = ResNet(block, _LAYERS[depth], **kwargs)
    model.default_cfg = default_cfgs[variant]
    return model


def _mk(variant, depth, stem):
    def fn(pretrained=False, **kwargs):
        return _gluon(variant, depth, stem, pretrained, **kwargs)

    fn.__name__ = variant
    return fn


for _d in ["18", "34", "50", "101", "152"]:
    register_model(_mk(f"gluon_resnet{_d}_v1b", _d, None))
for _d in ["50", "101", "152"]:
    for _s in ["c", "d", "e", "s"]:
        register_model(_mk(f"gluon_resnet{_d}_v1{_s}", _d, _s))


@register_model
def gluon_resnext50_32x4d(pretrained=False, **kwargs):
    return _gluon("gluon_resnext50_32x4d", "50", None, pretrained,
                  cardinality=32, base_width=4, **kwargs)


@register_model
def gluon_resnext101_32x4d(pretrained=False, **kwargs):
    return _gluon("gluon_resnext101_32x4d", "101", None, pretrained,
                  cardinality=32, base_width=4, **kwargs)


@register_model
def gluon_resnext101_64x4d(pretrained=False, **kwargs):
    return _gluon("gluon_resnext101_64x4d", "101", None, pretrained,
                  cardinality=64, base_width=4, **kwargs)


@register_model
def gluon_seresnext50_32x4d(pretrained=False, **kwargs):
    return _gluon("gluon_seresnext50_32x4d", "50", None, pretrained,
                  cardinality=32, base_width=4, use_se=True, **kwargs)


@register_model
def gluon_seresnext101_32x4d(pretrained=False, **kwargs):
    return _gluon("gluon_seresnext101_32x4d", "101", None, pretrained,
                  cardinality=32, base_width=4, use_se=True, **kwargs)


@register_model
def gluon_seresnext101_64x4d(pretrained=False, **kwargs):
    return _gluon("gluon_seresnext101_64x4d", "101", None, pretrained,
                  cardinality=64, base_width=4, use_se=True, **kwargs)


@register_model
def gluon_senet154(pretrained=False, **kwargs):
    """SENet-154 in Gluon form: deep stem, grouped 3x3, SE, avg-down,
    3x3 downsample convs (reference gluon_resnet.py)."""
    return _gluon("gluon_senet154", "152", None, pretrained,
                  cardinality=64, base_width=4, use_se=True, deep_stem=True,
                  stem_width=64, down_kernel_size=3, block_reduce_first=2,
                  **kwargs)
