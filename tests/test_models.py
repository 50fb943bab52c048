"""Model-zoo structural tests: param parity with the reference vendored-timm
(SURVEY.md §2.3 verified numbers), state-dict key layout, registry/factory."""

import pytest
import torch

import deepfake_detection_amd as dfd
from deepfake_detection_amd.models import list_models, is_model
from deepfake_detection_amd.models.blocks import make_divisible, round_channels


def test_registry_basic():
    assert is_model("efficientnet_deepfake_v4")
    assert "efficientnet_b0" in list_models()
    assert list_models("efficientnet_b*")[0] == "efficientnet_b0"


def test_round_channels():
    # reference semantics (efficientnet_blocks.py:55-69)
    assert make_divisible(32) == 32
    assert round_channels(16, 2.0) == 32
    assert round_channels(128, 2.0) == 256
    assert round_channels(1280, 1.4) == 1792
    assert round_channels(320, 2.0) == 640


def test_deepfake_v4_shape_parity():
    """Verified reference shape: 62,373,826 params, 1200 keys, stem
    Conv2d(12,256,3x3,s2), head Conv2d(640,256,1x1), blocks [4,7,7,10,10,13,4]
    (SURVEY.md §2.3)."""
    m = dfd.create_deepfake_model_v4("efficientnet_deepfake_v4", num_classes=2, in_chans=12)
    assert sum(p.numel() for p in m.parameters()) == 62373826
    sd = list(m.state_dict().keys())
    assert len(sd) == 1200
    assert sd[0] == "conv_stem.weight"
    assert sd[1].startswith("bn1.")
    assert sd[-2:] == ["classifier.weight", "classifier.bias"]
    assert [len(s) for s in m.blocks] == [4, 7, 7, 10, 10, 13, 4]
    assert m.conv_stem.in_channels == 12 and m.conv_stem.out_channels == 256
    assert m.conv_head.in_channels == 640 and m.conv_head.out_channels == 256


def test_deepfake_v4_block_key_names():
    m = dfd.create_deepfake_model_v4("efficientnet_deepfake_v4", num_classes=2, in_chans=12)
    sd = m.state_dict()
    # IR block key layout (reference efficientnet_blocks.py:260-348)
    for k in ["blocks.1.0.conv_pw.weight", "blocks.1.0.bn1.weight",
              "blocks.1.0.conv_dw.weight", "blocks.1.0.bn2.running_mean",
              "blocks.1.0.se.conv_reduce.weight", "blocks.1.0.se.conv_expand.bias",
              "blocks.1.0.conv_pwl.weight", "blocks.1.0.bn3.bias"]:
        assert k in sd, k
    # DS block key layout (stage 0)
    for k in ["blocks.0.0.conv_dw.weight", "blocks.0.0.bn1.weight",
              "blocks.0.0.se.conv_reduce.weight", "blocks.0.0.conv_pw.weight",
              "blocks.0.0.bn2.weight"]:
        assert k in sd, k


@pytest.mark.parametrize("name,expect", [
    ("efficientnet_b0", 5288548),
    ("efficientnet_b4", 19341616),
])
def test_imagenet_param_parity(name, expect):
    m = dfd.create_model(name)
    assert sum(p.numel() for p in m.parameters()) == expect


def test_forward_shapes():
    m = dfd.create_model("efficientnet_b0", num_classes=10)
    m.eval()
    with torch.no_grad():
        y = m(torch.randn(2, 3, 64, 64))
    assert y.shape == (2, 10)


def test_deepfake_v4_forward_backward():
    m = dfd.create_deepfake_model_v4("efficientnet_deepfake_v4", num_classes=2, in_chans=12)
    x = torch.randn(2, 12, 64, 64)
    y = m(x)
    assert y.shape == (2, 2)
    y.sum().backward()
    assert m.conv_stem.weight.grad is not None


def test_factory_asserts_model_name():
    with pytest.raises(AssertionError):
        dfd.create_deepfake_model_v4("efficientnet_b0")


def test_bn_momentum_plumbs():
    m = dfd.create_deepfake_model_v4(
        "efficientnet_deepfake_v4", num_classes=2, in_chans=12, bn_momentum=0.001)
    assert abs(m.bn1.momentum - 0.001) < 1e-12
    assert abs(m.blocks[3][2].bn2.momentum - 0.001) < 1e-12


def test_checkpoint_roundtrip(tmp_path):
    m = dfd.create_deepfake_model_v4("efficientnet_deepfake_v4", num_classes=2, in_chans=12)
    p = tmp_path / "ckpt.pth.tar"
    # module.-prefixed save must load (reference helpers.py:17-20)
    sd = {"module." + k: v for k, v in m.state_dict().items()}
    torch.save({"state_dict": sd, "epoch": 3, "version": 2, "arch": "efficientnet_deepfake_v4"}, p)
    m2 = dfd.create_deepfake_model_v4("efficientnet_deepfake_v4", num_classes=2, in_chans=12)
    from deepfake_detection_amd.models import resume_checkpoint

    other, epoch = resume_checkpoint(m2, str(p))
    assert epoch == 4  # version>=2 -> epoch+1 (reference helpers.py:62-66)
    for k in m.state_dict():
        assert torch.equal(m.state_dict()[k], m2.state_dict()[k])


def test_nonstrict_load_drops_mismatched(tmp_path):
    m = dfd.create_deepfake_model_v4("efficientnet_deepfake_v4", num_classes=2, in_chans=12)
    p = tmp_path / "c.pth.tar"
    torch.save({"state_dict": m.state_dict()}, p)
    # different head size: non-strict load drops the mismatched classifier
    m3 = dfd.create_deepfake_model_v4(
        "efficientnet_deepfake_v4", num_classes=5, in_chans=12, checkpoint_path=str(p))
    assert m3.classifier.out_features == 5


def test_efficientnet_features_backbone():
    """EfficientNetFeatures multi-stage extractor (reference
    efficientnet.py:458-518): features_only returns per-stage pyramids."""
    import deepfake_detection_amd as dfd

    m = dfd.create_model("efficientnet_b0", features_only=True)
    m.eval()
    with torch.no_grad():
        feats = m(torch.randn(1, 3, 128, 128))
    assert isinstance(feats, (list, tuple)) and len(feats) >= 4
    hw = [f.shape[-1] for f in feats]
    assert hw == sorted(hw, reverse=True)  # decreasing spatial pyramid


def test_model_shapes_tool():
    """tools/model_shapes.py inventory matches the known B4 block structure
    (32 depthwise convs; blocks [4,7,7,10,10,13,4] minus... dw per block)."""
    import os
    import sys

    sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", "tools"))
    import model_shapes

    import deepfake_detection_amd as dfd

    m = dfd.create_model("efficientnet_b0", num_classes=2)
    rows = model_shapes.collect(m, img_size=224)
    totals = {}
    for (kind, *_), n in rows.items():
        totals[kind] = totals.get(kind, 0) + n
    assert totals["dw"] == 16  # B0: one dw per block, 16 blocks
    # BN modules never run their own forward on this stack (the fused
    # HIP/CPU path consumes their parameters directly), so no "bn" rows
    assert "bn" not in totals


def test_stem_weight_pack_layout():
    """stemconv weight repack: [N, Kpad] with k = (kh*KW + kw)*Cin + c,
    zero-padded to a 32 multiple."""
    import torch

    from deepfake_detection_amd.ops.stemconv import _pack_weight

    w = torch.arange(2 * 3 * 3 * 3, dtype=torch.float32).reshape(2, 3, 3, 3)
    p = _pack_weight(w)
    assert p.shape == (2, 32) and p.dtype == torch.bfloat16
    # element (n=1, kh=2, kw=1, c=0) -> k = (2*3+1)*3 + 0 = 21
    assert p[1, 21].item() == w[1, 0, 2, 1].item()
    assert (p[:, 27:] == 0).all()


def test_conv_dispatch_classes():
    """create_conv2d routes 1x1 -> PointwiseConv2d, k3 s2 small-Cin ->
    StemConv2d, depthwise -> DepthwiseConv2d, else nn.Conv2d; all share the
    nn.Conv2d state_dict."""
    import torch.nn as nn

    from deepfake_detection_amd.models.layers import (DepthwiseConv2d,
                                                      PointwiseConv2d,
                                                      StemConv2d, create_conv2d)

    assert type(create_conv2d(64, 128, 1)) is PointwiseConv2d
    assert type(create_conv2d(3, 48, 3, stride=2)) is StemConv2d
    assert type(create_conv2d(64, 64, 3, depthwise=True)) is DepthwiseConv2d
    assert type(create_conv2d(64, 128, 3, stride=2)) is nn.Conv2d  # Cin > 16
    assert type(create_conv2d(64, 128, 3)) is nn.Conv2d
    for cls_conv in (create_conv2d(64, 128, 1), create_conv2d(3, 48, 3, stride=2)):
        sd = cls_conv.state_dict()
        assert set(sd) == {"weight"}


def test_pw_dispatch_gating():
    import torch

    from deepfake_detection_amd.ops.pwconv import pw_supported, pw_use_mfma

    assert pw_use_mfma(64, 128)
    assert not pw_use_mfma(63, 128)  # not 8-aligned
    w = torch.zeros(128, 64, 1, 1)
    x = torch.zeros(2, 64, 8, 8, dtype=torch.bfloat16)
    assert pw_supported(x, w, 1, 0, 1, 1)
    assert not pw_supported(x.float(), w, 1, 0, 1, 1)  # fp32 -> torch path
    assert not pw_supported(x, w, 2, 0, 1, 1)          # strided
    assert not pw_supported(x, torch.zeros(128, 64, 3, 3), 1, 1, 1, 1)
