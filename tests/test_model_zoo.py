"""Param-count parity for the secondary model families (canonical counts
where the architecture is standard)."""

import pytest
import torch

import deepfake_detection_amd as dfd


@pytest.mark.parametrize("name,expect", [
    ("resnet18", 11689512),
    ("resnet34", 21797672),
    ("resnet50", 25557032),
    ("resnet101", 44549160),
    ("resnext50_32x4d", 25028904),
    ("wide_resnet50_2", 68883240),
    ("mobilenetv3_large_100", 5483032),
    ("xception", 22855952),
])
def test_param_parity(name, expect):
    m = dfd.create_model(name)
    assert sum(p.numel() for p in m.parameters()) == expect


@pytest.mark.parametrize("name", [
    "resnet26d", "seresnext26_32x4d", "mobilenetv3_small_100", "mnasnet_a1",
    "mnasnet_b1", "fbnetc_100", "spnasnet_100", "efficientnet_es",
    "efficientnet_lite0", "efficientnet_cc_b0_4e", "tf_efficientnet_b0",
])
def test_forward_small(name):
    m = dfd.create_model(name, num_classes=4)
    m.eval()
    with torch.no_grad():
        y = m(torch.randn(2, 3, 96, 96))
    assert y.shape == (2, 4)


def test_xception_fp16_inference_path():
    """BASELINE config 4 shape: Xception 299px fp16 eval forward."""
    m = dfd.create_model("xception", num_classes=2)
    m.eval().half()
    with torch.no_grad():
        y = m(torch.randn(1, 3, 299, 299).half())
    assert y.shape == (1, 2)


@pytest.mark.parametrize("name,expect", [
    ("densenet121", 7978856),
    ("densenet161", 28681000),
    ("inception_v3", 23834568),
])
def test_param_parity_extra(name, expect):
    m = dfd.create_model(name)
    assert sum(p.numel() for p in m.parameters()) == expect
