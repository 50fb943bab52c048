"""AutoAugment engine + auxiliary layer tests."""

import numpy as np
import pytest
import torch
from PIL import Image

from deepfake_detection_amd.data.auto_augment import (
    AugmentOp,
    augment_and_mix_transform,
    auto_augment_transform,
    rand_augment_transform,
)
from deepfake_detection_amd.models.layers_extra import (
    AvgPool2dSame,
    CbamModule,
    EcaModule,
    FeatureHooks,
    MedianPool2d,
    SelectiveKernelConv,
    SEModule,
    SplitBatchNorm2d,
    TestTimePoolHead,
    convert_splitbn_model,
)


def _img(seed=0, size=32):
    rng = np.random.RandomState(seed)
    return Image.fromarray(rng.randint(0, 255, (size, size, 3), dtype=np.uint8))


def test_augment_ops_run():
    img = _img()
    for name in ["Rotate", "ShearX", "TranslateXRel", "Solarize", "SolarizeAdd",
                 "PosterizeTpu", "Color", "Contrast", "Brightness", "Sharpness",
                 "AutoContrast", "Equalize", "Invert"]:
        out = AugmentOp(name, prob=1.0, magnitude=7)(img)
        assert out.size == img.size


@pytest.mark.parametrize("cfg,builder", [
    ("original", auto_augment_transform),
    ("v0-mstd0.5", auto_augment_transform),
    ("rand-m9-n2-mstd0.5", rand_augment_transform),
    ("rand-m7-n3-w0", rand_augment_transform),
    ("augmix-m3-w3", augment_and_mix_transform),
])
def test_policy_configs(cfg, builder):
    t = builder(cfg, {"translate_const": 10, "img_mean": (128, 128, 128)})
    out = t(_img())
    assert out.size == (32, 32)


def test_create_transform_with_aa():
    from deepfake_detection_amd.data import create_transform

    t = create_transform((3, 48, 48), is_training=True, auto_augment="rand-m5-n2")
    out = t(_img(size=64))
    assert out.shape == (3, 48, 48)


def test_split_batchnorm():
    bn = SplitBatchNorm2d(8, num_splits=2)
    x = torch.randn(4, 8, 5, 5)
    y = bn(x)
    assert y.shape == x.shape
    m = torch.nn.Sequential(torch.nn.Conv2d(3, 8, 3), torch.nn.BatchNorm2d(8))
    m2 = convert_splitbn_model(m, num_splits=2)
    assert isinstance(m2[1], SplitBatchNorm2d)
    # eval path uses main stats only
    m2.eval()
    assert m2(torch.randn(2, 3, 8, 8)).shape == (2, 8, 6, 6)


def test_split_batchnorm_aux_stats_update_through_blocks():
    """A split-BN-converted model must dispatch through SplitBatchNorm2d.forward
    (not the fused plain-BN path): aux_bn running stats change after a
    training step (subclass bypass via isinstance guard)."""
    from deepfake_detection_amd.models.blocks import InvertedResidual

    torch.manual_seed(0)
    block = InvertedResidual(8, 8, exp_ratio=2.0, se_ratio=0.25, act_layer=torch.nn.SiLU)
    m = convert_splitbn_model(block, num_splits=2)
    assert isinstance(m.bn1, SplitBatchNorm2d)
    before = [a.running_mean.clone() for a in
              [m.bn1.aux_bn[0], m.bn2.aux_bn[0], m.bn3.aux_bn[0]]]
    m.train()
    y = m(torch.randn(4, 8, 9, 9) * 3 + 1)
    y.sum().backward()
    after = [a.running_mean for a in
             [m.bn1.aux_bn[0], m.bn2.aux_bn[0], m.bn3.aux_bn[0]]]
    for b, a in zip(before, after):
        assert not torch.allclose(b, a), "aux BN stats did not update (fused-path bypass)"
    # main stats must also have moved
    assert not torch.allclose(m.bn1.running_mean, torch.zeros(16))


def test_bn_act_subclass_routes_through_module_forward():
    """O.bn_act on a BN subclass must call the subclass forward."""
    from deepfake_detection_amd.ops import functional as O

    class CountingBN(torch.nn.BatchNorm2d):
        calls = 0

        def forward(self, x):
            CountingBN.calls += 1
            return super().forward(x)

    bn = CountingBN(4)
    x = torch.randn(2, 4, 5, 5)
    y = O.bn_act(x, bn, "relu")
    assert CountingBN.calls == 1
    assert torch.allclose(y, torch.relu(torch.nn.BatchNorm2d(4)(x)), atol=1e-5)


@pytest.mark.parametrize("mod", [
    lambda: SEModule(32),
    lambda: EcaModule(32),
    lambda: CbamModule(32),
])
def test_attention_modules(mod):
    m = mod()
    x = torch.randn(2, 32, 7, 7)
    y = m(x)
    assert y.shape == x.shape


def test_selective_kernel():
    m = SelectiveKernelConv(16, 32, stride=2)
    y = m(torch.randn(2, 16, 16, 16))
    assert y.shape == (2, 32, 8, 8)


def test_avg_pool_same():
    m = AvgPool2dSame(3, stride=2)
    assert m(torch.randn(1, 4, 7, 7)).shape == (1, 4, 4, 4)


def test_median_pool():
    m = MedianPool2d(3, same=True)
    x = torch.randn(1, 2, 9, 9)
    assert m(x).shape == x.shape


def test_test_time_pool_head():
    import deepfake_detection_amd as dfd

    model = dfd.create_model("efficientnet_lite0", num_classes=5)
    head = TestTimePoolHead(model, original_pool=7)
    y = head(torch.randn(1, 3, 256, 256))
    assert y.shape == (1, 5)


def test_feature_hooks():
    m = torch.nn.Sequential(torch.nn.Conv2d(3, 4, 3), torch.nn.Conv2d(4, 8, 3))
    hooks = FeatureHooks([{"name": "0"}, {"name": "1"}], m.named_modules())
    x = torch.randn(1, 3, 16, 16)
    m(x)
    outs = hooks.get_output(x.device)
    assert len(outs) == 2 and outs[0].shape[1] == 4 and outs[1].shape[1] == 8


def test_legacy_datasets(tmp_path):
    import tarfile

    from deepfake_detection_amd.data import DatasetTar, DeepFakeDataset_v1

    # DeepFakeDataset_v1 pair file
    fake = tmp_path / "f.jpg"
    real = tmp_path / "r.jpg"
    _img(1).save(fake)
    _img(2).save(real)
    pair_file = tmp_path / "pairs.txt"
    pair_file.write_text(f"{fake}:{real}:0\n")
    ds = DeepFakeDataset_v1(str(tmp_path), str(pair_file))
    assert len(ds) == 2
    img0, y0 = ds[0]
    img1, y1 = ds[1]
    assert (y0, y1) == (0, 1)

    # DatasetTar
    tar_path = tmp_path / "data.tar"
    d = tmp_path / "cls_a"
    d.mkdir()
    _img(3).save(d / "0.jpg")
    with tarfile.open(tar_path, "w") as tf:
        tf.add(d / "0.jpg", arcname="cls_a/0.jpg")
    dt = DatasetTar(str(tar_path))
    img, y = dt[0]
    assert y == 0 and img.size == (32, 32)


def test_aug_splits_separate_transform_and_augmix_dataset():
    """AugMix JSD data path: separate (primary, secondary, final) transform
    triple consumed by AugMixDataset (reference transforms_factory.py:239-318,
    dataset.py:633)."""
    import numpy as np
    from PIL import Image as PILImage

    from deepfake_detection_amd.data.dataset import AugMixDataset
    from deepfake_detection_amd.data.transforms_factory import transforms_imagenet_train

    triple = transforms_imagenet_train(64, auto_augment="augmix-m3",
                                       use_prefetcher=True, separate=True)
    assert isinstance(triple, tuple) and len(triple) == 3

    class _DS:
        transform = triple

        def __getitem__(self, i):
            img = PILImage.fromarray(
                (np.random.rand(80, 80, 3) * 255).astype(np.uint8))
            return self.transform(img), i % 2

        def __len__(self):
            return 4

    ds = AugMixDataset(_DS(), num_splits=3)
    xs, y = ds[0]
    assert len(xs) == 3
    assert all(x.shape == xs[0].shape for x in xs)


def test_every_autoaugment_op_applies():
    """Apply every registered AA op at several magnitudes — catches latent
    crashes in rarely-drawn ops (reference auto_augment.py op table)."""
    import numpy as np
    from PIL import Image as PILImage

    from deepfake_detection_amd.data.auto_augment import NAME_TO_OP, AugmentOp

    img = PILImage.fromarray((np.random.rand(64, 56, 3) * 255).astype(np.uint8))
    failures = []
    for name in NAME_TO_OP:
        for mag in (0, 5, 10):
            try:
                op = AugmentOp(name, prob=1.0, magnitude=mag)
                out = op(img)
                assert out.size[0] > 0
            except Exception as e:  # noqa: BLE001
                failures.append(f"{name}@{mag}: {e}")
    assert not failures, failures
