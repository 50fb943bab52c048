"""Data pipeline tests (SURVEY.md §4 test strategy): dataset split
determinism + label-balance grouping, transform shared randomness, collate
shapes, prefetcher normalize math, ordered sampler slicing."""

import os
import random

import numpy as np
import pytest
import torch
from PIL import Image

from deepfake_detection_amd.data import (
    DeepFakeDataset_v3,
    OrderedDistributedSampler,
    PrefetchLoader_v3,
    SyntheticDeepFakeDataset,
    create_deepfake_loader_v3,
    fast_collate,
    mixup_target,
    resolve_data_config,
    transforms_deepfake_eval_v3,
    transforms_deepfake_train_v3,
)
from deepfake_detection_amd.data.transforms import (
    MultiColorJitter,
    MultiRandomCrop,
    MultiRandomHorizontalFlip,
    MultiRandomResize,
)


def _make_dataset_tree(root, n_real=6, n_fake=10, img_num=4, size=32):
    for cls, n in (("real", n_real), ("fake", n_fake)):
        names = []
        for i in range(n):
            name = f"{cls}{i:03d}"
            d = os.path.join(root, cls, name)
            os.makedirs(d, exist_ok=True)
            use_num = img_num if i % 2 == 0 else 2  # some clips have <4 frames
            for f in range(use_num):
                Image.new("RGB", (size, size), color=(i * 10 % 255, f * 40 % 255, 0)).save(
                    os.path.join(d, f"{f}.jpg"))
            names.append((name, use_num))
        with open(os.path.join(root, f"{cls}_list.txt"), "w") as fh:
            for name, use_num in names:
                fh.write(f"{name}:{use_num}\n")


@pytest.fixture
def data_root(tmp_path):
    root = str(tmp_path / "ds")
    os.makedirs(root)
    _make_dataset_tree(root)
    return root


def test_dataset_split_deterministic(data_root):
    kw = dict(class_names="fake,real", train_split=True, train_ratio=0.5,
              random_state=42, label_balance=True)
    tr1 = DeepFakeDataset_v3(data_root, is_training=True, **kw)
    tr2 = DeepFakeDataset_v3(data_root, is_training=True, **kw)
    va = DeepFakeDataset_v3(data_root, is_training=False, **kw)
    assert [tuple(x) for x in tr1.real_images] == [tuple(x) for x in tr2.real_images]
    # val reals = complement of train reals
    train_names = {x[0] for x in tr1.real_images}
    val_names = {x[0] for x in va.real_images}
    assert train_names.isdisjoint(val_names)
    assert len(train_names) + len(val_names) == 6


def test_dataset_label_balance_groups(data_root):
    ds = DeepFakeDataset_v3(data_root, class_names="fake,real", label_balance=True)
    # 10 fakes split into 6 groups (len(real)) -> len = 6 groups + 6 reals
    assert len(ds.fake_images) == 6
    assert len(ds) == 12
    assert sum(len(g) for g in ds.fake_images) == 10


def test_dataset_item_shapes_and_labels(data_root):
    ds = DeepFakeDataset_v3(data_root, class_names="fake,real", label_balance=True)
    ds.set_transform(transforms_deepfake_eval_v3(24))
    x, y = ds[0]
    assert y == 0  # fake first
    assert x.shape == (12, 24, 24) and x.dtype == np.uint8
    x, y = ds[len(ds.fake_images)]
    assert y == 1  # first real


def test_dataset_epoch_round_robin(data_root):
    ds = DeepFakeDataset_v3(data_root, class_names="fake,real", label_balance=True)
    ds.set_transform(None)
    # group 0 cycles deterministically with set_epoch
    g = ds.fake_images[0]
    seen = []
    for epoch in range(len(g) * 2):
        ds.set_epoch(epoch)
        idx = epoch % len(g)
        assert tuple(g[idx]) == tuple(g[epoch % len(g)])
        seen.append(idx)
    assert seen[:len(g)] == list(range(len(g)))


def test_multi_transforms_share_randomness():
    imgs = [Image.new("RGB", (40, 40), color=(i * 60, 0, 0)) for i in range(4)]
    random.seed(0)
    crop = MultiRandomCrop(24, pad_if_needed=True)
    out = crop(imgs)
    assert all(o.size == (24, 24) for o in out)

    # shared flip: either all flipped or none
    imgs = [Image.fromarray(np.random.RandomState(i).randint(0, 255, (8, 8, 3), dtype=np.uint8))
            for i in range(4)]
    flip = MultiRandomHorizontalFlip(p=0.5)
    for _ in range(10):
        out = flip(imgs)
        flipped = [np.array(o) [0, 0, 0] != np.array(i)[0, 0, 0] for o, i in zip(out, imgs)]
        assert all(flipped) or not any(flipped)

    # shared resize: all frames same output size
    rr = MultiRandomResize(scale=(0.5, 1.5))
    out = rr(imgs)
    assert len({o.size for o in out}) == 1


def test_train_transform_pipeline_shapes():
    t = transforms_deepfake_train_v3(32, color_jitter=0.4, rotate_range=10)
    imgs = [Image.new("RGB", (48, 48)) for _ in range(4)]
    out = t(imgs)
    assert out.shape == (12, 32, 32) and out.dtype == np.uint8


def test_fast_collate():
    batch = [(np.zeros((12, 8, 8), dtype=np.uint8), 0), (np.ones((12, 8, 8), dtype=np.uint8), 1)]
    x, y = fast_collate(batch)
    assert x.shape == (2, 12, 8, 8) and x.dtype == torch.uint8
    assert y.tolist() == [0, 1]


def test_prefetch_normalize_math():
    ds = SyntheticDeepFakeDataset(length=4, img_size=16, img_num=4)
    loader = torch.utils.data.DataLoader(ds, batch_size=2, collate_fn=fast_collate)
    pf = PrefetchLoader_v3(loader, fp16=False, img_num=4)
    x, y = next(iter(pf))
    assert x.shape == (2, 12, 16, 16)
    # check against direct math on the raw batch
    raw, _ = fast_collate([ds[0], ds[1]])
    expect = (raw.float() - pf.mean) / pf.std
    assert torch.allclose(x.contiguous(), expect, atol=1e-5)


def test_loader_factory_end_to_end(data_root):
    ds = DeepFakeDataset_v3(data_root, class_names="fake,real", label_balance=True)
    loader = create_deepfake_loader_v3(
        ds, input_size=(12, 24, 24), batch_size=4, is_training=True,
        num_workers=0, fp16=False, rotate_range=5)
    x, y = next(iter(loader))
    assert x.shape == (4, 12, 24, 24)
    assert x.dtype == torch.float32


def test_ordered_distributed_sampler():
    ds = list(range(10))
    s0 = OrderedDistributedSampler(ds, num_replicas=4, rank=0)
    s3 = OrderedDistributedSampler(ds, num_replicas=4, rank=3)
    i0, i3 = list(iter(s0)), list(iter(s3))
    assert len(i0) == len(i3) == 3  # ceil(10/4)
    assert i0 == [0, 4, 8]
    assert i3 == [3, 7, 1]  # padded with wrapped indices


def test_mixup_target():
    t = torch.tensor([0, 1])
    y = mixup_target(t, 2, lam=0.7, smoothing=0.0, device="cpu")
    assert torch.allclose(y, torch.tensor([[0.7, 0.3], [0.3, 0.7]]))


def test_resolve_data_config_v2_string():
    cfg = resolve_data_config({"input_size_v2": "12,600,600", "model": "efficientnet_deepfake_v4"})
    assert cfg["input_size"] == (12, 600, 600)
    assert cfg["mean"] == (0.485, 0.456, 0.406)


def test_tf_preprocessing_eval_transform():
    """TF-style eval path (reference tf_preprocessing.py) without a TF dep."""
    from PIL import Image

    from deepfake_detection_amd.data.transforms_factory import create_transform

    import numpy as np

    t = create_transform(224, is_training=False, tf_preprocessing=True,
                         use_prefetcher=True)
    img = Image.fromarray((torch.rand(300, 260, 3) * 255).byte().numpy())
    out = t(img)
    # reference contract (tf_preprocessing.py:219-226): uint8 CHW in [0,255],
    # normalized later on-device by the PrefetchLoader
    assert isinstance(out, np.ndarray)
    assert out.shape == (3, 224, 224)
    assert out.dtype == np.uint8
    assert out.max() > 1  # not a truncated [0,1] float
    # without the prefetcher the standard eval pipeline serves (normalizes on CPU)
    t2 = create_transform(224, is_training=False, tf_preprocessing=True)
    out2 = t2(img)
    assert out2.shape == (3, 224, 224) and out2.dtype == torch.float32


def test_legacy_dataset_and_loader_variants(tmp_path):
    """Legacy v1_bak/v2 datasets + v1 collate (reference dataset.py:126,284,
    loader.py:48-99) exist and roundtrip."""
    import numpy as np

    import deepfake_detection_amd.data as D

    # v1_bak: pair file fake:real:rotated
    img = tmp_path / "i.jpg"
    from PIL import Image as PILImage

    PILImage.fromarray(np.zeros((8, 8, 3), dtype=np.uint8)).save(img)
    pair_file = tmp_path / "pairs.txt"
    pair_file.write_text(f"{img}:{img}:0\n" * 3)
    ds = D.DeepFakeDataset_v1_bak(str(tmp_path), str(pair_file))
    fake, real, fr, rr = ds[0]
    assert len(ds) == 3 and fr == 0 and rr == 0

    # v2: per-root list files; names carry the _<rot>.jpg suffix
    for sub in ("real", "fake"):
        (tmp_path / sub).mkdir()
        PILImage.fromarray(np.zeros((8, 8, 3), dtype=np.uint8)).save(tmp_path / sub / "a_0.jpg")
    (tmp_path / "real_list.txt").write_text("a_0.jpg\n")
    (tmp_path / "fake_list.txt").write_text("a_0.jpg\n")
    ds2 = D.DeepFakeDataset_v2(str(tmp_path), "fake,real")
    assert len(ds2) == 2
    _, t0 = ds2[0]
    _, t1 = ds2[1]
    assert (t0, t1) == (0, 1)


def test_fast_collate_mixup_math():
    """Collate-time uint8 mixing: x_i*lam + x_rev_i*(1-lam); soft targets sum
    to 1 (reference mixup.py:27-51)."""
    import numpy as np

    from deepfake_detection_amd.data import FastCollateMixup

    c = FastCollateMixup(mixup_alpha=0.8, label_smoothing=0.1, num_classes=4)
    batch = [(np.full((3, 8, 8), v, dtype=np.uint8), v % 4) for v in (10, 200)]
    x, y = c(batch)
    assert x.shape == (2, 3, 8, 8) and x.dtype == torch.uint8
    assert y.shape == (2, 4)
    assert torch.allclose(y.sum(1), torch.ones(2), atol=1e-5)
    # mixed pixel values lie between the two sources
    assert 10 <= x[0, 0, 0, 0].item() <= 200

    c.mixup_enabled = False
    x2, y2 = c(batch)
    assert x2[0, 0, 0, 0].item() == 10 and x2[1, 0, 0, 0].item() == 200


def test_random_erasing_per_frame_slices():
    """img_num-aware GPU RandomErasing erases each 3-channel frame slice
    independently (reference random_erasing.py:96-100)."""
    from deepfake_detection_amd.data import RandomErasing

    torch.manual_seed(0)
    re = RandomErasing(probability=1.0, mode="const", img_num=4, device="cpu")
    x = torch.ones(2, 12, 32, 32)
    out = re(x.clone())
    assert out.shape == x.shape
    assert not torch.equal(out, x)  # something was erased
