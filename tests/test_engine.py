"""End-to-end trainer tests on CPU: one tiny epoch through the real
train_epoch/validate loops + the full runner main() on synthetic data."""

import os
import types

import torch

import deepfake_detection_amd as dfd
from deepfake_detection_amd.data import SyntheticDeepFakeDataset
from deepfake_detection_amd.data.loader import PrefetchLoader_v3, fast_collate
from deepfake_detection_amd.engine import train_epoch, validate
from deepfake_detection_amd.loss import LabelSmoothingCrossEntropy


def _loader(n=8, img=32, img_num=4, bs=4):
    ds = SyntheticDeepFakeDataset(length=n, img_size=img, img_num=img_num)
    dl = torch.utils.data.DataLoader(ds, batch_size=bs, collate_fn=fast_collate)
    return PrefetchLoader_v3(dl, fp16=False, img_num=img_num)


def _args(**kw):
    d = dict(amp=False, prefetcher=True, log_interval=2, recovery_interval=0, tta=0)
    d.update(kw)
    return types.SimpleNamespace(**d)


def test_train_epoch_and_validate_cpu():
    torch.manual_seed(0)
    model = dfd.create_model("efficientnet_lite0", num_classes=2, in_chans=12)
    device = torch.device("cpu")
    opt = torch.optim.SGD(model.parameters(), lr=0.01)
    loss_fn = LabelSmoothingCrossEntropy(0.1)
    loader = _loader()
    args = _args()

    before = model.conv_stem.weight.clone()
    metrics = train_epoch(0, model, loader, opt, loss_fn, args, device)
    assert "loss" in metrics and "prec1" in metrics
    assert not torch.equal(before, model.conv_stem.weight)  # params moved

    eval_metrics = validate(model, loader, torch.nn.CrossEntropyLoss(), args, device)
    assert set(eval_metrics) == {"loss", "prec1"}


def test_runner_main_synthetic(tmp_path, monkeypatch):
    """Full runner main(): 1 epoch, tiny model, synthetic data, checkpoint
    written + summary.csv produced."""
    from deepfake_detection_amd.runners.train import _parse_args, main

    monkeypatch.chdir(tmp_path)
    args, args_text = _parse_args([
        "--synthetic-data", "--synthetic-len", "8", "--model", "efficientnet_lite0",
        "--num-classes", "2", "--input-size-v2", "12,32,32", "-b", "4",
        "--epochs", "1", "--sched", "step", "--decay-epochs", "2",
        "--warmup-epochs", "0", "--opt", "rmsproptf", "--opt-eps", "0.001",
        "--workers", "0", "--log-interval", "1", "--no-amp",
        "--eval-metric", "loss", "--model-version", "t0",
    ])
    main(0, args, args_text)
    out_dir = tmp_path / "output" / "t0-efficientnet_lite0"
    assert (out_dir / "summary.csv").exists()
    assert (out_dir / "args.yaml").exists()
    assert (out_dir / "checkpoint-0.pth.tar").exists()
    assert (out_dir / "model_best.pth.tar").exists()
    # resume from the checkpoint through the factory
    ck = torch.load(str(out_dir / "checkpoint-0.pth.tar"), weights_only=False)
    assert ck["version"] == 2 and "optimizer" in ck


def test_inference_runner_cpu(tmp_path):
    """Reference test.sh smoke equivalent: random checkpoint + sample image
    through the fp-inference path (scripts/test.sh semantics)."""
    import numpy as np
    from PIL import Image

    from deepfake_detection_amd.runners.test import test_img

    m = dfd.create_deepfake_model_v4("efficientnet_deepfake_v4", num_classes=2, in_chans=12)
    ckpt = tmp_path / "model_half.pth.tar"
    torch.save({"state_dict": {k: v.half() for k, v in m.state_dict().items()}}, ckpt)

    img_path = tmp_path / "sample.png"
    Image.fromarray(np.random.RandomState(0).randint(0, 255, (64, 48, 3), dtype=np.uint8)).save(img_path)

    results = test_img([str(img_path)], checkpoint_path=str(ckpt), device="cpu")
    assert len(results) == 1
    score = results[0][1]
    assert 0.0 <= score <= 1.0


def test_validate_tta_reduces_output():
    """--tta > 1 oversamples the batch and averages predictions back to B
    rows (the first cut repeated inputs without reducing outputs)."""
    import types

    import deepfake_detection_amd as dfd
    from deepfake_detection_amd.engine import validate

    model = dfd.create_model("resnet18", num_classes=3).eval()
    batches = [(torch.randn(4, 3, 64, 64), torch.randint(0, 3, (4,)))]
    args = types.SimpleNamespace(log_interval=100, prefetcher=False, tta=3, amp=False)
    metrics = validate(model, batches, torch.nn.CrossEntropyLoss(), args,
                       torch.device("cpu"))
    assert set(metrics) == {"loss", "prec1"}
    assert torch.isfinite(torch.tensor(metrics["loss"]))


def test_train_epoch_recovery_interval(tmp_path):
    """Mid-epoch recovery checkpoints every --recovery-interval batches
    (reference train.py:686-689, utils.py:128-149)."""
    import glob
    import types

    import deepfake_detection_amd as dfd
    from deepfake_detection_amd.engine import train_epoch
    from deepfake_detection_amd.utils import CheckpointSaver

    model = dfd.create_model("resnet18", num_classes=2)
    opt = torch.optim.SGD(model.parameters(), lr=0.01)
    saver = CheckpointSaver(checkpoint_dir=str(tmp_path), recovery_dir=str(tmp_path))
    batches = [(torch.randn(2, 3, 32, 32), torch.randint(0, 2, (2,)))
               for _ in range(8)]
    args = types.SimpleNamespace(log_interval=100, recovery_interval=2,
                                 save_images=False, tta=0, prefetcher=False)
    train_epoch(0, model, batches, opt, torch.nn.CrossEntropyLoss(), args,
                torch.device("cpu"), saver=saver)
    recs = glob.glob(str(tmp_path / "recovery-*"))
    # reference cleanup lags one save (utils.py:128-140): current + previous
    assert len(recs) == 2
    assert saver.find_recovery()


def test_train_epoch_save_images(tmp_path):
    """--save-images dumps input batches at log points (reference
    train.py:679-684)."""
    import glob
    import types

    import deepfake_detection_amd as dfd
    from deepfake_detection_amd.engine import train_epoch

    model = dfd.create_model("resnet18", num_classes=2)
    opt = torch.optim.SGD(model.parameters(), lr=0.01)
    batches = [(torch.rand(2, 3, 32, 32), torch.randint(0, 2, (2,)))]
    args = types.SimpleNamespace(log_interval=1, recovery_interval=0,
                                 save_images=True, tta=0, prefetcher=False)
    train_epoch(0, model, batches, opt, torch.nn.CrossEntropyLoss(), args,
                torch.device("cpu"), output_dir=str(tmp_path))
    assert glob.glob(str(tmp_path / "train-batch-*.jpg"))


def test_fused_head_gating_logic():
    """_use_fused_head: eligible only for plain single-GPU CE-family setups."""
    import deepfake_detection_amd as dfd
    from deepfake_detection_amd.engine import _use_fused_head
    from deepfake_detection_amd.loss import LabelSmoothingCrossEntropy

    m = dfd.create_model("efficientnet_b0", num_classes=2)
    t = torch.randint(0, 2, (4,))
    ce = torch.nn.CrossEntropyLoss()
    # CPU (use_cuda False) -> no
    assert not _use_fused_head(m, ce, t, use_cuda=False)
    # soft targets -> no
    soft = torch.rand(4, 2)
    assert not _use_fused_head(m, ce, soft, use_cuda=True)
    # smoothing via torch CE kwarg -> no (fused op wants explicit smoothing)
    assert not _use_fused_head(m, torch.nn.CrossEntropyLoss(label_smoothing=0.1),
                               t, use_cuda=True)
    # JSD-style loss -> no
    assert not _use_fused_head(m, torch.nn.MSELoss(), t, use_cuda=True)
    # DataParallel wrapper -> no
    assert not _use_fused_head(torch.nn.DataParallel(m), ce, t, use_cuda=True)
    # eligible shape (actual GPU dispatch still gated by gpu_ops_required)
    ls = LabelSmoothingCrossEntropy(0.1)
    assert isinstance(ls.smoothing, float) or ls.smoothing == 0.1
