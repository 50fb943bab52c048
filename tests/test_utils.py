"""Utils tests: CheckpointSaver ranked retention + layout, EMA math,
meters, summary CSV."""

import glob
import os
import types

import torch

import deepfake_detection_amd as dfd
from deepfake_detection_amd.utils import (
    AverageMeter,
    CheckpointSaver,
    ModelEma,
    accuracy,
    update_summary,
)


def _tiny_model():
    return torch.nn.Sequential(torch.nn.Linear(4, 4), torch.nn.Linear(4, 2))


def test_checkpoint_saver_layout(tmp_path):
    model = _tiny_model()
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    args = types.SimpleNamespace(model="tiny")
    saver = CheckpointSaver(checkpoint_dir=str(tmp_path), recovery_dir=str(tmp_path),
                            decreasing=True, max_history=3)
    saver.save_checkpoint(model, opt, args, epoch=0, metric=1.0)
    ck = torch.load(str(tmp_path / "checkpoint-0.pth.tar"), weights_only=False)
    # reference save-state dict layout (timm/utils.py:97-112)
    for key in ("epoch", "arch", "state_dict", "optimizer", "args", "version"):
        assert key in ck, key
    assert ck["version"] == 2
    assert ck["arch"] == "tiny"
    assert os.path.exists(tmp_path / "model_best.pth.tar")


def test_checkpoint_saver_ranked_retention(tmp_path):
    model = _tiny_model()
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    args = types.SimpleNamespace(model="tiny")
    saver = CheckpointSaver(checkpoint_dir=str(tmp_path), recovery_dir=str(tmp_path),
                            decreasing=True, max_history=2)
    # losses: 1.0, 0.5, 2.0 -> keep the two best (0.5, 1.0)
    saver.save_checkpoint(model, opt, args, epoch=0, metric=1.0)
    saver.save_checkpoint(model, opt, args, epoch=1, metric=0.5)
    best_metric, best_epoch = saver.save_checkpoint(model, opt, args, epoch=2, metric=2.0)
    files = sorted(os.path.basename(f) for f in glob.glob(str(tmp_path / "checkpoint-*")))
    assert files == ["checkpoint-0.pth.tar", "checkpoint-1.pth.tar"]
    assert best_metric == 0.5 and best_epoch == 1


def test_checkpoint_recovery_cleanup(tmp_path):
    model = _tiny_model()
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    args = types.SimpleNamespace(model="tiny")
    saver = CheckpointSaver(checkpoint_dir=str(tmp_path), recovery_dir=str(tmp_path))
    saver.save_recovery(model, opt, args, epoch=0, batch_idx=10)
    saver.save_recovery(model, opt, args, epoch=0, batch_idx=20)
    saver.save_recovery(model, opt, args, epoch=0, batch_idx=30)
    files = glob.glob(str(tmp_path / "recovery-*"))
    assert len(files) == 2  # previous-previous removed
    assert saver.find_recovery()


def test_model_ema_math():
    m = torch.nn.Linear(2, 2, bias=False)
    with torch.no_grad():
        m.weight.fill_(1.0)
    ema = ModelEma(m, decay=0.9)
    with torch.no_grad():
        m.weight.fill_(2.0)
    ema.update(m)
    # ema = 1*0.9 + 2*0.1 = 1.1 (reference timm/utils.py:329-340)
    assert torch.allclose(ema.ema.weight, torch.full((2, 2), 1.1))


def test_resume_roundtrip_through_saver(tmp_path):
    model = dfd.create_model("efficientnet_lite0", num_classes=2)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    args = types.SimpleNamespace(model="efficientnet_lite0")
    saver = CheckpointSaver(checkpoint_dir=str(tmp_path), recovery_dir=str(tmp_path))
    saver.save_checkpoint(model, opt, args, epoch=5, metric=0.3)

    model2 = dfd.create_model("efficientnet_lite0", num_classes=2)
    from deepfake_detection_amd.models import resume_checkpoint

    other, epoch = resume_checkpoint(model2, str(tmp_path / "checkpoint-5.pth.tar"))
    assert epoch == 6
    assert "optimizer" in other
    assert torch.equal(model.conv_stem.weight, model2.conv_stem.weight)


def test_meters_and_accuracy():
    m = AverageMeter()
    m.update(1.0, 2)
    m.update(2.0, 2)
    assert m.avg == 1.5
    out = torch.tensor([[0.9, 0.1], [0.2, 0.8], [0.6, 0.4]])
    tgt = torch.tensor([0, 1, 1])
    (top1,) = accuracy(out, tgt)
    assert abs(top1.item() - 200.0 / 3) < 1e-4


def test_update_summary(tmp_path):
    f = str(tmp_path / "summary.csv")
    update_summary(0, {"loss": 1.0}, {"loss": 2.0, "prec1": 50.0}, f, write_header=True)
    update_summary(1, {"loss": 0.5}, {"loss": 1.5, "prec1": 60.0}, f)
    lines = open(f).read().strip().splitlines()
    assert lines[0] == "epoch,train_loss,eval_loss,eval_prec1"
    assert len(lines) == 3


def test_checkpoint_v4_interchange_with_reference_layout(tmp_path):
    """Full §3.5 round trip on the flagship model: saved dict carries
    module.-prefix-tolerant timm keys; resume returns epoch+1 for version>=2;
    EMA state under state_dict_ema; fp16-cast state loads back (the
    model_half.pth.tar path)."""
    import deepfake_detection_amd as dfd
    from deepfake_detection_amd.models import resume_checkpoint
    from deepfake_detection_amd.utils import CheckpointSaver, ModelEma

    model = dfd.create_model("efficientnet_b0", num_classes=2)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    ema = ModelEma(model, decay=0.9)
    args = types.SimpleNamespace(model="efficientnet_b0")
    saver = CheckpointSaver(checkpoint_dir=str(tmp_path), recovery_dir=str(tmp_path))
    saver.save_checkpoint(model, opt, args, epoch=3, model_ema=ema, metric=0.9)

    ck = torch.load(str(tmp_path / "checkpoint-3.pth.tar"), weights_only=False)
    assert "state_dict_ema" in ck and "metric" in ck
    keys = list(ck["state_dict"])
    assert keys[0].startswith("conv_stem") and keys[-1].startswith("classifier")

    # simulate a DDP-saved dict: module. prefix must be stripped on load
    ck["state_dict"] = {"module." + k: v for k, v in ck["state_dict"].items()}
    p = tmp_path / "ddp.pth.tar"
    torch.save(ck, str(p))
    model2 = dfd.create_model("efficientnet_b0", num_classes=2)
    resume_state, resume_epoch = resume_checkpoint(model2, str(p))
    assert resume_epoch == 4  # version>=2 -> epoch+1 (reference helpers.py:47-73)

    # fp16 "model_half" style weights load into an fp32 model (test.py path)
    half = {k: v.half() for k, v in model.state_dict().items()}
    torch.save({"state_dict": half}, str(tmp_path / "model_half.pth.tar"))
    model3 = dfd.create_model(
        "efficientnet_b0", num_classes=2,
        checkpoint_path=str(tmp_path / "model_half.pth.tar"))
    sd = model3.state_dict()
    ref = model.state_dict()
    assert torch.allclose(sd["conv_stem.weight"].float(),
                          ref["conv_stem.weight"].half().float())
