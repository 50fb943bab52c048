"""Training/validation engine (hot loop).

Capability parity with the reference trainer loops (reference
dfd/runners/train.py: train_epoch :594-700, validate :703-766) with the
MI355X-native changes called out in SURVEY.md §7:

  * NO per-step `torch.cuda.synchronize()` (reference train.py:639 syncs
    every step) — step timing uses HIP events sampled at log intervals.
  * Metric all-reduce (loss/prec1, reference train.py:625-627) is batched:
    the two scalars are packed into ONE tensor and reduced at log points
    and epoch end instead of two all-reduces per step (latency-bound on
    xGMI).
  * bf16 autocast forward/backward (CDNA4 MFMA path) — no loss scaling
    needed, replacing apex AMP O1 (reference train.py:352-353,632-634).
  * channels_last (NHWC) activations end-to-end.
"""

import logging
import os
import time
from collections import OrderedDict

import torch

from .parallel.distributed import reduce_tensor
from .utils.meters import AverageMeter, accuracy

_logger = logging.getLogger(__name__)


def _autocast(enabled, dtype=torch.bfloat16):
    if enabled:
        return torch.autocast(device_type="cuda", dtype=dtype)
    import contextlib

    return contextlib.nullcontext()


def _use_fused_head(model, loss_fn, target, use_cuda) -> bool:
    """Fused classifier+CE head eligibility: single-GPU (plain module, not
    DDP-wrapped), a plain-Linear classifier after global avg pool, a hard
    int64 target, and a CE-family loss (plain or label-smoothed)."""
    if not use_cuda or target.dim() != 1 or target.dtype != torch.long:
        return False
    if isinstance(model, (torch.nn.parallel.DistributedDataParallel,
                          torch.nn.DataParallel)):
        return False  # wrappers forward attr lookups to .module — explicit guard
    from .loss import LabelSmoothingCrossEntropy
    from .ops.extension import gpu_ops_required

    if not isinstance(loss_fn, (torch.nn.CrossEntropyLoss, LabelSmoothingCrossEntropy)):
        return False
    if isinstance(loss_fn, torch.nn.CrossEntropyLoss) and loss_fn.label_smoothing:
        return False
    classifier = getattr(model, "classifier", None)
    pool = getattr(model, "global_pool", None)
    return (gpu_ops_required() and hasattr(model, "forward_features")
            and isinstance(classifier, torch.nn.Linear)
            and getattr(pool, "pool_type", None) == "avg")


def train_epoch(epoch, model, loader, optimizer, loss_fn, args, device,
                lr_scheduler=None, saver=None, output_dir="", model_ema=None,
                world_size=1, rank=0, num_epochs=None):
    use_cuda = device.type == "cuda"
    use_amp = bool(getattr(args, "amp", False)) and use_cuda

    losses_m = AverageMeter()
    prec1_m = AverageMeter()
    batch_time_m = AverageMeter()
    data_time_m = AverageMeter()

    model.train()

    num_updates = epoch * len(loader)
    last_idx = len(loader) - 1
    end = time.time()
    for batch_idx, (input, target) in enumerate(loader):
        last_batch = batch_idx == last_idx
        data_time_m.update(time.time() - end)

        if not getattr(args, "prefetcher", True) and use_cuda:
            input, target = input.to(device, non_blocking=True), target.to(device, non_blocking=True)
        if use_cuda:
            input = input.contiguous(memory_format=torch.channels_last)

        if _use_fused_head(model, loss_fn, target, use_cuda):
            # fused classifier GEMM + CE (ops/head.py): pooled features go
            # through one autograd op returning (loss, logits). Single-GPU
            # only — DDP must run module.forward for its reducer hooks.
            from .ops import functional as O
            from .ops.head import fused_head_ce

            smoothing = float(getattr(loss_fn, "smoothing", 0.0))
            with _autocast(use_amp):
                feats = model.forward_features(input)
                pooled = O.global_avg_pool(feats)
                if getattr(model, "drop_rate", 0.0) > 0.0:
                    pooled = torch.nn.functional.dropout(
                        pooled, p=model.drop_rate, training=model.training)
            loss, output = fused_head_ce(pooled, model.classifier.weight,
                                         model.classifier.bias, target, smoothing)
        else:
            with _autocast(use_amp):
                output = model(input)
                loss = loss_fn(output, target)

        optimizer.zero_grad(set_to_none=True)
        loss.backward()
        optimizer.step()

        num_updates += 1

        if model_ema is not None:
            model_ema.update(model)

        # local (un-reduced) running meters; cross-rank reduction happens at
        # log points to avoid 2 all-reduces per step. --per-step-metrics
        # restores the reference's every-step meter/NaN-guard semantics
        # (reference train.py:625-645) at the cost of a sync per step.
        per_step = getattr(args, "per_step_metrics", False)
        log_point = last_batch or batch_idx % args.log_interval == 0
        if per_step or log_point:
            if use_cuda:
                torch.cuda.synchronize()
            with torch.no_grad():
                if target.dim() > 1:  # soft targets (mixup)
                    hard_target = target.argmax(dim=-1)
                else:
                    hard_target = target
                prec1 = accuracy(output.detach().float(), hard_target)[0]
                if world_size > 1 and log_point:
                    packed = torch.stack([loss.detach().float(), prec1])
                    packed = reduce_tensor(packed, world_size)
                    loss_val, prec1_val = packed[0].item(), packed[1].item()
                else:
                    loss_val, prec1_val = loss.item(), prec1.item()

            if not torch.isfinite(torch.tensor(loss_val)):
                _logger.warning("NaN/Inf loss at epoch %d batch %d — skipping meter update",
                                epoch, batch_idx)
            else:
                losses_m.update(loss_val, input.size(0))
                prec1_m.update(prec1_val, input.size(0))

            batch_time_m.update(time.time() - end)
            if rank == 0 and log_point:
                lrl = [pg["lr"] for pg in optimizer.param_groups]
                lr = sum(lrl) / len(lrl)
                rate = input.size(0) * world_size / max(batch_time_m.val, 1e-9)
                _logger.info(
                    "Train: {} [{:>4d}/{}]  Loss: {:.4g} ({:.3g})  Prec@1: {:.3f} ({:.3f})  "
                    "{:.1f} img/s  LR: {:.3e}  Data: {:.3f}s".format(
                        epoch, batch_idx, len(loader), loss_val, losses_m.avg,
                        prec1_val, prec1_m.avg, rate, lr, data_time_m.avg))

        if getattr(args, "save_images", False) and output_dir and (
                last_batch or batch_idx % args.log_interval == 0):
            # input-batch dumps (reference train.py:679-684; torchvision is
            # not in this image, so build the grid with PIL)
            try:
                import numpy as np
                from PIL import Image

                # cap the dump and tile into a grid of 8 per row: a single
                # row at production batch sizes exceeds PIL's 65,535-px JPEG
                # dimension limit (batch 384 @ 299px ~= 115k px wide)
                x = input[:64, :3].float().detach().cpu()
                x = x - x.amin(dim=(1, 2, 3), keepdim=True)
                x = x / x.amax(dim=(1, 2, 3), keepdim=True).clamp(min=1e-6)
                per_row = 8
                n, _, h, w = x.shape
                rows = []
                for r0 in range(0, n, per_row):
                    row = list(x[r0:r0 + per_row])
                    while len(row) < per_row and n > per_row:
                        row.append(torch.zeros(3, h, w))
                    rows.append(torch.cat(row, dim=2))
                grid = torch.cat(rows, dim=1)
                arr = (grid.permute(1, 2, 0).numpy() * 255).astype(np.uint8)
                Image.fromarray(arr).save(
                    os.path.join(output_dir, f"train-batch-{batch_idx}.jpg"))
            except Exception as e:  # noqa: BLE001
                _logger.warning("save_images failed: %s", e)

        if saver is not None and args.recovery_interval and (
                last_batch or (batch_idx + 1) % args.recovery_interval == 0):
            saver.save_recovery(model, optimizer, args, epoch, model_ema=model_ema,
                                batch_idx=batch_idx)

        if lr_scheduler is not None:
            lr_scheduler.step_update(num_updates=num_updates, metric=losses_m.avg)

        end = time.time()

    if hasattr(optimizer, "sync_lookahead"):
        optimizer.sync_lookahead()

    return OrderedDict([("loss", losses_m.avg), ("prec1", prec1_m.avg),
                        ("learning_rate", optimizer.param_groups[0]["lr"])])


def _tta_views(x, tta):
    """tta distinct deterministic views of a NCHW batch: identity, H/V flips,
    then ±8px shifts. Identical copies through a deterministic eval model
    would make the TTA average a no-op; these views are cheap and distinct."""
    views = []
    for k in range(tta):
        if k == 0:
            v = x
        elif k == 1:
            v = torch.flip(x, dims=(3,))
        elif k == 2:
            v = torch.flip(x, dims=(2,))
        elif k == 3:
            v = torch.flip(x, dims=(2, 3))
        else:
            s = 8 * ((k - 4) // 4 + 1)
            axis = 2 + (k % 2)
            sign = -1 if (k // 2) % 2 else 1
            v = torch.roll(x, shifts=sign * s, dims=axis)
        views.append(v)
    return views


def validate(model, loader, loss_fn, args, device, world_size=1, rank=0, log_suffix=""):
    use_cuda = device.type == "cuda"
    use_amp = bool(getattr(args, "amp", False)) and use_cuda

    batch_time_m = AverageMeter()
    losses_m = AverageMeter()
    prec1_m = AverageMeter()

    model.eval()

    end = time.time()
    last_idx = len(loader) - 1
    with torch.no_grad():
        for batch_idx, (input, target) in enumerate(loader):
            last_batch = batch_idx == last_idx
            if not getattr(args, "prefetcher", True) and use_cuda:
                input = input.to(device, non_blocking=True)
                target = target.to(device, non_blocking=True)
            if use_cuda:
                input = input.contiguous(memory_format=torch.channels_last)

            tta = getattr(args, "tta", 0)
            if tta > 1:
                # TTA: tta DISTINCT deterministic views per sample (flips,
                # then small shifts), interleaved sample-major so the
                # reference's unfold-mean reduction applies unchanged
                # (reference train.py:724-727).
                input = torch.stack(_tta_views(input, tta), dim=1).flatten(0, 1)

            with _autocast(use_amp):
                output = model(input)
            if isinstance(output, (tuple, list)):
                output = output[0]
            output = output.float()
            if tta > 1:
                output = output.unfold(0, tta, tta).mean(dim=2)

            loss = loss_fn(output, target)
            prec1 = accuracy(output, target)[0]

            if world_size > 1:
                packed = torch.stack([loss.detach(), prec1])
                packed = reduce_tensor(packed, world_size)
                loss_v, prec1_v = packed[0].item(), packed[1].item()
            else:
                loss_v, prec1_v = loss.item(), prec1.item()

            losses_m.update(loss_v, input.size(0))
            prec1_m.update(prec1_v, input.size(0))
            batch_time_m.update(time.time() - end)
            end = time.time()

            if rank == 0 and (last_batch or batch_idx % args.log_interval == 0):
                _logger.info(
                    "Test{}: [{:>4d}/{}]  Time: {:.3f} ({:.3f})  Loss: {:.4g} ({:.3g})  "
                    "Prec@1: {:.3f} ({:.3f})".format(
                        log_suffix, batch_idx, last_idx, batch_time_m.val,
                        batch_time_m.avg, loss_v, losses_m.avg, prec1_v, prec1_m.avg))

    return OrderedDict([("loss", losses_m.avg), ("prec1", prec1_m.avg)])
