"""Distributed training runner (CLI).

Capability parity with reference dfd/runners/train.py: two-stage argparse
with YAML config override (:53-253), server-JSON cluster topology +
mp.spawn per-GPU launch (:769-816), per-rank setup / model / optimizer /
loaders / epoch loop / validation / EMA / checkpointing (:256-592).

MI355X-native deltas:
  * one process per GPU over RCCL ("nccl" backend IS RCCL on ROCm);
    torchrun env-based rendezvous is first-class; the reference's
    file-store + server-JSON + mp.spawn path is kept for compat.
  * bf16 autocast (CDNA4 MFMA) instead of apex AMP O1; no loss scaling.
  * torch-DDP bucketed all-reduce overlapped with backward (parallel/
    distributed.py) instead of apex delay_allreduce.
  * channels_last model + activations end-to-end.

Usage (single node, 8 GPUs):
  torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 8 \
      -m deepfake_detection_amd.runners.train --data /path -c cfg.yaml ...
or the reference-style launcher:
  python -m deepfake_detection_amd.runners.train --json_file servers.json ...
"""

import argparse
import logging
import os
import time
from datetime import datetime

import torch
import torch.multiprocessing as mp
import yaml

from .. import create_deepfake_model_v4, create_model
from ..data import (
    DeepFakeDataset_v3,
    SyntheticDeepFakeDataset,
    create_deepfake_loader_v3,
    resolve_data_config,
)
from ..engine import train_epoch, validate
from ..loss import LabelSmoothingCrossEntropy, SoftTargetCrossEntropy
from ..optim import create_optimizer
from ..parallel import distribute_bn, init_distributed, parse_server, wrap_ddp
from ..parallel.distributed import convert_sync_batchnorm
from ..scheduler import create_scheduler
from ..utils import CheckpointSaver, ModelEma, setup_default_logging, update_summary
from ..models import resume_checkpoint

_logger = logging.getLogger("train")


def _build_parser():
    parser = argparse.ArgumentParser(description="MI355X deepfake-detection training")
    # Dataset / model
    parser.add_argument("--data", default=None, type=str, metavar="DIR")
    parser.add_argument("--synthetic-data", action="store_true", default=False,
                        help="use the synthetic face-crop dataset (benching)")
    parser.add_argument("--synthetic-len", type=int, default=2048)
    parser.add_argument("--validation_frac", default=0.1, type=float,
                        help="val split when --train_frac is default")
    parser.add_argument("--train_frac", default=1, type=float,
                        help="train split ratio (reference train.py:65,437)")
    parser.add_argument("--label_balance", action="store_true", default=False)
    parser.add_argument("--model", default="efficientnet_deepfake_v4", type=str, metavar="MODEL")
    parser.add_argument("--class_names", default="fake,real", type=str)
    parser.add_argument("--share_file", default="", type=str, metavar="SF")
    parser.add_argument("--master_share_file", default="", type=str, metavar="SF")
    parser.add_argument("--json_file", type=str, default=None, metavar="JS")
    parser.add_argument("--model-version", default=None, type=str)
    parser.add_argument("--input-size-v2", default=None, type=str,
                        help='input size "channel,height,width"')
    parser.add_argument("--pretrained", action="store_true", default=False)
    parser.add_argument("--initial-checkpoint", default="", type=str, metavar="PATH")
    parser.add_argument("--resume", default="", type=str, metavar="PATH")
    parser.add_argument("--no-resume-opt", action="store_true", default=False)
    parser.add_argument("--num-classes", type=int, default=2, metavar="N")
    parser.add_argument("--flicker", type=float, default=0.0)
    parser.add_argument("--gp", default="avg", type=str, metavar="POOL")
    parser.add_argument("--img-size", type=int, default=None, metavar="N")
    parser.add_argument("--crop-pct", default=None, type=float)
    parser.add_argument("--mean", type=float, nargs="+", default=None)
    parser.add_argument("--std", type=float, nargs="+", default=None)
    parser.add_argument("--interpolation", default="", type=str)
    parser.add_argument("-b", "--batch-size", type=int, default=32, metavar="N")
    parser.add_argument("-vb", "--validation-batch-size-multiplier", type=int, default=1)
    parser.add_argument("--drop", type=float, default=0.0, metavar="PCT")
    parser.add_argument("--drop-connect", type=float, default=None)
    parser.add_argument("--drop-path", type=float, default=None)
    parser.add_argument("--drop-block", type=float, default=None)
    parser.add_argument("--jsd", action="store_true", default=False)
    # Optimizer
    parser.add_argument("--opt", default="rmsproptf", type=str, metavar="OPTIMIZER")
    parser.add_argument("--opt-eps", default=1e-8, type=float)
    parser.add_argument("--momentum", type=float, default=0.9)
    parser.add_argument("--weight-decay", type=float, default=0.0001)
    # LR schedule
    parser.add_argument("--sched", default="step", type=str, metavar="SCHEDULER")
    parser.add_argument("--basic_lr", type=float, default=0.0000625, metavar="BLR")
    parser.add_argument("--lr", type=float, default=None,
                        help="explicit LR; default = batch_size*world_size*basic_lr")
    parser.add_argument("--lr-noise", type=float, nargs="+", default=None)
    parser.add_argument("--lr-noise-pct", type=float, default=0.67)
    parser.add_argument("--lr-noise-std", type=float, default=1.0)
    parser.add_argument("--warmup-lr", type=float, default=0.0001)
    parser.add_argument("--min-lr", type=float, default=1e-5)
    parser.add_argument("--epochs", type=int, default=200, metavar="N")
    parser.add_argument("--start-epoch", default=None, type=int)
    parser.add_argument("--decay-epochs", type=float, default=30)
    parser.add_argument("--warmup-epochs", type=int, default=3)
    parser.add_argument("--cooldown-epochs", type=int, default=10)
    parser.add_argument("--patience-epochs", type=int, default=10)
    parser.add_argument("--decay-rate", "--dr", type=float, default=0.1)
    # Augmentation
    parser.add_argument("--blur_radiu", type=int, default=1)
    parser.add_argument("--blur_prob", type=float, default=0)
    parser.add_argument("--color-jitter", type=float, default=0.2)
    parser.add_argument("--aa", type=str, default=None, metavar="NAME",
                        help="AutoAugment policy ('v0', 'original', "
                             "'rand-m9-mstd0.5', 'augmix-m3-w3')")
    parser.add_argument("--aug-splits", type=int, default=0)
    parser.add_argument("--reprob", type=float, default=0.0)
    parser.add_argument("--remode", type=str, default="pixel")
    parser.add_argument("--recount", type=int, default=1)
    parser.add_argument("--rotate_range", type=int, default=0)
    parser.add_argument("--remax", type=float, default=0.02)
    parser.add_argument("--resplit", action="store_true", default=False)
    parser.add_argument("--mixup", type=float, default=0.0)
    parser.add_argument("--mixup-off-epoch", default=0, type=int)
    parser.add_argument("--smoothing", type=float, default=0.1)
    parser.add_argument("--train-interpolation", type=str, default="random")
    # BatchNorm
    parser.add_argument("--bn-tf", action="store_true", default=False)
    parser.add_argument("--bn-momentum", type=float, default=None)
    parser.add_argument("--bn-eps", type=float, default=None)
    parser.add_argument("--sync-bn", action="store_true")
    parser.add_argument("--dist-bn", type=str, default="")
    parser.add_argument("--split-bn", action="store_true")
    # EMA
    parser.add_argument("--model-ema", action="store_true", default=False)
    parser.add_argument("--model-ema-force-cpu", action="store_true", default=False)
    parser.add_argument("--model-ema-decay", type=float, default=0.9998)
    # Misc
    parser.add_argument("--seed", type=int, default=42, metavar="S")
    parser.add_argument("--log-interval", type=int, default=50)
    parser.add_argument("--recovery-interval", type=int, default=0)
    parser.add_argument("-j", "--workers", type=int, default=4)
    parser.add_argument("--num-gpu", type=int, default=1)
    parser.add_argument("--save-images", action="store_true", default=False)
    parser.add_argument("--amp", action="store_true", default=True,
                        help="bf16 autocast (default on; CDNA4 MFMA path)")
    parser.add_argument("--no-amp", dest="amp", action="store_false")
    parser.add_argument("--pin-mem", action="store_true", default=True)
    parser.add_argument("--no-prefetcher", action="store_true", default=False)
    parser.add_argument("--output", default="", type=str, metavar="PATH")
    parser.add_argument("--eval-metric", default="prec1", type=str)
    parser.add_argument("--tta", type=int, default=0)
    parser.add_argument("--per-step-metrics", action="store_true", default=False,
                        help="reference train.py:625-645 every-step meter semantics (syncs per step)")
    parser.add_argument("--local_rank", default=0, type=int)
    return parser


def _parse_args(args=None):
    """Two-stage parse: -c/--config YAML keys become defaults of the main
    parser (reference train.py:53-57,238-253)."""
    config_parser = argparse.ArgumentParser(description="Training Config", add_help=False)
    config_parser.add_argument("-c", "--config", default="", type=str, metavar="FILE")

    parser = _build_parser()
    args_config, remaining = config_parser.parse_known_args(args)
    if args_config.config:
        with open(args_config.config, "r") as f:
            cfg = yaml.safe_load(f)
        parser.set_defaults(**cfg)

    parsed = parser.parse_args(remaining)
    args_text = yaml.safe_dump(parsed.__dict__, default_flow_style=False)
    return parsed, args_text


def main(rank, args, args_text, world_size=None, start_rank=0):
    args.prefetcher = not args.no_prefetcher
    device, world_size, whole_rank, local_rank = init_distributed(
        world_size=world_size, rank=start_rank + rank if world_size else None,
        local_rank=rank if world_size else None)
    args.distributed = world_size > 1
    use_cuda = device.type == "cuda"

    torch.manual_seed(args.seed + whole_rank)

    in_chans = 12
    if args.input_size_v2:
        in_chans = int(args.input_size_v2.split(",")[0])

    if getattr(args, "drop_connect", None):
        # DEPRECATED alias (reference train.py:311): drop_connect -> drop_path
        args.drop_path = args.drop_connect
    if args.model == "efficientnet_deepfake_v4":
        model = create_deepfake_model_v4(
            args.model, pretrained=args.pretrained, num_classes=args.num_classes,
            in_chans=in_chans, drop_rate=args.drop, drop_path_rate=args.drop_path,
            global_pool=args.gp, bn_tf=args.bn_tf, bn_momentum=args.bn_momentum,
            bn_eps=args.bn_eps, checkpoint_path=args.initial_checkpoint)
    else:
        model = create_model(
            args.model, pretrained=args.pretrained, num_classes=args.num_classes,
            in_chans=in_chans, drop_rate=args.drop, drop_path_rate=args.drop_path,
            global_pool=args.gp, bn_tf=args.bn_tf, bn_momentum=args.bn_momentum,
            bn_eps=args.bn_eps, checkpoint_path=args.initial_checkpoint)

    data_config = resolve_data_config(vars(args), model=model, verbose=whole_rank == 0)

    # AugMix splits + auxiliary split-BN (reference train.py:330-337)
    num_aug_splits = 0
    if args.aug_splits > 0:
        assert args.aug_splits > 1, "A split of 1 makes no sense"
        num_aug_splits = args.aug_splits
    if args.split_bn:
        assert num_aug_splits > 1 or args.resplit
        from ..models.layers_extra import convert_splitbn_model

        model = convert_splitbn_model(model, max(num_aug_splits, 2))

    model = model.to(device)
    if use_cuda:
        model = model.to(memory_format=torch.channels_last)

    # linear LR scaling (reference train.py:814)
    if args.lr is None:
        args.lr = args.batch_size * world_size * args.basic_lr

    optimizer = create_optimizer(args, model)

    resume_state = {}
    resume_epoch = None
    if args.resume:
        resume_state, resume_epoch = resume_checkpoint(model, args.resume)
    if resume_state and not args.no_resume_opt and "optimizer" in resume_state:
        optimizer.load_state_dict(resume_state["optimizer"])

    model_ema = None
    if args.model_ema:
        model_ema = ModelEma(
            model, decay=args.model_ema_decay,
            device="cpu" if args.model_ema_force_cpu else "",
            resume=args.resume)

    if args.distributed:
        if args.sync_bn:
            model = convert_sync_batchnorm(model)
        model = wrap_ddp(model, device)

    lr_scheduler, num_epochs = create_scheduler(args, optimizer)
    start_epoch = 0
    if args.start_epoch is not None:
        start_epoch = args.start_epoch
    elif resume_epoch is not None:
        start_epoch = resume_epoch
    if lr_scheduler is not None and start_epoch > 0:
        lr_scheduler.step(start_epoch)

    # datasets
    img_size = int(data_config["input_size"][-1])
    if args.synthetic_data or args.data is None:
        img_num = max(1, in_chans // 3)
        dataset_train = SyntheticDeepFakeDataset(
            length=args.synthetic_len, img_size=img_size, img_num=img_num)
        dataset_eval = SyntheticDeepFakeDataset(
            length=max(64, args.synthetic_len // 8), img_size=img_size, img_num=img_num, seed=1)
        loader_train = _synthetic_loader(dataset_train, args, data_config, is_training=True)
        loader_eval = _synthetic_loader(dataset_eval, args, data_config, is_training=False)
    else:
        train_ratio = args.train_frac if args.train_frac < 1 else 1.0 - args.validation_frac
        dataset_train = DeepFakeDataset_v3(
            args.data, args.class_names, train_split=True,
            train_ratio=train_ratio, random_state=args.seed,
            is_training=True, label_balance=args.label_balance)
        dataset_eval = DeepFakeDataset_v3(
            args.data, args.class_names, train_split=True,
            train_ratio=train_ratio, random_state=args.seed,
            is_training=False, label_balance=args.label_balance)
        # collate-time mixup when the device prefetcher owns normalization
        # (reference train.py:442-445)
        collate_fn = None
        if args.prefetcher and args.mixup > 0:
            assert not num_aug_splits, "mixup collate conflicts with aug splits"
            from ..data import FastCollateMixup

            collate_fn = FastCollateMixup(args.mixup, args.smoothing, args.num_classes)
        if num_aug_splits > 1:
            from ..data.dataset import AugMixDataset

            dataset_train = AugMixDataset(dataset_train, num_splits=num_aug_splits)
        loader_train = create_deepfake_loader_v3(
            dataset_train, input_size=data_config["input_size"], batch_size=args.batch_size,
            collate_fn=collate_fn,
            is_training=True, use_prefetcher=args.prefetcher, re_prob=args.reprob,
            re_mode=args.remode, re_count=args.recount, re_split=args.resplit,
            re_max=args.remax, color_jitter=args.color_jitter,
            auto_augment=args.aa, num_aug_splits=num_aug_splits,
            mean=data_config["mean"], std=data_config["std"], num_workers=args.workers,
            distributed=args.distributed, pin_memory=args.pin_mem,
            fp16=False, dtype="bfloat16" if args.amp else "float32",
            flicker=args.flicker, rotate_range=args.rotate_range,
            blur_radiu=args.blur_radiu, blur_prob=args.blur_prob)
        loader_eval = create_deepfake_loader_v3(
            dataset_eval, input_size=data_config["input_size"],
            batch_size=args.validation_batch_size_multiplier * args.batch_size,
            is_training=False, use_prefetcher=args.prefetcher,
            mean=data_config["mean"], std=data_config["std"], num_workers=args.workers,
            distributed=args.distributed, pin_memory=args.pin_mem,
            fp16=False, dtype="bfloat16" if args.amp else "float32")

    # loss selection (reference train.py:506-520)
    if args.jsd:
        assert num_aug_splits > 1, "JSD only valid with aug splits set"
        from ..loss import JsdCrossEntropy

        train_loss_fn = JsdCrossEntropy(num_splits=num_aug_splits,
                                        smoothing=args.smoothing).to(device)
    elif args.mixup > 0.0:
        train_loss_fn = SoftTargetCrossEntropy().to(device)
    elif args.smoothing:
        train_loss_fn = LabelSmoothingCrossEntropy(smoothing=args.smoothing).to(device)
    else:
        train_loss_fn = torch.nn.CrossEntropyLoss().to(device)
    validate_loss_fn = torch.nn.CrossEntropyLoss().to(device)

    eval_metric = args.eval_metric
    best_metric = None
    best_epoch = None
    saver = None
    output_dir = ""
    if whole_rank == 0:
        output_base = args.output if args.output else "./output"
        exp_name = "-".join([
            args.model_version or datetime.now().strftime("%Y%m%d-%H%M%S"), args.model])
        output_dir = os.path.join(output_base, exp_name)
        backup_dir = output_dir + "_bak"
        os.makedirs(output_dir, exist_ok=True)
        os.makedirs(backup_dir, exist_ok=True)
        decreasing = eval_metric == "loss"
        saver = CheckpointSaver(
            checkpoint_dir=output_dir, recovery_dir=output_dir, backup_dir=backup_dir,
            decreasing=decreasing)
        with open(os.path.join(output_dir, "args.yaml"), "w") as f:
            f.write(args_text)

    try:
        for epoch in range(start_epoch, num_epochs):
            if hasattr(dataset_train, "set_epoch"):
                dataset_train.set_epoch(epoch)
            if args.distributed and hasattr(loader_train.sampler, "set_epoch"):
                loader_train.sampler.set_epoch(epoch)

            if args.distributed and args.dist_bn in ("broadcast", "reduce"):
                distribute_bn(model, world_size, args.dist_bn == "reduce")

            # mixup cutoff (reference train.py:597-599)
            if args.prefetcher and args.mixup > 0 and getattr(loader_train, "mixup_enabled", False):
                if args.mixup_off_epoch and epoch >= args.mixup_off_epoch:
                    loader_train.mixup_enabled = False

            train_metrics = train_epoch(
                epoch, model, loader_train, optimizer, train_loss_fn, args, device,
                lr_scheduler=lr_scheduler, saver=saver, output_dir=output_dir,
                model_ema=model_ema, world_size=world_size, rank=whole_rank)

            eval_metrics = validate(model, loader_eval, validate_loss_fn, args, device,
                                    world_size=world_size, rank=whole_rank)
            if model_ema is not None and not args.model_ema_force_cpu:
                if args.distributed and args.dist_bn in ("broadcast", "reduce"):
                    distribute_bn(model_ema.ema, world_size, args.dist_bn == "reduce")
                ema_eval_metrics = validate(
                    model_ema.ema, loader_eval, validate_loss_fn, args, device,
                    world_size=world_size, rank=whole_rank, log_suffix=" (EMA)")
                eval_metrics = ema_eval_metrics

            if lr_scheduler is not None:
                lr_scheduler.step(epoch + 1, eval_metrics[eval_metric])

            if output_dir:
                update_summary(
                    epoch, train_metrics, eval_metrics,
                    os.path.join(output_dir, "summary.csv"),
                    write_header=best_metric is None)

            if saver is not None:
                best_metric, best_epoch = saver.save_checkpoint(
                    model, optimizer, args, epoch=epoch, model_ema=model_ema,
                    metric=eval_metrics[eval_metric])
    except KeyboardInterrupt:
        pass
    if best_metric is not None:
        _logger.info("*** Best metric: {0} (epoch {1})".format(best_metric, best_epoch))


def _synthetic_loader(dataset, args, data_config, is_training):
    """Loader for the synthetic dataset: fast_collate + device prefetcher,
    skipping the PIL transform stage (tensors come pre-shaped)."""
    from ..data.loader import PrefetchLoader_v3, fast_collate

    loader = torch.utils.data.DataLoader(
        dataset, batch_size=args.batch_size, shuffle=is_training,
        num_workers=args.workers, collate_fn=fast_collate,
        pin_memory=args.pin_mem, drop_last=is_training,
        persistent_workers=args.workers > 0)
    return PrefetchLoader_v3(
        loader, mean=data_config["mean"], std=data_config["std"],
        dtype="bfloat16" if args.amp else "float32",
        img_num=max(1, data_config["input_size"][0] // 3))


def launch_main(argv=None):
    setup_default_logging()
    args, args_text = _parse_args(argv)

    if "WORLD_SIZE" in os.environ and int(os.environ.get("WORLD_SIZE", "1")) >= 1 \
            and "RANK" in os.environ:
        # torchrun path: env rendezvous, this process IS one rank
        main(int(os.environ.get("LOCAL_RANK", 0)), args, args_text)
        return

    if args.json_file:
        # reference-style launcher: topology JSON + spawn one proc per GPU
        hostname, gpus, world_size, local_size, start_rank = parse_server(args.json_file)
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29500")
        os.environ["WORLD_SIZE"] = str(world_size)
        if args.share_file and os.path.exists(args.share_file) and start_rank == 0:
            os.remove(args.share_file)
        os.environ["CUDA_VISIBLE_DEVICES"] = gpus
        if local_size > 1:
            ctx = mp.spawn(_spawn_worker, nprocs=local_size,
                           args=(args, args_text, world_size, start_rank), join=True)
        else:
            main(0, args, args_text, world_size=world_size, start_rank=start_rank)
        return

    # single process
    main(0, args, args_text)


def _spawn_worker(rank, args, args_text, world_size, start_rank):
    os.environ["RANK"] = str(start_rank + rank)
    os.environ["LOCAL_RANK"] = str(rank)
    main(rank, args, args_text, world_size=world_size, start_rank=start_rank)


if __name__ == "__main__":
    launch_main()
