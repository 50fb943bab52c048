#!/usr/bin/env python3
"""Flagship training benchmark — the driver contract.

Measures the BASELINE.json metric: images/sec (whole node) for
EfficientNet-B4 299px bf16 training on synthetic face crops with
random-init weights, at 1..8 MI355X (one rank per GPU over RCCL when
launched by torchrun / torch.distributed.run).

Timing protocol: W untimed warmup steps, then EXACTLY K timed steps
bracketed by dist.barrier() + torch.cuda.synchronize() on both sides;
elapsed = MAX over ranks; rank 0 prints ONE JSON line.
"""

import argparse
import json
import os
import time

import torch


def parse_args():
    p = argparse.ArgumentParser(description="MI355X deepfake-detection bench")
    p.add_argument("--gpus", type=int, default=1, help="world size expected (informational)")
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--model", default="efficientnet_b4")
    p.add_argument("--img-size", type=int, default=299)
    p.add_argument("--in-chans", type=int, default=3)
    # per-GPU micro-batch sized for 288 GB HBM3E: 1536 at B4-299 bf16 uses
    # 218 GB allocated / 252 GB reserved and measures 2697 img/s (768: 2504,
    # 384: 2085). The r01 default of 384 only existed to dodge MIOpen find
    # time — every conv now runs on in-tree HIP kernels, so a fresh process
    # reaches the first timed step in seconds.
    p.add_argument("--batch-size", type=int, default=1536, help="per-GPU micro-batch")
    p.add_argument("--num-classes", type=int, default=2)
    p.add_argument("--opt", default="rmsproptf", choices=["rmsproptf", "adamw"])
    p.add_argument("--lr", type=float, default=1e-4)
    p.add_argument("--dtype", default="bf16", choices=["bf16", "fp32"])
    p.add_argument("--graph", action="store_true", default=True,
                   help="capture the train step in a hipGraph (single-GPU only; "
                        "auto-disabled when distributed)")
    p.add_argument("--no-graph", dest="graph", action="store_false")
    p.add_argument("--channels-last", dest="channels_last", action="store_true", default=True)
    p.add_argument("--no-fused-ops", action="store_true", default=False,
                   help="A/B: run plain torch ops instead of the HIP kernels")
    return p.parse_args()


def main():
    args = parse_args()
    if args.no_fused_ops:
        os.environ["DFD_AMD_FORCE_TORCH_OPS"] = "1"

    import deepfake_detection_amd as dfd
    from deepfake_detection_amd.optim import AdamW, RMSpropTF, add_weight_decay

    world_size = int(os.environ.get("WORLD_SIZE", 1))
    rank = int(os.environ.get("RANK", 0))
    local_rank = int(os.environ.get("LOCAL_RANK", rank))

    use_cuda = torch.cuda.is_available()
    if use_cuda:
        torch.cuda.set_device(local_rank)
        device = torch.device("cuda", local_rank)
    else:
        device = torch.device("cpu")

    distributed = world_size > 1
    if distributed:
        import torch.distributed as dist

        dist.init_process_group("nccl" if use_cuda else "gloo",
                                world_size=world_size, rank=rank)

    torch.manual_seed(42 + rank)
    model = dfd.create_model(args.model, num_classes=args.num_classes,
                             in_chans=args.in_chans)
    model = model.to(device)
    if use_cuda and args.channels_last:
        model = model.to(memory_format=torch.channels_last)

    if distributed:
        from deepfake_detection_amd.parallel import wrap_ddp

        run_model = wrap_ddp(model, device)
    else:
        run_model = model

    params = add_weight_decay(model, 1e-5)
    if args.opt == "adamw":
        optimizer = AdamW(params, lr=args.lr, weight_decay=0.0)
    else:
        optimizer = RMSpropTF(params, lr=args.lr, alpha=0.9, eps=1e-3, momentum=0.9)

    loss_fn = torch.nn.CrossEntropyLoss().to(device)
    use_bf16 = args.dtype == "bf16" and use_cuda

    # synthetic face crops: a pool of resident uint8 batches; each step runs
    # the device-side normalize (the prefetcher's work) + fwd + bwd + step.
    from deepfake_detection_amd.ops.prefetch_ops import normalize_uint8

    n_pool = 4
    g = torch.Generator(device="cpu").manual_seed(1234 + rank)
    pool = [
        torch.randint(0, 256, (args.batch_size, args.in_chans, args.img_size, args.img_size),
                      dtype=torch.uint8, generator=g).to(device)
        for _ in range(n_pool)
    ]
    targets = [
        torch.randint(0, args.num_classes, (args.batch_size,), generator=g).to(device)
        for _ in range(n_pool)
    ]
    mean = torch.tensor([0.485, 0.456, 0.406] * (args.in_chans // 3 or 1),
                        device=device)[: args.in_chans] * 255
    std = torch.tensor([0.229, 0.224, 0.225] * (args.in_chans // 3 or 1),
                       device=device)[: args.in_chans] * 255
    mean = mean.view(1, args.in_chans, 1, 1)
    std = std.view(1, args.in_chans, 1, 1)
    norm_dtype = torch.bfloat16 if use_bf16 else torch.float32

    # fused classifier+CE head (ops/head.py) on the single-GPU path; DDP
    # must go through run_model.forward for the reducer hooks
    from deepfake_detection_amd.ops import functional as O
    from deepfake_detection_amd.ops.extension import gpu_ops_required

    use_fused_head = (use_cuda and not distributed and not args.no_fused_ops
                      and gpu_ops_required()
                      and hasattr(model, "forward_features")
                      and isinstance(model.classifier, torch.nn.Linear))

    def one_step(i):
        x = normalize_uint8(pool[i % n_pool], mean, std, out_dtype=norm_dtype,
                            channels_last=args.channels_last and use_cuda)
        t = targets[i % n_pool]
        if use_fused_head:
            from deepfake_detection_amd.ops.head import fused_head_ce

            with torch.autocast("cuda", torch.bfloat16, enabled=use_bf16):
                feats = model.forward_features(x)
                pooled = O.global_avg_pool(feats)
            loss, _ = fused_head_ce(pooled, model.classifier.weight,
                                    model.classifier.bias, t)
        elif use_bf16:
            with torch.autocast("cuda", torch.bfloat16):
                out = run_model(x)
                loss = loss_fn(out, t)
        else:
            out = run_model(x)
            loss = loss_fn(out, t)
        optimizer.zero_grad(set_to_none=True)
        loss.backward()
        optimizer.step()
        return loss

    def barrier_sync():
        if distributed:
            import torch.distributed as dist

            dist.barrier()
        if use_cuda:
            torch.cuda.synchronize()

    for i in range(args.warmup):
        one_step(i)
    barrier_sync()

    graph = None
    if args.graph and use_cuda and not distributed:
        # capture the whole train step in a hipGraph; replay K times. If a
        # fallback op in this model config is not capture-safe, log and run
        # eager — the bench must never die on the capture path.
        try:
            static_idx = 0
            graph = torch.cuda.CUDAGraph()
            one_step(static_idx)
            torch.cuda.synchronize()
            with torch.cuda.graph(graph):
                one_step(static_idx)
            barrier_sync()
        except Exception as e:  # noqa: BLE001
            import sys

            print(f"[bench] hipGraph capture failed ({e}); running eager",
                  file=sys.stderr)
            graph = None
            torch.cuda.synchronize()

    t0 = time.perf_counter()
    for i in range(args.steps):
        if graph is not None:
            graph.replay()
        else:
            one_step(args.warmup + i)
    barrier_sync()
    elapsed = time.perf_counter() - t0

    if distributed:
        import torch.distributed as dist

        e = torch.tensor([elapsed], device=device if use_cuda else None)
        dist.all_reduce(e, op=dist.ReduceOp.MAX)
        elapsed = e.item()

    global_batch = args.batch_size * world_size
    images_per_sec = args.steps * global_batch / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if rank == 0:
        result = {
            "metric": "images/sec (whole node) EfficientNet-B4 299px bf16 train at 1/2/4/8 MI355X",
            "value": images_per_sec,
            "unit": "images/sec",
            "n_gpus": world_size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if use_bf16 else "fp32",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": global_batch,
                "img_size": args.img_size,
                "in_chans": args.in_chans,
                "optimizer": args.opt,
                "parallelism": f"dp{world_size}",
                "graph": bool(graph is not None),
            },
        }
        print(json.dumps(result), flush=True)

    if distributed:
        import torch.distributed as dist

        dist.destroy_process_group()


if __name__ == "__main__":
    main()
