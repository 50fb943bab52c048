"""Multi-process CPU (gloo, world_size=2) distributed-plumbing tests —
BASELINE.json config 1: reduce_tensor, distribute_bn, DDP gradient sync,
server-JSON topology parsing."""

import json
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from deepfake_detection_amd.parallel import parse_server

WORLD = 2


def _run_dist(rank, world_size, fn, port, results):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world_size)
    try:
        out = fn(rank, world_size)
        results[rank] = out
    finally:
        dist.barrier()
        dist.destroy_process_group()


def _spawn(fn, port):
    ctx = mp.get_context("spawn")
    with ctx.Manager() as manager:
        results = manager.dict()
        mp.start_processes(
            _run_dist, args=(WORLD, fn, port, results), nprocs=WORLD,
            join=True, start_method="spawn")
        return dict(results)


def _reduce_worker(rank, world_size):
    from deepfake_detection_amd.parallel import reduce_tensor

    t = torch.tensor([float(rank + 1)])
    out = reduce_tensor(t, world_size)
    return out.item()  # (1+2)/2 = 1.5


def _bn_worker(rank, world_size):
    from deepfake_detection_amd.parallel import distribute_bn

    model = torch.nn.BatchNorm1d(3)
    with torch.no_grad():
        model.running_mean.fill_(float(rank))
        model.running_var.fill_(float(rank + 1))
    distribute_bn(model, world_size, reduce=True)
    return (model.running_mean.tolist(), model.running_var.tolist())


def _ddp_worker(rank, world_size):
    from deepfake_detection_amd.parallel import wrap_ddp

    torch.manual_seed(0)  # identical init on both ranks
    model = torch.nn.Linear(4, 2)
    device = torch.device("cpu")
    ddp = wrap_ddp(model, device, static_graph=False)
    opt = torch.optim.SGD(ddp.parameters(), lr=0.1)
    # different data per rank -> grads averaged by DDP
    torch.manual_seed(rank)
    x = torch.randn(8, 4)
    y = torch.randint(0, 2, (8,))
    loss = torch.nn.functional.cross_entropy(ddp(x), y)
    opt.zero_grad()
    loss.backward()
    opt.step()
    return model.weight.detach().flatten().tolist()


@pytest.mark.parametrize("fn,port", [(_reduce_worker, 29701)])
def test_reduce_tensor_gloo(fn, port):
    results = _spawn(fn, port)
    assert abs(results[0] - 1.5) < 1e-6
    assert abs(results[1] - 1.5) < 1e-6


def test_distribute_bn_gloo():
    results = _spawn(_bn_worker, 29702)
    for rank in (0, 1):
        mean, var = results[rank]
        assert all(abs(m - 0.5) < 1e-6 for m in mean)  # (0+1)/2
        assert all(abs(v - 1.5) < 1e-6 for v in var)  # (1+2)/2


def test_ddp_params_stay_synced():
    results = _spawn(_ddp_worker, 29703)
    assert results[0] == pytest.approx(results[1], abs=1e-6)


def test_parse_server(tmp_path):
    cfg = {"servers": [
        {"hostname": "nodeA", "gpus": "0,1,2,3"},
        {"hostname": "nodeB", "gpus": "0,1,2,3,4,5,6,7"},
    ]}
    p = tmp_path / "servers.json"
    p.write_text(json.dumps(cfg))
    hostname, gpus, world_size, local_size, start_rank = parse_server(str(p), hostname="nodeB")
    assert world_size == 12
    assert local_size == 8
    assert start_rank == 4
    assert gpus == "0,1,2,3,4,5,6,7"
    with pytest.raises(RuntimeError):
        parse_server(str(p), hostname="nodeC")


def _e2e_worker(rank, world_size):
    """Full engine step under DDP with a real EfficientNet (tiny input):
    the round-end multi-GPU scaling run exercises this exact path over RCCL."""
    import types

    import deepfake_detection_amd as dfd
    from deepfake_detection_amd.engine import train_epoch
    from deepfake_detection_amd.optim import RMSpropTF, add_weight_decay
    from deepfake_detection_amd.parallel import distribute_bn, wrap_ddp

    torch.manual_seed(0)
    model = dfd.create_model("efficientnet_b0", num_classes=2)
    device = torch.device("cpu")
    ddp = wrap_ddp(model, device, static_graph=False)
    opt = RMSpropTF(add_weight_decay(model, 1e-5), lr=1e-3, alpha=0.9,
                    eps=1e-3, momentum=0.9)
    torch.manual_seed(100 + rank)
    batches = [(torch.randn(2, 3, 64, 64), torch.randint(0, 2, (2,)))
               for _ in range(2)]
    args = types.SimpleNamespace(log_interval=100, recovery_interval=0,
                                 save_images=False, tta=0)
    metrics = train_epoch(0, ddp, batches, opt, torch.nn.CrossEntropyLoss(),
                          args, device, world_size=world_size, rank=rank)
    distribute_bn(ddp, world_size, reduce=True)
    return [p.detach().sum().item() for p in list(model.parameters())[:3]] + \
           [metrics["loss"]]


def test_ddp_engine_end_to_end_efficientnet():
    results = _spawn(_e2e_worker, 29704)
    # parameters identical across ranks after the DDP step; loss all-reduced
    assert results[0] == pytest.approx(results[1], abs=1e-5)
