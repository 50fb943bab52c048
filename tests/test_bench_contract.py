"""bench.py driver contract: one JSON line on stdout with the BASELINE.json
metric fields (the round driver parses this)."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract():
    res = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--model", "resnet18",
         "--img-size", "64", "--batch-size", "4", "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=600, cwd=REPO)
    assert res.returncode == 0, res.stderr[-2000:]
    lines = [l for l in res.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, res.stdout
    out = json.loads(lines[0])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in out, key
    assert out["metric"].startswith("images/sec")
    assert out["n_gpus"] == 1 and out["steps"] == 2 and out["warmup"] == 1
    assert out["value"] > 0 and out["higher_is_better"] is True
    assert out["scaling"] == "weak" and out["data"] == "synthetic"
    assert out["config"]["global_batch"] == 4
    assert out["config"]["parallelism"] == "dp1"


def test_bench_ws2_gloo_contract():
    """The driver launches bench.py under torch.distributed.run for N>1 —
    exercise that exact path at world_size 2 over gloo (CPU) so the
    round-end SCALE run can't hit a distributed-only bug."""
    env = dict(os.environ)
    env.pop("WORLD_SIZE", None)
    res = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--standalone",
         "--local-addr", "127.0.0.1", "--nproc-per-node", "2",
         os.path.join(REPO, "bench.py"), "--gpus", "2", "--model", "resnet18",
         "--img-size", "64", "--batch-size", "4", "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=900, cwd=REPO, env=env)
    assert res.returncode == 0, (res.stderr[-3000:], res.stdout[-500:])
    lines = [l for l in res.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, res.stdout  # rank 0 only
    out = json.loads(lines[0])
    assert out["n_gpus"] == 2
    assert out["config"]["parallelism"] == "dp2"
    assert out["config"]["global_batch"] == 8
    assert out["value"] > 0
