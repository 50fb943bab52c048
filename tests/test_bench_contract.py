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
