"""bench.py contract tests: single-process CPU run and the driver's
torchrun world=2 launch (gloo on CPU), validating the JSON line schema."""
import json
import os
import subprocess
import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parents[1]

REQUIRED_KEYS = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
                 "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                 "dtype", "data", "config"}


def _last_json_line(stdout: str) -> dict:
    for line in reversed(stdout.strip().splitlines()):
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output: {stdout[-500:]}")


@pytest.mark.slow
def test_bench_single_process_cpu():
    r = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=600, cwd=REPO)
    assert r.returncode == 0, r.stderr[-2000:]
    res = _last_json_line(r.stdout)
    assert REQUIRED_KEYS <= set(res)
    assert res["n_gpus"] == 1
    assert res["metric"] == "images/sec"
    # default invocation measures BOTH headline models (BASELINE.json names
    # "ResNet-50 & ViT-B/16"); value is the ResNet-50 number
    assert res["config"]["model"] == "resnet50 & vit_b16"
    assert set(res["config"]["models"]) == {"resnet50", "vit_b16"}
    assert res["config"]["models"]["vit_b16"]["images_per_sec"] > 0
    assert res["value"] == res["config"]["models"]["resnet50"]["images_per_sec"]
    assert res["value"] > 0


@pytest.mark.slow
@pytest.mark.timeout(600)
def test_bench_torchrun_world2_cpu():
    """Exactly the driver's launch: torch.distributed.run, nproc 2, gloo."""
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29531", "bench.py", "--gpus", "2",
         "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=560, cwd=REPO, env=env)
    assert r.returncode == 0, r.stderr[-2000:]
    res = _last_json_line(r.stdout)
    assert res["n_gpus"] == 2
    assert res["config"]["parallelism"] == "dp2"
    # whole-job value: global batch = 2x per-rank batch
    assert res["config"]["global_batch"] == 16  # 8 per rank on CPU


@pytest.mark.slow
@pytest.mark.timeout(600)
def test_classification_cli_torchrun_world2(tmp_path):
    """Full classification training CLI under torchrun world=2 (gloo):
    DDP wrap, DistributedSampler, rank-0 checkpointing."""
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29533",
         "projects/classification/mnist/train.py", "--epochs", "1",
         "--batch-size", "8", "--synthetic-size", "32", "--workers", "0",
         "--device", "cpu", "--output", str(tmp_path)],
        capture_output=True, text=True, timeout=560, cwd=REPO, env=env)
    assert r.returncode == 0, (r.stdout[-1000:], r.stderr[-2000:])
    weights = tmp_path / "mnist" / "weights"
    assert (weights / "model_0.pth").exists()
    assert (weights / "best_model.pth").exists()
