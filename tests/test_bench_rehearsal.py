"""torchrun rehearsal of bench.py's distributed path (VERDICT round-1
item 2): world=2 over gloo on CPU exercises the SAME rendezvous /
barrier / MAX-all-reduce code the driver's 8-GPU SCALE run uses, just
with backend=gloo instead of nccl(=RCCL). Do NOT wait for 8-GPU
hardware to find wire bugs."""
import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(600)
def test_bench_torchrun_world2_cpu():
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29517", os.path.join(ROOT, "bench.py"),
         "--gpus", "2", "--steps", "3", "--warmup", "1", "--batch", "4",
         "--model", "lenet", "--device", "cpu"],
        capture_output=True, text=True, timeout=540, env=env, cwd=ROOT)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [ln for ln in out.stdout.splitlines()
             if ln.startswith("{") and '"metric"' in ln]
    assert len(lines) == 1, out.stdout  # exactly ONE JSON line (rank 0)
    rec = json.loads(lines[0])
    assert rec["n_gpus"] == 2
    assert rec["steps"] == 3
    assert rec["config"]["parallelism"] == "dp2"
    assert rec["config"]["global_batch"] == 8
    assert rec["value"] > 0


@pytest.mark.timeout(300)
def test_bench_single_process_cpu():
    out = subprocess.run(
        [sys.executable, os.path.join(ROOT, "bench.py"), "--steps", "3",
         "--warmup", "1", "--batch", "4", "--model", "lenet",
         "--device", "cpu"],
        capture_output=True, text=True, timeout=240, cwd=ROOT)
    assert out.returncode == 0, out.stderr[-2000:]
    rec = json.loads(out.stdout.strip().splitlines()[-1])
    assert rec["metric"] == "images_per_sec"
    assert rec["n_gpus"] == 1


@pytest.mark.timeout(900)
def test_bench_torchrun_world8_cpu():
    """Full-width rehearsal of the driver's 8-GPU SCALE launch line
    (gloo instead of nccl): one JSON line, dp8 config."""
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "8", "--master-addr", "127.0.0.1",
         "--master-port", "29533", os.path.join(ROOT, "bench.py"),
         "--gpus", "8", "--steps", "2", "--warmup", "1", "--batch", "4",
         "--model", "lenet", "--device", "cpu"],
        capture_output=True, text=True, timeout=840, env=env, cwd=ROOT)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [ln for ln in out.stdout.splitlines()
             if ln.startswith("{") and '"metric"' in ln]
    assert len(lines) == 1, out.stdout
    rec = json.loads(lines[0])
    assert rec["n_gpus"] == 8
    assert rec["config"]["parallelism"] == "dp8"
