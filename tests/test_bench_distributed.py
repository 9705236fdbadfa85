"""bench.py driver-contract regression: the torchrun 2-rank CPU path must
produce one valid JSON line (rank 0) with whole-job aggregation. This is
the exact launch shape the driver uses for multi-GPU scaling runs."""
import json
import os
import subprocess
import sys

import pytest


@pytest.mark.timeout(300)
def test_bench_two_rank_gloo(tmp_path):
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", "--nproc-per-node", "2",
        "--master-addr", "127.0.0.1", "--master-port", "29781",
        os.path.join(root, "bench.py"),
        "--gpus", "2", "--steps", "3", "--warmup", "1",
        "--device", "cpu", "--backend", "gloo",
        "--actor-batch-size", "8", "--num-actor-batches", "2",
        "--num-actor-cpus", "2", "--batch-size", "4",
        "--virtual-batch-size", "8",
    ]
    out = subprocess.run(
        cmd, cwd=str(tmp_path), capture_output=True, text=True, timeout=280,
        env={**os.environ, "PYTHONPATH": root},
    )
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout  # exactly one JSON line (rank 0 only)
    j = json.loads(lines[0])
    assert j["n_gpus"] == 2 and j["steps"] == 3
    assert j["scaling"] == "weak" and j["value"] > 0
    assert j["config"]["parallelism"] == "dp2"


@pytest.mark.timeout(180)
def test_bench_r2d2_config_json(tmp_path):
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, os.path.join(root, "bench.py"), "--config", "r2d2", "--steps", "8"],
        cwd=str(tmp_path), capture_output=True, text=True, timeout=170,
        env={**os.environ, "PYTHONPATH": root},
    )
    assert out.returncode == 0, out.stdout[-1500:] + out.stderr[-1500:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout
    j = json.loads(lines[0])
    assert "replay" in j["metric"] and j["unit"] == "sequences/s"
    assert j["config"]["capacity_seqs"] == 2048
