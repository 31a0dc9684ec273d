"""bench.py contract: the driver launches it via torch.distributed.run —
verify the exact multi-process invocation end-to-end on CPU (gloo)."""

import json
import os
import socket
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def test_bench_single_process_cpu(tmp_path):
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--cpu",
         "--steps", "1", "--warmup", "0", "--batch", "64", "--hidden", "32",
         "--train-steps-per-iter", "3", "--eval-batches", "2"],
        capture_output=True, text=True, timeout=600, cwd=str(tmp_path))
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    assert d["metric"] == "adanet_iterations_per_hour"
    assert d["n_gpus"] == 1 and d["steps"] == 1
    assert d["value"] > 0 and d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["dtype"] == "fp32"  # --cpu debug mode; bf16 on GPU
    assert "final_ensemble_accuracy" in d["config"]


def test_bench_torchrun_two_ranks_cpu(tmp_path):
    """The driver's exact launch shape at N=2 (gloo stands in for RCCL)."""
    port = _free_port()
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port), os.path.join(REPO, "bench.py"),
         "--gpus", "2", "--cpu", "--steps", "1", "--warmup", "0",
         "--batch", "64", "--hidden", "32", "--train-steps-per-iter", "3",
         "--eval-batches", "2"],
        capture_output=True, text=True, timeout=900, cwd=str(tmp_path))
    assert out.returncode == 0, (out.stdout[-1000:], out.stderr[-2000:])
    # exactly one JSON line, from rank 0
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout[-2000:]
    d = json.loads(lines[0])
    assert d["n_gpus"] == 2
    assert d["config"]["candidates_per_iter"] == 4  # 2 per rank (weak)
    assert "round_robin2" in d["config"]["parallelism"]
    assert d["value"] > 0


def test_bench_torchrun_replication_cpu(tmp_path):
    """DP placement at N=2 over gloo: flat-bucket all-reduce path."""
    port = _free_port()
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port), os.path.join(REPO, "bench.py"),
         "--gpus", "2", "--cpu", "--steps", "1", "--warmup", "0",
         "--batch", "64", "--hidden", "32", "--train-steps-per-iter", "3",
         "--eval-batches", "2", "--placement", "replication"],
        capture_output=True, text=True, timeout=900, cwd=str(tmp_path))
    assert out.returncode == 0, (out.stdout[-1000:], out.stderr[-2000:])
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1
    d = json.loads(lines[0])
    assert d["config"]["parallelism"] == "dp2"
    assert d["value"] > 0


def test_bench_self_launches_torchrun(tmp_path):
    """A plain `python bench.py --gpus 2` (no torchrun env) must self-launch
    under torch.distributed.run and report n_gpus=2 — never silently measure
    one rank."""
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--gpus", "2",
         "--cpu", "--steps", "1", "--warmup", "0", "--batch", "64",
         "--hidden", "32", "--train-steps-per-iter", "3",
         "--eval-batches", "2"],
        capture_output=True, text=True, timeout=900, cwd=str(tmp_path))
    assert out.returncode == 0, (out.stdout[-1000:], out.stderr[-2000:])
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout[-2000:]
    assert json.loads(lines[0])["n_gpus"] == 2


def test_bench_world_size_mismatch_fails(tmp_path):
    """Under a torchrun env whose world size contradicts --gpus, bench.py
    must fail instead of reporting a wrong n_gpus."""
    port = _free_port()
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port), os.path.join(REPO, "bench.py"),
         "--gpus", "4", "--cpu", "--steps", "1", "--warmup", "0",
         "--batch", "64", "--hidden", "32", "--train-steps-per-iter", "3",
         "--eval-batches", "2"],
        capture_output=True, text=True, timeout=900, cwd=str(tmp_path))
    assert out.returncode != 0
