"""bench.py driver-contract tests: JSON line schema, and the exact
torchrun multi-rank launch the round-end driver uses (CPU/gloo here;
RCCL on the GPU box)."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED_KEYS = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
                 "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                 "dtype", "data", "config"}


def _parse_last_json(out: str):
    lines = [l for l in out.strip().splitlines() if l.startswith("{")]
    assert lines, f"no JSON line in output:\n{out[-2000:]}"
    return json.loads(lines[-1])


@pytest.mark.timeout(300)
def test_bench_single_process_cpu():
    r = subprocess.run(
        [sys.executable, "bench.py", "--steps", "4", "--warmup", "1",
         "--n_stocks", "24", "--seq_len", "5", "--num_factor", "4",
         "--hidden_size", "16", "--num_portfolio", "8", "--n_days", "2",
         "--engine", "eager"],
        cwd=REPO, capture_output=True, text=True, timeout=240)
    assert r.returncode == 0, r.stderr[-2000:]
    d = _parse_last_json(r.stdout)
    assert REQUIRED_KEYS.issubset(d.keys())
    assert d["n_gpus"] == 1
    assert d["metric"] == "training cross-sections/sec"
    assert d["value"] > 0 and d["higher_is_better"] is True
    assert d["config"]["parallelism"] == "dp1"


@pytest.mark.timeout(600)
def test_bench_torchrun_two_ranks_gloo():
    """Exactly the driver's launch shape (nnodes=1, nproc-per-node N,
    master-addr 127.0.0.1), CPU/gloo standing in for RCCL."""
    env = dict(os.environ)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29571", "bench.py", "--gpus", "2",
         "--steps", "4", "--warmup", "1", "--n_stocks", "24",
         "--seq_len", "5", "--num_factor", "4", "--hidden_size", "16",
         "--num_portfolio", "8", "--n_days", "2", "--engine", "eager"],
        cwd=REPO, capture_output=True, text=True, timeout=540, env=env)
    assert r.returncode == 0, (r.stdout[-1500:], r.stderr[-2000:])
    d = _parse_last_json(r.stdout)
    assert d["n_gpus"] == 2
    assert d["config"]["parallelism"] == "dp2"
    assert d["config"]["global_batch"] == 48
    assert d["scaling"] == "weak"


@pytest.mark.timeout(300)
def test_bench_forced_dist_world_size_1():
    """FV_FORCE_DIST=1: a single-process run initializes the real
    collective backend at world_size=1 (gloo on CPU; RCCL on a GPU box —
    the hardware smoke for the DP path) and still reports n_gpus=1."""
    env = dict(os.environ)
    env.update({"FV_FORCE_DIST": "1", "MASTER_ADDR": "127.0.0.1",
                "MASTER_PORT": "29572", "WORLD_SIZE": "1", "RANK": "0",
                "LOCAL_RANK": "0"})
    r = subprocess.run(
        [sys.executable, "bench.py", "--steps", "4", "--warmup", "1",
         "--n_stocks", "24", "--seq_len", "5", "--num_factor", "4",
         "--hidden_size", "16", "--num_portfolio", "8", "--n_days", "2",
         "--engine", "eager"],
        cwd=REPO, capture_output=True, text=True, timeout=240, env=env)
    assert r.returncode == 0, (r.stdout[-1500:], r.stderr[-2000:])
    d = _parse_last_json(r.stdout)
    assert d["n_gpus"] == 1
    assert d["config"]["parallelism"] == "dp1"


@pytest.mark.timeout(600)
def test_bench_torchrun_four_ranks_gloo():
    """4-rank ladder rung of the driver's SCALE launch (CPU/gloo)."""
    env = dict(os.environ)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "4", "--master-addr", "127.0.0.1",
         "--master-port", "29573", "bench.py", "--gpus", "4",
         "--steps", "4", "--warmup", "1", "--n_stocks", "16",
         "--seq_len", "4", "--num_factor", "4", "--hidden_size", "16",
         "--num_portfolio", "8", "--n_days", "2", "--engine", "eager"],
        cwd=REPO, capture_output=True, text=True, timeout=540, env=env)
    assert r.returncode == 0, (r.stdout[-1500:], r.stderr[-2000:])
    d = _parse_last_json(r.stdout)
    assert d["n_gpus"] == 4
    assert d["config"]["parallelism"] == "dp4"
    assert d["config"]["global_batch"] == 64


@pytest.mark.timeout(600)
def test_bench_torchrun_eight_ranks_gloo():
    """Full-node rung of the driver's SCALE launch (nproc-per-node 8,
    CPU/gloo): the exact width the round-end 8×MI355X run uses, incl.
    n_days(2) < world_size(8) — wrap-padding must keep all 8 ranks in
    lockstep."""
    env = dict(os.environ)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "8", "--master-addr", "127.0.0.1",
         "--master-port", "29574", "bench.py", "--gpus", "8",
         "--steps", "3", "--warmup", "1", "--n_stocks", "12",
         "--seq_len", "4", "--num_factor", "3", "--hidden_size", "16",
         "--num_portfolio", "6", "--n_days", "2", "--engine", "eager"],
        cwd=REPO, capture_output=True, text=True, timeout=540, env=env)
    assert r.returncode == 0, (r.stdout[-1500:], r.stderr[-2000:])
    d = _parse_last_json(r.stdout)
    assert d["n_gpus"] == 8
    assert d["config"]["parallelism"] == "dp8"
    assert d["config"]["global_batch"] == 96
