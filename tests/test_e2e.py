"""End-to-end slices: CSR optimizer runs, the example scripts, and the
bench.py contract (hook bracketing, JSON line) on CPU."""

import json
import math
import subprocess
import sys
import os

import torch

from sparkagd_amd import (
    LogisticGradient,
    SimpleUpdater,
    run,
    run_mini_batch,
)
from sparkagd_amd.data import generate_csr_problem
from sparkagd_amd import ops

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_agd_on_csr_shard_cpu():
    shard, _ = generate_csr_problem(3000, 500, 12, seed=6)
    w0 = torch.zeros(500, dtype=torch.float32)
    w, hist = run(shard, LogisticGradient(), SimpleUpdater(), 1e-10, 15, 0.0,
                  w0, 1.0, math.inf, 0.5, 0.9, True)
    assert hist[-1] < hist[0]
    assert hist[-1] < math.log(2.0)  # beats the all-zeros predictor


def test_minibatch_on_csr_shard_cpu():
    shard, _ = generate_csr_problem(3000, 500, 12, seed=6)
    w0 = torch.zeros(500, dtype=torch.float32)
    w, hist = run_mini_batch(shard, LogisticGradient(), SimpleUpdater(), 1.0,
                             15, 0.0, 0.5, w0)
    assert hist[-1] < hist[0]


def test_bench_contract_cpu():
    """bench.py with no GPU prints exactly one valid JSON line with the
    driver-contract fields."""
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=300, cwd=REPO,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1
    j = json.loads(lines[0])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in j, key
    assert j["steps"] == 2 and j["warmup"] == 1 and j["n_gpus"] == 1
    assert j["scaling"] == "weak" and j["data"] == "synthetic"
    assert j["value"] > 0 and j["ms_per_step"] > 0
    assert j["config"]["evals_per_step"] >= 2.0


def test_example_scripts_cpu():
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "examples", "train_logistic.py"),
         "--n", "2000", "--d", "32", "--iters", "8"],
        capture_output=True, text=True, timeout=300, cwd=REPO,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    assert "loss:" in out.stdout
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "examples", "train_sparse_svm.py"),
         "--n", "2000", "--d", "300", "--nnz-per-row", "8", "--iters", "6"],
        capture_output=True, text=True, timeout=300, cwd=REPO,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    assert "AGD" in out.stdout
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "examples", "train_multiclass.py")],
        capture_output=True, text=True, timeout=300, cwd=REPO,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    assert "save/load round-trip: ok" in out.stdout


def test_baseline_config1_plumbing():
    """BASELINE.json config #1 at its exact shape: dense logistic n=1k,
    d=100, CPU — the reference's `local[2]` plumbing tier. AGD reaches GD's
    loss with far fewer iterations (the suite's iteration-advantage
    contract, Suite.scala:60-90)."""
    import math

    import torch

    from sparkagd_amd import (LogisticGradient, SimpleUpdater, run,
                              run_mini_batch)
    from sparkagd_amd.data import generate_dense_problem

    shard, _ = generate_dense_problem(1000, 100, seed=12, dtype=torch.float64)
    w0 = torch.zeros(100, dtype=torch.float64)
    w_agd, h_agd = run(shard, LogisticGradient(), SimpleUpdater(), 1e-12, 10,
                       0.0, w0, 1.0, math.inf, 0.5, 0.9, True)
    w_gd, h_gd = run_mini_batch(shard, LogisticGradient(), SimpleUpdater(),
                                1.0, 50, 0.0, 1.0, w0)
    assert len(h_agd) == 10 and len(h_gd) == 50
    # AGD(10) at or below GD(50) within the suite's 2% relTol
    assert h_agd[-1] <= h_gd[-1] * 1.02, (h_agd[-1], h_gd[-1])
