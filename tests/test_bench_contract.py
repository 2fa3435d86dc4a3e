"""The driver contract on bench.py: one JSON line with the mandated fields
(the driver parses this at round end — BENCH/SCALE records). Runs the CPU
smoke fallback; the GPU path emits the identical schema."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run_bench(*extra):
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"),
         "--steps", "3", "--warmup", "1", *extra],
        capture_output=True, text=True, timeout=300, cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    return json.loads(line)


def test_bench_json_contract():
    d = _run_bench()
    # the driver-mandated fields, with the mandated types/semantics
    assert d["metric"].startswith("examples/sec + iters-to-")
    assert isinstance(d["value"], float) and d["value"] > 0
    assert d["unit"] == "examples/s"
    assert d["n_gpus"] == 1
    assert d["steps"] == 3 and d["warmup"] == 1
    assert d["ms_per_step"] > 0
    assert d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["data"] == "synthetic"
    assert isinstance(d["dtype"], str)
    assert d["rows_per_sec"] > 0
    c = d["config"]
    for key in ("model", "d", "rows_per_gpu", "global_rows", "parallelism",
                "solver", "reg_param", "evals_per_step",
                "data_passes_per_step", "examples_definition",
                "iters_to_eps", "eps", "loss_star", "shard_gb"):
        assert key in c, key
    assert c["parallelism"] == "dp1"
    # the metric's second half must be non-null and meaningful by default
    assert isinstance(c["iters_to_eps"], int) and c["iters_to_eps"] >= 1
    assert c["loss_star"] is not None
    # eval-examples definition: value = rows * evals/step * steps / elapsed
    ratio = d["value"] / d["rows_per_sec"]
    assert abs(ratio - c["evals_per_step"]) < 1e-6


def test_bench_gram_contract():
    d = _run_bench("--solver", "gram", "--eps-iters", "0")
    assert d["config"]["solver"] == "gram"
    assert d["config"]["gram_build_seconds"] is not None
    assert d["value"] > 0  # fused paths must keep the eval accounting alive


def test_bench_csr_contract():
    d = _run_bench("--csr", "--eps-iters", "0")
    assert d["config"]["model"].startswith("csr_")
    assert d["config"]["csr_dist"] == "uniform"
    assert d["value"] > 0
