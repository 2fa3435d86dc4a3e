"""The examples are living documentation — run each at a small size so they
cannot rot, and check the public API surface re-exports cleanly."""

import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(script, *args):
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "examples", script), *args],
        capture_output=True, text=True, timeout=300, cwd=REPO)
    assert out.returncode == 0, f"{script}: {out.stderr[-2000:]}"
    return out.stdout


@pytest.mark.parametrize("script,args", [
    ("train_logistic.py", ["--n", "5000", "--d", "64", "--iters", "10"]),
    ("train_multiclass.py", ["--n", "3000", "--d", "32", "--classes", "4",
                             "--iters", "8"]),
    ("train_sparse_svm.py", ["--n", "5000", "--d", "2000",
                             "--nnz-per-row", "8", "--iters", "10"]),
    ("train_mixed.py", ["--n-dense", "1500", "--n-sparse", "2500",
                        "--d", "300", "--iters", "8"]),
    ("train_save_serve.py", ["--n", "8000", "--d", "96", "--iters", "12"]),
])
def test_example_runs(script, args):
    _run(script, *args)


def test_public_api_exports_resolve():
    import sparkagd_amd as m

    assert m.__version__
    missing = [n for n in m.__all__ if not hasattr(m, n)]
    assert not missing, missing


def test_iters_to_eps_script_runs():
    """The iteration-advantage script (the metric's second half) stays
    runnable; on CPU it uses its small fallback problem."""
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "benchmarks", "iters_to_eps.py")],
        capture_output=True, text=True, timeout=300, cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    assert "AGD" in out.stdout and "L*" in out.stdout
