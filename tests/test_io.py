"""Shard ingestion/persistence tests (sparkagd_amd.io)."""

import numpy as np
import pytest
import torch

from sparkagd_amd import ops
from sparkagd_amd.data import generate_csr_problem, generate_dense_problem
from sparkagd_amd.io import (
    csr_from_scipy,
    dense_from_arrays,
    load_shard,
    save_shard,
)


def test_dense_from_arrays_sharded():
    X = np.random.default_rng(0).normal(size=(101, 7)).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.float32)
    s0 = dense_from_arrays(X, y, rank=0, world_size=2)
    s1 = dense_from_arrays(X, y, rank=1, world_size=2)
    assert s0.n + s1.n == 101 and abs(s0.n - s1.n) <= 1
    torch.testing.assert_close(
        torch.cat([s0.features, s1.features]), torch.from_numpy(X)
    )


def test_csr_from_scipy_matches_dense_eval():
    import scipy.sparse as sp

    rng = np.random.default_rng(1)
    X = sp.random(200, 50, density=0.1, format="csr", random_state=2, dtype=np.float64)
    y = rng.integers(0, 2, 200).astype(np.float64)
    shard = csr_from_scipy(X, y)
    w = torch.randn(50, dtype=torch.float32,
                    generator=torch.Generator().manual_seed(21))
    g_csr, lc = shard.eval(w, ops.LOSS_LOGISTIC)
    g_ref, lc_ref = ops.reference.dense_eval(
        torch.from_numpy(X.toarray()).to(torch.float32), torch.from_numpy(y).float(), w, ops.LOSS_LOGISTIC
    )
    torch.testing.assert_close(g_csr, g_ref, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(lc, lc_ref, rtol=1e-6, atol=1e-6)  # f32 paths, different sum order


def test_dense_shard_roundtrip(tmp_path):
    shard, _ = generate_dense_problem(64, 10, seed=3)
    p = str(tmp_path / "dense.safetensors")
    save_shard(p, shard)
    back = load_shard(p)
    assert torch.equal(back.features, shard.features)
    assert torch.equal(back.labels, shard.labels)


def test_csr_shard_roundtrip_sharded(tmp_path):
    shard, _ = generate_csr_problem(100, 40, 5, seed=4)
    p = str(tmp_path / "csr.safetensors")
    save_shard(p, shard)
    s0 = load_shard(p, rank=0, world_size=2)
    s1 = load_shard(p, rank=1, world_size=2)
    assert s0.n + s1.n == 100
    assert s0.nnz + s1.nnz == shard.nnz
    w = torch.randn(40, dtype=torch.float32,
                    generator=torch.Generator().manual_seed(22))
    g_full, lc_full = shard.eval(w, ops.LOSS_LEAST_SQUARES)
    g0, lc0 = s0.eval(w, ops.LOSS_LEAST_SQUARES)
    g1, lc1 = s1.eval(w, ops.LOSS_LEAST_SQUARES)
    torch.testing.assert_close(g0 + g1, g_full, rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(lc0 + lc1, lc_full, rtol=1e-6, atol=1e-6)


def test_svmlight_loader(tmp_path):
    pytest.importorskip("sklearn")
    p = str(tmp_path / "data.svm")
    with open(p, "w") as f:
        f.write("1 1:0.5 3:1.5\n0 2:2.0\n1 1:-1.0 2:0.25 3:0.75\n")
    shard = load_svmlight_helper(p)
    assert shard.n == 3 and shard.nnz == 6


def load_svmlight_helper(p):
    from sparkagd_amd.io import load_svmlight

    return load_svmlight(p)
