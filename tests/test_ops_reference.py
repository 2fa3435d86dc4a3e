"""Unit tests of the reference (oracle) ops against hand-written NumPy math.

These pin the MLlib-1.3 loss/updater semantics so the
HIP kernels can later be validated against `ops.reference` transitively.
"""

import math

import numpy as np
import pytest
import torch

from sparkagd_amd import ops
from sparkagd_amd.models.gradient import (
    HingeGradient,
    LeastSquaresGradient,
    LogisticGradient,
)
from sparkagd_amd.models.updater import L1Updater, SimpleUpdater, SquaredL2Updater
from sparkagd_amd.data import DenseShard, CSRShard, generate_csr_problem


def _np_eval(A, y, w, loss_type):
    z = A @ w
    if loss_type == ops.LOSS_LOGISTIC:
        mult = 1.0 / (1.0 + np.exp(-z)) - y
        loss = np.where(y > 0, np.log1p(np.exp(-np.abs(z))) + np.maximum(-z, 0),
                        np.log1p(np.exp(-np.abs(z))) + np.maximum(z, 0))
    elif loss_type == ops.LOSS_LEAST_SQUARES:
        mult = 2.0 * (z - y)
        loss = (z - y) ** 2
    elif loss_type == ops.LOSS_HINGE:
        s = 2.0 * y - 1.0
        viol = s * z < 1.0
        mult = np.where(viol, -s, 0.0)
        loss = np.maximum(0.0, 1.0 - s * z)
    else:  # smoothed hinge
        s = 2.0 * y - 1.0
        sz = s * z
        mult = np.where(sz >= 1.0, 0.0, np.where(sz > 0.0, -s * (1.0 - sz), -s))
        loss = np.where(sz >= 1.0, 0.0, np.where(sz > 0.0, 0.5 * (1.0 - sz) ** 2, 0.5 - sz))
    return A.T @ mult, loss.sum()


@pytest.mark.parametrize("loss_type", [ops.LOSS_LOGISTIC, ops.LOSS_LEAST_SQUARES, ops.LOSS_HINGE, ops.LOSS_SMOOTH_HINGE])
def test_dense_eval_matches_numpy(loss_type):
    rng = np.random.default_rng(0)
    n, d = 257, 13
    A = rng.normal(size=(n, d))
    y = (rng.normal(size=n) > 0).astype(np.float64)
    if loss_type == ops.LOSS_LEAST_SQUARES:
        y = rng.normal(size=n)
    w = rng.normal(size=d)
    grad_np, loss_np = _np_eval(A, y, w, loss_type)

    grad, lc = ops.reference.dense_eval(
        torch.from_numpy(A), torch.from_numpy(y), torch.from_numpy(w), loss_type
    )
    np.testing.assert_allclose(grad.numpy(), grad_np, rtol=1e-10)
    assert abs(float(lc[0]) - loss_np) < 1e-8 * max(1.0, abs(loss_np))
    assert float(lc[1]) == n


@pytest.mark.parametrize("loss_type", [ops.LOSS_LOGISTIC, ops.LOSS_LEAST_SQUARES, ops.LOSS_HINGE])
def test_dense_eval_mask(loss_type):
    rng = np.random.default_rng(1)
    n, d = 100, 7
    A = rng.normal(size=(n, d))
    y = (rng.normal(size=n) > 0).astype(np.float64)
    w = rng.normal(size=d)
    mask = (rng.random(n) < 0.5)
    grad_np, loss_np = _np_eval(A[mask], y[mask], w, loss_type)
    grad, lc = ops.reference.dense_eval(
        torch.from_numpy(A), torch.from_numpy(y), torch.from_numpy(w), loss_type,
        mask=torch.from_numpy(mask.astype(np.uint8)),
    )
    np.testing.assert_allclose(grad.numpy(), grad_np, rtol=1e-10)
    assert abs(float(lc[0]) - loss_np) < 1e-8 * max(1.0, abs(loss_np))
    assert float(lc[1]) == mask.sum()


def test_csr_eval_matches_dense():
    shard, _ = generate_csr_problem(n=300, d=50, nnz_per_row=5, seed=3)
    w = torch.randn(50, dtype=torch.float32,
                    generator=torch.Generator().manual_seed(12))
    grad_c, lc_c = shard.eval(w, ops.LOSS_LOGISTIC)
    # densify
    A = torch.zeros(300, 50)
    for i in range(300):
        for k in range(int(shard.rowptr[i]), int(shard.rowptr[i + 1])):
            A[i, int(shard.col[k])] += float(shard.val[k])
    grad_d, lc_d = ops.reference.dense_eval(A, shard.labels, w, ops.LOSS_LOGISTIC)
    torch.testing.assert_close(grad_c, grad_d, rtol=1e-4, atol=1e-4)
    # both sides compute in f32 with different summation orders
    torch.testing.assert_close(lc_c, lc_d, rtol=1e-6, atol=1e-6)


def test_csc_build_matches_csr():
    """CSRShard's CSC copy re-expresses exactly the same matrix."""
    shard, _ = generate_csr_problem(n=200, d=60, nnz_per_row=7, seed=8)
    colptr, row, val = shard.csc
    A1 = torch.zeros(200, 60, dtype=torch.float64)
    for i in range(200):
        for k in range(int(shard.rowptr[i]), int(shard.rowptr[i + 1])):
            A1[i, int(shard.col[k])] += float(shard.val[k])
    A2 = torch.zeros(200, 60, dtype=torch.float64)
    for c in range(60):
        for k in range(int(colptr[c]), int(colptr[c + 1])):
            A2[int(row[k]), c] += float(val[k])
    torch.testing.assert_close(A1, A2)


def test_gradient_compute_per_example_parity():
    """The MLlib per-example Gradient.compute contract (AGD.scala:198)."""
    rng = np.random.default_rng(2)
    d = 5
    for g_cls, label in [(LogisticGradient, 1.0), (LeastSquaresGradient, 0.37), (HingeGradient, 0.0)]:
        g = g_cls()
        x = torch.from_numpy(rng.normal(size=d))
        w = torch.from_numpy(rng.normal(size=d))
        cum = torch.zeros(d, dtype=torch.float64)
        loss = g.compute(x, label, w, cum)
        grad_np, loss_np = _np_eval(x.numpy().reshape(1, -1), np.array([label]), w.numpy(), g.LOSS_TYPE)
        np.testing.assert_allclose(cum.numpy(), grad_np, rtol=1e-10)
        assert abs(loss - loss_np) < 1e-10 * max(1.0, abs(loss_np))


def test_updaters_mllib_semantics():
    rng = np.random.default_rng(4)
    d = 11
    w = rng.normal(size=d)
    g = rng.normal(size=d)
    wt, gt = torch.from_numpy(w), torch.from_numpy(g)

    # SimpleUpdater: w - (s/sqrt(iter)) g, reg 0
    w2, reg = SimpleUpdater().compute(wt, gt, 0.7, 4, 0.3)
    np.testing.assert_allclose(w2.numpy(), w - (0.7 / 2.0) * g, rtol=1e-12)
    assert float(reg) == 0.0

    # L1Updater: soft threshold, reg = lam*||w'||_1
    lam, s0, it = 0.3, 0.7, 9
    s = s0 / math.sqrt(it)
    w1 = w - s * g
    expected = np.sign(w1) * np.maximum(np.abs(w1) - lam * s, 0.0)
    w2, reg = L1Updater().compute(wt, gt, s0, it, lam)
    np.testing.assert_allclose(w2.numpy(), expected, rtol=1e-12)
    assert abs(float(reg) - lam * np.abs(expected).sum()) < 1e-10

    # SquaredL2Updater: w(1-s*lam) - s*g, reg = lam/2 ||w'||^2
    w2, reg = SquaredL2Updater().compute(wt, gt, s0, it, lam)
    expected = w * (1 - s * lam) - s * g
    np.testing.assert_allclose(w2.numpy(), expected, rtol=1e-12)
    assert abs(float(reg) - 0.5 * lam * (expected**2).sum()) < 1e-10


def test_sample_weight_semantics():
    """Per-example weights: grad/loss scale by w_i, count = sum of weights;
    integer weights equal example duplication."""
    from sparkagd_amd.data import DenseShard

    rng = np.random.default_rng(10)
    n, d = 50, 6
    A = rng.normal(size=(n, d))
    y = (rng.normal(size=n) > 0).astype(np.float64)
    w = rng.normal(size=d)
    sw = rng.integers(0, 4, n).astype(np.float32)

    sh = DenseShard(torch.from_numpy(A), torch.from_numpy(y),
                    sample_weight=torch.from_numpy(sw))
    grad_w, lc_w = sh.eval(torch.from_numpy(w), ops.LOSS_LOGISTIC)

    # duplicate rows per weight
    reps = sw.astype(int)
    A_dup = np.repeat(A, reps, axis=0)
    y_dup = np.repeat(y, reps)
    grad_np, loss_np = _np_eval(A_dup, y_dup, w, ops.LOSS_LOGISTIC)
    np.testing.assert_allclose(grad_w.numpy(), grad_np, rtol=1e-9)
    assert abs(float(lc_w[0]) - loss_np) < 1e-8 * max(1.0, abs(loss_np))
    assert float(lc_w[1]) == reps.sum()

    # weighted AGD runs end-to-end (with margin tracking) and matches the
    # duplicated-data run
    from sparkagd_amd import run, LogisticGradient, SquaredL2Updater
    w0 = torch.zeros(d, dtype=torch.float64)
    sh_dup = DenseShard(torch.from_numpy(A_dup), torch.from_numpy(y_dup))
    args = (LogisticGradient(), SquaredL2Updater(), 1e-12, 8, 0.1, w0,
            1.0, math.inf, 0.5, 0.9, True)
    w_a, h_a = __import__("sparkagd_amd").run(sh, *args)
    w_b, h_b = __import__("sparkagd_amd").run(sh_dup, *args)
    torch.testing.assert_close(w_a, w_b, rtol=1e-8, atol=1e-10)


def test_elastic_net_updater():
    from sparkagd_amd.models.updater import ElasticNetUpdater

    rng = np.random.default_rng(6)
    d = 13
    w = rng.normal(size=d)
    g = rng.normal(size=d)
    lam, ratio, s0, it = 0.4, 0.3, 0.7, 4
    s = s0 / math.sqrt(it)
    l1, l2 = ratio * lam, (1 - ratio) * lam
    w1 = w - s * g
    wsoft = np.sign(w1) * np.maximum(np.abs(w1) - l1 * s, 0.0)
    expected = wsoft / (1.0 + s * l2)
    w2, reg = ElasticNetUpdater(l1_ratio=ratio).compute(
        torch.from_numpy(w), torch.from_numpy(g), s0, it, lam)
    np.testing.assert_allclose(w2.numpy(), expected, rtol=1e-12)
    exp_reg = l1 * np.abs(expected).sum() + 0.5 * l2 * (expected**2).sum()
    assert abs(float(reg) - exp_reg) < 1e-10
    with pytest.raises(ValueError):
        ElasticNetUpdater(l1_ratio=1.5)


def test_add_intercept():
    from sparkagd_amd.data import add_intercept

    sh = DenseShard(torch.randn(9, 3, dtype=torch.float64), torch.zeros(9))
    sh2 = add_intercept(sh)
    assert sh2.d == 4
    assert torch.all(sh2.features[:, 0] == 1.0)
    torch.testing.assert_close(sh2.features[:, 1:], sh.features)


def test_fused_scalars():
    rng = np.random.default_rng(5)
    d = 33
    x, y, gy, xo = (rng.normal(size=d) for _ in range(4))
    out = ops.reference.fused_scalars(*(torch.from_numpy(v) for v in (x, y, gy, xo))).numpy()
    np.testing.assert_allclose(out[0], ((x - y) ** 2).sum(), rtol=1e-12)
    np.testing.assert_allclose(out[1], ((x - y) * gy).sum(), rtol=1e-12)
    np.testing.assert_allclose(out[2], (x * x).sum(), rtol=1e-12)
    np.testing.assert_allclose(out[3], ((x - xo) ** 2).sum(), rtol=1e-12)
    np.testing.assert_allclose(out[4], (gy * (x - xo)).sum(), rtol=1e-12)


def test_axpby():
    x = torch.randn(17, dtype=torch.float64)
    y = torch.randn(17, dtype=torch.float64)
    out = ops.reference.axpby(0.3, x, -1.7, y)
    torch.testing.assert_close(out, 0.3 * x - 1.7 * y)


def test_fp8_reference_eval():
    """float8_e4m3fn shards evaluate through the oracle (f32 accumulation)."""
    from sparkagd_amd.data import generate_dense_problem

    shard, _ = generate_dense_problem(128, 64, seed=5, dtype=torch.float8_e4m3fn)
    w = torch.randn(64, dtype=torch.float32) * 0.1
    grad, lc = shard.eval(w, ops.LOSS_LOGISTIC)
    g_ref, lc_ref = ops.reference.dense_eval(
        shard.features.to(torch.float32), shard.labels, w, ops.LOSS_LOGISTIC
    )
    torch.testing.assert_close(grad, g_ref, rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(lc, lc_ref)


def test_shard_properties():
    sh = DenseShard(torch.randn(10, 4), torch.zeros(10))
    assert sh.n == 10 and sh.d == 4 and sh.nbytes > 0
    c, _ = generate_csr_problem(20, 8, 3, seed=0)
    assert c.n == 20 and c.d == 8 and c.nnz == 60
