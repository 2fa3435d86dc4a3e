"""Multinomial (softmax) logistic regression — CPU tier."""

import math

import numpy as np
import pytest
import torch

from sparkagd_amd import (
    MultinomialLogisticGradient,
    SimpleUpdater,
    SquaredL2Updater,
    generate_multiclass_problem,
    run,
    run_mini_batch,
)
from sparkagd_amd.ops import multiclass as mc


def _np_softmax_eval(A, y, Wflat, K):
    n, d = A.shape
    W = Wflat.reshape(d, K)
    Z = A @ W
    Zs = Z - Z.max(axis=1, keepdims=True)
    P = np.exp(Zs) / np.exp(Zs).sum(axis=1, keepdims=True)
    lse = np.log(np.exp(Zs).sum(axis=1)) + Z.max(axis=1)
    loss = (lse - Z[np.arange(n), y.astype(int)]).sum()
    M = P.copy()
    M[np.arange(n), y.astype(int)] -= 1.0
    return (A.T @ M).reshape(-1), loss


def test_oracle_matches_numpy():
    rng = np.random.default_rng(0)
    n, d, K = 200, 12, 5
    A = rng.normal(size=(n, d))
    y = rng.integers(0, K, n).astype(np.float64)
    W = rng.normal(size=d * K)
    g_np, l_np = _np_softmax_eval(A, y, W, K)
    grad, lc = mc.ref_eval_multi(torch.from_numpy(A), torch.from_numpy(y),
                                 torch.from_numpy(W), K)
    np.testing.assert_allclose(grad.numpy(), g_np, rtol=1e-9)
    assert abs(float(lc[0]) - l_np) < 1e-8 * max(1.0, abs(l_np))
    assert float(lc[1]) == n


def test_oracle_finite_difference():
    """grad matches a central finite difference of the loss."""
    rng = np.random.default_rng(1)
    n, d, K = 40, 5, 3
    A = torch.from_numpy(rng.normal(size=(n, d)))
    y = torch.from_numpy(rng.integers(0, K, n).astype(np.float64))
    W = torch.from_numpy(rng.normal(size=d * K))
    grad, _ = mc.ref_eval_multi(A, y, W, K)
    eps = 1e-6
    for idx in [0, 7, d * K - 1]:
        Wp, Wm = W.clone(), W.clone()
        Wp[idx] += eps
        Wm[idx] -= eps
        _, lp = mc.ref_eval_multi(A, y, Wp, K, need_grad=False)
        _, lm = mc.ref_eval_multi(A, y, Wm, K, need_grad=False)
        fd = (float(lp[0]) - float(lm[0])) / (2 * eps)
        assert abs(fd - float(grad[idx])) < 1e-4 * max(1.0, abs(fd))


def test_padded_margins_roundtrip():
    """K not a multiple of 4 exercises the class padding."""
    rng = np.random.default_rng(2)
    n, d, K = 64, 10, 6  # KC = 8
    A = torch.from_numpy(rng.normal(size=(n, d)))
    y = torch.from_numpy(rng.integers(0, K, n).astype(np.float64))
    W = torch.from_numpy(rng.normal(size=d * K))
    zf = mc.margins_multi(A, W, K)
    assert zf.numel() == n * mc.padded_k(K)
    g1, lc1 = mc.eval_multi_from_margins(A, zf, y, K)
    g2, lc2 = mc.ref_eval_multi(A, y, W, K)
    torch.testing.assert_close(g1, g2, rtol=1e-9, atol=1e-10)
    torch.testing.assert_close(lc1, lc2)


def test_agd_multiclass_converges():
    shard, w_true = generate_multiclass_problem(3000, 20, 4, seed=3,
                                                dtype=torch.float64,
                                                label_noise=0.15)
    K = 4
    grad = MultinomialLogisticGradient(K)
    w0 = torch.zeros(20 * K, dtype=torch.float64)
    w, h = run(shard, grad, SquaredL2Updater(), 1e-10, 60, 0.001, w0,
               1.0, math.inf, 0.5, 0.9, True)
    assert h[-1] < 0.65 * h[0]  # log(4) start; noise floor bounds the drop
    # accuracy of the fitted model
    Z = shard.features @ w.reshape(20, K)
    acc = float((Z.argmax(dim=1).to(torch.float32) == shard.labels).float().mean())
    assert acc > 0.8
    # margin tracking matches the untracked path (trajectory identity)
    w_u, h_u = run(shard, grad, SquaredL2Updater(), 1e-10, 60, 0.001, w0,
                   1.0, math.inf, 0.5, 0.9, True, track_margins=False)
    assert len(h) == len(h_u)
    for a, b in zip(h, h_u):
        assert abs(a - b) < 1e-8 * max(1.0, abs(b))


def test_minibatch_multiclass():
    shard, _ = generate_multiclass_problem(2000, 10, 3, seed=5, dtype=torch.float64)
    grad = MultinomialLogisticGradient(3)
    w0 = torch.zeros(30, dtype=torch.float64)
    w, h = run_mini_batch(shard, grad, SimpleUpdater(), 1.0, 15, 0.0, 0.5, w0)
    assert h[-1] < h[0]


def test_softmax_trainer_and_model():
    from sparkagd_amd import SoftmaxRegressionWithAGD

    shard, _ = generate_multiclass_problem(3000, 20, 4, seed=9,
                                           dtype=torch.float64,
                                           label_noise=0.1)
    model = SoftmaxRegressionWithAGD.train(shard, num_classes=4,
                                           num_iterations=60, reg_param=0.001)
    assert model.loss_history[-1] < model.loss_history[0]
    pred = model.predict(shard.features)
    acc = float((pred == shard.labels).float().mean())
    assert acc > 0.8
    proba = model.predict_proba(shard.features)
    assert proba.shape == (3000, 4)
    torch.testing.assert_close(proba.sum(dim=1),
                               torch.ones(3000, dtype=proba.dtype))


def test_csr_multiclass_matches_dense_oracle():
    """CSR multiclass margins/grad equal the dense oracle on the densified
    matrix (CPU tier)."""
    from sparkagd_amd.data import generate_multiclass_csr_problem

    shard, _ = generate_multiclass_csr_problem(300, 40, 12, num_classes=5, seed=11)
    K = 5
    dense = torch.zeros((shard.n, shard.d), dtype=torch.float64)
    rows = torch.repeat_interleave(
        torch.arange(shard.n), torch.diff(shard.rowptr.to(torch.int64)))
    dense.index_put_((rows, shard.col.to(torch.int64)),
                     shard.val.to(torch.float64), accumulate=True)
    W = torch.randn(shard.d * K, dtype=torch.float64,
                    generator=torch.Generator().manual_seed(3))
    grad_d, lc_d = mc.ref_eval_multi(dense, shard.labels.to(torch.float64), W, K)

    grad = MultinomialLogisticGradient(K)
    # CSR path computes in f32 (val dtype); compare at f32-appropriate tolerance
    g_c, lc_c = grad.eval(shard, W)
    torch.testing.assert_close(g_c.to(torch.float64), grad_d, rtol=2e-5, atol=2e-5)
    torch.testing.assert_close(lc_c, lc_d, rtol=1e-5, atol=1e-5)
    # loss-only and from-margins agree
    zf = grad.margins(shard, W)
    g2, lc2 = grad.eval_from_margins(shard, zf)
    torch.testing.assert_close(g2, g_c)
    torch.testing.assert_close(lc2, lc_c)


def test_csr_multiclass_agd_converges():
    from sparkagd_amd.data import generate_multiclass_csr_problem

    K = 4
    shard, _ = generate_multiclass_csr_problem(4000, 60, 10, num_classes=K,
                                               seed=13, label_noise=0.1)
    grad = MultinomialLogisticGradient(K)
    w0 = torch.zeros(60 * K, dtype=torch.float32)
    w, h = run(shard, grad, SquaredL2Updater(), 1e-10, 50, 0.001, w0,
               1.0, math.inf, 0.5, 0.9, True)
    assert h[-1] < 0.6 * h[0]
    # tracked and untracked trajectories agree
    w_u, h_u = run(shard, grad, SquaredL2Updater(), 1e-10, 50, 0.001, w0,
                   1.0, math.inf, 0.5, 0.9, True, track_margins=False)
    assert len(h) == len(h_u)
    for a, b in zip(h, h_u):
        assert abs(a - b) < 1e-6 * max(1.0, abs(b))


def test_multiclass_gram_matches_direct():
    """The dual-space (Gram) solver on a multiclass problem follows the
    direct solver's trajectory (padded class columns through the
    coefficient-space machinery)."""
    K = 5
    shard, _ = generate_multiclass_problem(400, 24, K, seed=17,
                                           dtype=torch.float64,
                                           label_noise=0.2)
    grad = MultinomialLogisticGradient(K)
    w0 = torch.zeros(24 * K, dtype=torch.float64)
    args = (grad, SquaredL2Updater(), 1e-12, 25, 0.01, w0,
            1.0, math.inf, 0.5, 0.9, True)
    w_d, h_d = run(shard, *args, loss_history_mode="backtrack")
    w_g, h_g = run(shard, *args, loss_history_mode="backtrack", solver="gram")
    assert len(h_d) == len(h_g)
    for a, b in zip(h_d, h_g):
        assert abs(a - b) < 1e-8 * max(1.0, abs(b)), (a, b)
    torch.testing.assert_close(w_g, w_d, rtol=1e-6, atol=1e-9)
    # nonzero initial weights exercise the x0 margin basis row
    w1 = torch.randn(24 * K, dtype=torch.float64,
                     generator=torch.Generator().manual_seed(18)) * 0.05
    w_d2, h_d2 = run(shard, grad, SquaredL2Updater(), 1e-12, 10, 0.01, w1,
                     1.0, math.inf, 0.5, 0.9, True)
    w_g2, h_g2 = run(shard, grad, SquaredL2Updater(), 1e-12, 10, 0.01, w1,
                     1.0, math.inf, 0.5, 0.9, True, solver="gram")
    for a, b in zip(h_d2, h_g2):
        assert abs(a - b) < 1e-8 * max(1.0, abs(b))
    torch.testing.assert_close(w_g2, w_d2, rtol=1e-6, atol=1e-9)


def test_multiclass_guards():
    with pytest.raises(ValueError):
        MultinomialLogisticGradient(1)


def test_multiclass_regularization_path():
    """regularization_path with a multiclass gradient returns
    MultinomialModels and reuses the Gram operator across lambdas."""
    from sparkagd_amd import MultinomialModel, regularization_path

    shard, _ = generate_multiclass_problem(600, 15, 4, seed=19,
                                           dtype=torch.float64,
                                           label_noise=0.2)
    models = regularization_path(shard, [0.1, 0.01, 0.001],
                                 gradient=MultinomialLogisticGradient(4),
                                 num_iterations=25)
    assert len(models) == 3
    assert all(isinstance(m, MultinomialModel) for m in models)
    acc = float((models[-1].predict(shard.features) == shard.labels)
                .float().mean())
    assert acc > 0.7
