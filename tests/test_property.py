"""Property-based tests (hypothesis): the oracle ops hold their invariants
across random shapes, losses, masks and weights — the base the HIP kernels
are validated against transitively."""

import math

import numpy as np
import torch
from hypothesis import given, settings, strategies as st

from sparkagd_amd import ops


losses = st.sampled_from([ops.LOSS_LOGISTIC, ops.LOSS_LEAST_SQUARES,
                          ops.LOSS_HINGE, ops.LOSS_SMOOTH_HINGE])


@settings(max_examples=40, deadline=None)
@given(n=st.integers(1, 64), d=st.integers(1, 32), loss=losses,
       seed=st.integers(0, 2**31 - 1), use_mask=st.booleans(),
       use_weight=st.booleans())
def test_eval_invariants(n, d, loss, seed, use_mask, use_weight):
    rng = np.random.default_rng(seed)
    A = torch.from_numpy(rng.normal(size=(n, d)))
    y = torch.from_numpy((rng.normal(size=n) > 0).astype(np.float64))
    w = torch.from_numpy(rng.normal(size=d))
    mask = torch.from_numpy((rng.random(n) < 0.6).astype(np.uint8)) if use_mask else None
    sw = torch.from_numpy(rng.random(n).astype(np.float32) * 2) if use_weight else None

    grad, lc = ops.reference.dense_eval(A, y, w, loss, mask, True, sw)
    loss_sum, count = float(lc[0]), float(lc[1])

    # invariants
    assert math.isfinite(loss_sum) and loss_sum >= -1e-12 or loss == ops.LOSS_LEAST_SQUARES
    assert count >= 0
    assert torch.isfinite(grad).all()
    # loss-only evaluation returns the identical loss_count
    _, lc2 = ops.reference.dense_eval(A, y, w, loss, mask, False, sw)
    torch.testing.assert_close(lc, lc2)
    # gradient additivity over row partitions (the all-reduce identity)
    if n >= 2:
        k = n // 2
        g1, l1 = ops.reference.dense_eval(A[:k], y[:k], w, loss,
                                          None if mask is None else mask[:k],
                                          True, None if sw is None else sw[:k])
        g2, l2 = ops.reference.dense_eval(A[k:], y[k:], w, loss,
                                          None if mask is None else mask[k:],
                                          True, None if sw is None else sw[k:])
        torch.testing.assert_close(g1 + g2, grad, rtol=1e-9, atol=1e-9)
        torch.testing.assert_close(l1 + l2, lc, rtol=1e-9, atol=1e-9)
    # margins decomposition: eval == eval_from_margins(margins)
    m = ops.reference.dense_margins(A, w)
    g3, l3 = ops.reference.dense_eval_from_margins(A, m, y, loss, mask, True, sw)
    torch.testing.assert_close(g3, grad, rtol=1e-9, atol=1e-9)
    torch.testing.assert_close(l3, lc, rtol=1e-9, atol=1e-9)


@settings(max_examples=30, deadline=None)
@given(kind=st.sampled_from([ops.PROX_SIMPLE, ops.PROX_L1,
                             ops.PROX_SQUARED_L2, ops.PROX_ELASTIC_NET]),
       d=st.integers(1, 50), step=st.floats(0.0, 2.0),
       lam=st.floats(0.0, 1.0), seed=st.integers(0, 2**31 - 1))
def test_prox_invariants(kind, d, step, lam, seed):
    rng = np.random.default_rng(seed)
    w = torch.from_numpy(rng.normal(size=d))
    g = torch.from_numpy(rng.normal(size=d))
    out, reg = ops.reference.prox(kind, w, g, step, lam, lam2=0.2 * lam)
    assert torch.isfinite(out).all()
    assert float(reg) >= 0.0
    if step == 0.0 and kind in (ops.PROX_SIMPLE, ops.PROX_L1, ops.PROX_SQUARED_L2):
        torch.testing.assert_close(out, w)  # step 0 is the identity (reg-value trick)
    if kind in (ops.PROX_L1, ops.PROX_ELASTIC_NET) and lam > 0 and step > 0:
        # soft-thresholding never increases magnitude beyond the plain step
        plain = w - step * g
        assert bool((out.abs() <= plain.abs() + 1e-12).all())


@settings(max_examples=30, deadline=None)
@given(n=st.integers(1, 48), d=st.integers(1, 24), k=st.integers(2, 40),
       seed=st.integers(0, 2**31 - 1), use_mask=st.booleans(),
       use_weight=st.booleans())
def test_multiclass_invariants(n, d, k, seed, use_mask, use_weight):
    """Softmax oracle: additivity over rows, margins decomposition, loss
    positivity, and multiplier rows summing to zero-mass."""
    from sparkagd_amd.ops import multiclass as mc

    rng = np.random.default_rng(seed)
    A = torch.from_numpy(rng.normal(size=(n, d)))
    y = torch.from_numpy(rng.integers(0, k, n).astype(np.float64))
    W = torch.from_numpy(rng.normal(size=d * k))
    mask = torch.from_numpy((rng.random(n) < 0.6).astype(np.uint8)) if use_mask else None
    sw = torch.from_numpy(rng.random(n) * 2) if use_weight else None

    grad, lc = mc.ref_eval_multi(A, y, W, k, mask, True, sw)
    assert float(lc[0]) >= -1e-12 and torch.isfinite(grad).all()
    # margins decomposition
    zf = mc.ref_margins_multi(A, W, k)
    g2, lc2 = mc.ref_eval_multi_from_margins(zf, y, A, k, mask, True, sw)
    torch.testing.assert_close(g2, grad, rtol=1e-9, atol=1e-9)
    torch.testing.assert_close(lc2, lc)
    # row additivity (the all-reduce identity)
    if n >= 2:
        j = n // 2
        g_a, l_a = mc.ref_eval_multi(A[:j], y[:j], W, k,
                                     None if mask is None else mask[:j], True,
                                     None if sw is None else sw[:j])
        g_b, l_b = mc.ref_eval_multi(A[j:], y[j:], W, k,
                                     None if mask is None else mask[j:], True,
                                     None if sw is None else sw[j:])
        torch.testing.assert_close(g_a + g_b, grad, rtol=1e-9, atol=1e-9)
        torch.testing.assert_close(l_a + l_b, lc, rtol=1e-9, atol=1e-9)
    # multiplier rows sum to ~0 per example (softmax sums to 1, minus onehot)
    m, _ = mc.ref_multiplier_multi(zf.reshape(n, k), y)
    torch.testing.assert_close(m.sum(dim=1), torch.zeros(n, dtype=m.dtype),
                               rtol=0, atol=1e-12)


@settings(max_examples=30, deadline=None)
@given(n=st.integers(1, 40), d=st.integers(2, 60), nnz=st.integers(1, 8),
       k=st.integers(2, 12), seed=st.integers(0, 2**31 - 1))
def test_csr_multiclass_matches_densified(n, d, nnz, k, seed):
    """CSR multiclass oracle == dense oracle on the densified matrix,
    including duplicate column indices within a row (accumulated)."""
    from sparkagd_amd.ops import multiclass as mc

    rng = np.random.default_rng(seed)
    col = torch.from_numpy(rng.integers(0, d, n * nnz).astype(np.int32))
    val = torch.from_numpy(rng.normal(size=n * nnz))
    rowptr = torch.arange(0, n * nnz + 1, nnz, dtype=torch.int32)
    y = torch.from_numpy(rng.integers(0, k, n).astype(np.float64))
    W = torch.from_numpy(rng.normal(size=d * k))

    dense = torch.zeros((n, d), dtype=torch.float64)
    rows = torch.repeat_interleave(torch.arange(n), nnz)
    dense.index_put_((rows, col.to(torch.int64)), val, accumulate=True)

    zf = mc.ref_csr_margins_multi(rowptr, col, val, W, k, d)
    zd = mc.ref_margins_multi(dense, W, k)
    torch.testing.assert_close(zf, zd, rtol=1e-9, atol=1e-9)
    m, _ = mc.ref_multiplier_multi(zf.reshape(n, k), y)
    g1 = mc.ref_csr_grad_multi(rowptr, col, val, m, d)
    g2, _ = mc.ref_eval_multi_from_margins(zd, y, dense, k)
    torch.testing.assert_close(g1, g2, rtol=1e-9, atol=1e-9)


@settings(max_examples=25, deadline=None)
@given(n=st.integers(2, 40), d=st.integers(1, 24),
       theta=st.floats(0.01, 1.0), step=st.floats(0.0, 2.0),
       lam=st.floats(0.0, 0.5), seed=st.integers(0, 2**31 - 1))
def test_margin_tracking_linearity(n, d, theta, step, lam, seed):
    """The tracked-margin algebra (optimizer.py) is exact for affine prox:
    margins(update(w)) == update_margins(margins(w)) in f64."""
    from sparkagd_amd.models.updater import SimpleUpdater, SquaredL2Updater

    rng = np.random.default_rng(seed)
    A = torch.from_numpy(rng.normal(size=(n, d)))
    x = torch.from_numpy(rng.normal(size=d))
    z = torch.from_numpy(rng.normal(size=d))
    g = torch.from_numpy(rng.normal(size=d))
    mA = lambda v: ops.reference.dense_margins(A, v)
    for upd in (SimpleUpdater(), SquaredL2Updater()):
        y = ops.axpby(1.0 - theta, x, theta, z)
        ym = ops.axpby(1.0 - theta, mA(x), theta, mA(z))
        torch.testing.assert_close(mA(y), ym, rtol=1e-12, atol=1e-12)
        z2, _ = upd.compute(z, g, step, 1, lam)
        zm2 = upd.prox_margins(mA(z), mA(g), step, lam)
        torch.testing.assert_close(mA(z2), zm2, rtol=1e-10, atol=1e-10)
