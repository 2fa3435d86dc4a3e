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
