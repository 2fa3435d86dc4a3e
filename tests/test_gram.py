"""Dual-space (Gram) solver: trajectory identity with the direct solver."""

import math

import pytest
import torch

from sparkagd_amd import (
    LogisticGradient,
    LeastSquaresGradient,
    L1Updater,
    SimpleUpdater,
    SquaredL2Updater,
    run,
)
from sparkagd_amd.data import generate_dense_problem, generate_logistic_data
from sparkagd_amd.gram import GramOperator, run_gram
from sparkagd_amd import ops


def _args(data, updater, reg, w0, iters=12, tol=1e-12):
    return (data, LogisticGradient(), updater, tol, iters, reg, w0,
            1.0, math.inf, 0.5, 0.9, True)


@pytest.mark.parametrize("reg,updater_cls", [(0.0, SimpleUpdater), (0.2, SquaredL2Updater)])
def test_gram_matches_direct(reg, updater_cls):
    data = generate_logistic_data(2.0, -1.5, 4000, seed=21)
    w0 = torch.tensor([0.3, 0.12], dtype=torch.float64)
    w_d, h_d = run(*_args(data, updater_cls(), reg, w0))
    w_g, h_g = run(*_args(data, updater_cls(), reg, w0), solver="gram")
    assert len(h_d) == len(h_g)
    for a, b in zip(h_d, h_g):
        assert abs(a - b) < 1e-8 * max(1.0, abs(b)), (a, b)
    torch.testing.assert_close(w_g, w_d, rtol=1e-7, atol=1e-9)


def test_gram_nonzero_x0_and_wide_d():
    shard, _ = generate_dense_problem(300, 900, seed=4, dtype=torch.float64)
    w0 = torch.randn(900, dtype=torch.float64, generator=torch.Generator().manual_seed(0)) * 0.1
    args = (shard, LogisticGradient(), SquaredL2Updater(), 1e-12, 10, 0.05, w0,
            1.0, math.inf, 0.5, 0.9, True)
    w_d, h_d = run(*args)
    w_g, h_g = run(*args, solver="gram")
    for a, b in zip(h_d, h_g):
        assert abs(a - b) < 1e-8 * max(1.0, abs(b))
    torch.testing.assert_close(w_g, w_d, rtol=1e-6, atol=1e-8)


def test_gram_least_squares_beta_ge_1():
    shard, _ = generate_dense_problem(200, 500, seed=7,
                                      loss_type=ops.LOSS_LEAST_SQUARES,
                                      dtype=torch.float64)
    w0 = torch.zeros(500, dtype=torch.float64)
    feats = shard.features
    Lsafe = float(2 * (feats * feats).sum() / feats.shape[0])
    args = (shard, LeastSquaresGradient(), SimpleUpdater(), 1e-12, 15, 0.0, w0,
            Lsafe, Lsafe, 1.5, 1.0, True)
    w_d, h_d = run(*args)
    w_g, h_g = run(*args, solver="gram", loss_history_mode="backtrack")
    assert len(h_d) == len(h_g)
    torch.testing.assert_close(w_g, w_d, rtol=1e-6, atol=1e-8)


def test_gram_hinge_converges():
    """Hinge multiplier is discontinuous, so bitwise trajectory identity is
    not expected; the Gram solver must still converge."""
    from sparkagd_amd import HingeGradient

    shard, _ = generate_dense_problem(1000, 300, seed=9,
                                      loss_type=ops.LOSS_HINGE,
                                      dtype=torch.float64)
    w0 = torch.zeros(300, dtype=torch.float64)
    w, h = run(shard, HingeGradient(), SquaredL2Updater(), 1e-12, 20, 0.01, w0,
               1.0, math.inf, 0.5, 0.9, True, solver="gram")
    assert h[-1] < h[0] and h[-1] < 1.0


def test_gram_via_class_api_config():
    from sparkagd_amd import AcceleratedGradientDescent, AGDConfig

    data = generate_logistic_data(2.0, -1.5, 1500, seed=2)
    cfg = AGDConfig(num_iterations=6, convergence_tol=1e-12, solver="gram")
    opt = AcceleratedGradientDescent(LogisticGradient(), SimpleUpdater(), cfg)
    w = opt.optimize(data, torch.zeros(2, dtype=torch.float64))
    assert len(opt.loss_history) == 6


def test_gram_rejects_nonaffine_and_checkpoint():
    data = generate_logistic_data(2.0, -1.5, 500, seed=1)
    w0 = torch.zeros(2, dtype=torch.float64)
    with pytest.raises(ValueError):
        run(*_args(data, L1Updater(), 0.1, w0), solver="gram")
    with pytest.raises(ValueError):
        run(*_args(data, SimpleUpdater(), 0.0, w0), solver="gram",
            checkpoint_path="/tmp/x.safetensors", checkpoint_every=1)


def test_gram_operator_reuse():
    """A prebuilt GramOperator can be reused across solves (warm restarts)."""
    from sparkagd_amd.parallel.comm import Communicator

    data = generate_logistic_data(2.0, -1.5, 1000, seed=3)
    op = GramOperator(data, Communicator())
    assert op.K.shape == (1000, 1000)
    w0 = torch.zeros(2, dtype=torch.float64)
    w1, h1 = run_gram(data, LogisticGradient(), SimpleUpdater(), 1e-12, 5, 0.0,
                      w0, 1.0, math.inf, 0.5, 0.9, True, gram_op=op)
    w2, h2 = run_gram(data, LogisticGradient(), SimpleUpdater(), 1e-12, 5, 0.0,
                      w0, 1.0, math.inf, 0.5, 0.9, True, gram_op=op)
    assert torch.equal(w1, w2) and h1 == h2


def test_gram_checkpoint_resume_explicit_error():
    """Checkpoint/resume is a direct-solver capability; the Gram solver must
    refuse it LOUDLY, not silently skip checkpointing."""
    import pytest

    full = generate_logistic_data(2.0, -1.5, 500, seed=9)
    w0 = torch.tensor([0.1, 0.1], dtype=torch.float64)
    with pytest.raises(ValueError, match="direct solver"):
        run(full, LogisticGradient(), SquaredL2Updater(), 1e-12, 3, 0.1, w0,
            1.0, math.inf, 0.5, 0.9, True, solver="gram",
            checkpoint_path="/tmp/never_written.safetensors", checkpoint_every=1)
    with pytest.raises(ValueError, match="direct solver"):
        run(full, LogisticGradient(), SquaredL2Updater(), 1e-12, 3, 0.1, w0,
            1.0, math.inf, 0.5, 0.9, True, solver="gram",
            resume_from="/tmp/does_not_exist.safetensors")


def test_gram_alternate_backtracking_matches_direct():
    """backtrack_tol=inf forces the ALTERNATE backtracking test from trial 2
    on — in the fused GPU path this mixes the fused y-registration with the
    generic x-basis registration, so the bookkeeping (T, XB, Mstore, G)
    must stay consistent across both."""
    data = generate_logistic_data(2.0, -1.5, 3000, seed=77)
    w0 = torch.tensor([0.3, 0.12], dtype=torch.float64)
    args = (data, LogisticGradient(), SquaredL2Updater(), 1e-12, 8, 0.1, w0,
            1.0, math.inf, 0.5, 0.9, True)
    w_d, h_d = run(*args, backtrack_tol=math.inf)
    w_g, h_g = run(*args, solver="gram", backtrack_tol=math.inf)
    assert len(h_d) == len(h_g)
    for a, b in zip(h_d, h_g):
        assert abs(a - b) < 1e-7 * max(1.0, abs(b)), (a, b)
    torch.testing.assert_close(torch.as_tensor(w_g), torch.as_tensor(w_d),
                               rtol=1e-6, atol=1e-9)


@pytest.mark.parametrize("seed", [101, 202, 303])
def test_gram_matches_direct_randomized(seed):
    """Property sweep: random problem shapes/regs/losses — the Gram solver's
    trajectory equals the direct solver's (f64 CPU, tight tolerance)."""
    import numpy as _np

    rng = _np.random.RandomState(seed)
    n = int(rng.randint(300, 1500))
    d = int(rng.randint(3, 40))
    reg = float(rng.choice([0.0, 0.01, 0.3]))
    iters = int(rng.randint(4, 12))
    torch.manual_seed(seed)
    feats = torch.randn(n, d, dtype=torch.float64) / math.sqrt(d)
    labels = (torch.rand(n, dtype=torch.float64) < 0.5).double()
    from sparkagd_amd.data import DenseShard

    data = DenseShard(feats, labels)
    w0 = torch.randn(d, dtype=torch.float64) * 0.1
    upd = SquaredL2Updater() if reg > 0 else SimpleUpdater()
    w_d, h_d = run(data, LogisticGradient(), upd, 1e-14, iters, reg, w0,
                   1.0, math.inf, 0.5, 0.9, True)
    w_g, h_g = run(data, LogisticGradient(), upd, 1e-14, iters, reg, w0,
                   1.0, math.inf, 0.5, 0.9, True, solver="gram")
    assert len(h_d) == len(h_g)
    for a, b in zip(h_d, h_g):
        assert abs(a - b) < 1e-7 * max(1.0, abs(b)), (seed, a, b)
    torch.testing.assert_close(torch.as_tensor(w_g), torch.as_tensor(w_d),
                               rtol=1e-6, atol=1e-9)
