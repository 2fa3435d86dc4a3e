"""Checkpoint/resume tests (a new capability — the reference has none,
SURVEY.md §5 'Checkpoint / resume')."""

import math

import torch

from sparkagd_amd import (
    LogisticGradient,
    SquaredL2Updater,
    generate_logistic_data,
    run,
)
from sparkagd_amd.utils.checkpoint import load_checkpoint, save_checkpoint


def test_roundtrip_bitwise(tmp_path):
    p = str(tmp_path / "ckpt.safetensors")
    x = torch.randn(1000, dtype=torch.float32)
    z = torch.randn(1000, dtype=torch.float32)
    save_checkpoint(p, x=x, z=z, theta=math.inf, L=3.25, iter=7,
                    backtrack_simple=False, loss_history=[1.0, 0.5, 0.25])
    st = load_checkpoint(p)
    assert torch.equal(st["x"], x) and torch.equal(st["z"], z)
    assert math.isinf(st["theta"]) and st["L"] == 3.25 and st["iter"] == 7
    assert st["backtrack_simple"] is False
    assert st["loss_history"] == [1.0, 0.5, 0.25]


def test_theta_finite_roundtrip(tmp_path):
    p = str(tmp_path / "c2.safetensors")
    x = torch.randn(8, dtype=torch.float64)
    save_checkpoint(p, x=x, z=x, theta=0.123456789012345, L=1e-3, iter=1,
                    backtrack_simple=True, loss_history=[])
    st = load_checkpoint(p)
    assert st["theta"] == 0.123456789012345  # repr() round-trip is exact


def test_resume_equals_uninterrupted(tmp_path):
    """optimize(..., resume_from=...) continues bit-identically: run 10 iters
    straight vs 5 iters + checkpoint + resume for 5 more (track_margins=False
    — tracked margins are recomputed on resume, which is equivalent only up
    to fp rounding; the tolerant variant is tested below)."""
    data = generate_logistic_data(2.0, -1.5, 2000, seed=11)
    w0 = torch.tensor([0.3, 0.12], dtype=torch.float64)
    args = (data, LogisticGradient(), SquaredL2Updater(), 0.0, 10, 0.2, w0,
            1.0, math.inf, 0.5, 0.9, True)

    w_full, hist_full = run(*args, track_margins=False)

    p = str(tmp_path / "mid.safetensors")
    args5 = (data, LogisticGradient(), SquaredL2Updater(), 0.0, 5, 0.2, w0,
             1.0, math.inf, 0.5, 0.9, True)
    run(*args5, checkpoint_path=p, checkpoint_every=5, track_margins=False)
    w_res, hist_res = run(*args, resume_from=p, track_margins=False)

    assert torch.equal(w_full, w_res)
    assert hist_res == hist_full


def test_resume_with_margin_tracking_close(tmp_path):
    data = generate_logistic_data(2.0, -1.5, 2000, seed=11)
    w0 = torch.tensor([0.3, 0.12], dtype=torch.float64)
    args = (data, LogisticGradient(), SquaredL2Updater(), 0.0, 10, 0.2, w0,
            1.0, math.inf, 0.5, 0.9, True)
    w_full, hist_full = run(*args)
    p = str(tmp_path / "mid2.safetensors")
    run(data, LogisticGradient(), SquaredL2Updater(), 0.0, 5, 0.2, w0,
        1.0, math.inf, 0.5, 0.9, True, checkpoint_path=p, checkpoint_every=5)
    w_res, hist_res = run(*args, resume_from=p)
    torch.testing.assert_close(w_res, w_full, rtol=1e-9, atol=1e-12)
    assert len(hist_res) == len(hist_full)


def test_margin_tracking_matches_untracked():
    """Tracking is an algebraic identity: same trajectory up to fp rounding,
    and exactly one fewer data-pass class (validated on CPU fp64)."""
    data = generate_logistic_data(2.0, -1.5, 5000, seed=13)
    w0 = torch.tensor([0.3, 0.12], dtype=torch.float64)
    args = (data, LogisticGradient(), SquaredL2Updater(), 1e-12, 12, 0.2, w0,
            1.0, math.inf, 0.5, 0.9, True)
    w_t, hist_t = run(*args)                         # auto -> tracking
    w_u, hist_u = run(*args, track_margins=False)
    assert len(hist_t) == len(hist_u)
    for a, b in zip(hist_t, hist_u):
        assert abs(a - b) < 1e-9 * max(1.0, abs(b))
    torch.testing.assert_close(w_t, w_u, rtol=1e-8, atol=1e-10)


def test_margin_refresh_every():
    data = generate_logistic_data(2.0, -1.5, 2000, seed=14)
    w0 = torch.zeros(2, dtype=torch.float64)
    w1, h1 = run(data, LogisticGradient(), SquaredL2Updater(), 1e-12, 10, 0.1,
                 w0, 1.0, math.inf, 0.5, 0.9, True, margin_refresh_every=3)
    w2, h2 = run(data, LogisticGradient(), SquaredL2Updater(), 1e-12, 10, 0.1,
                 w0, 1.0, math.inf, 0.5, 0.9, True)
    torch.testing.assert_close(w1, w2, rtol=1e-8, atol=1e-10)
