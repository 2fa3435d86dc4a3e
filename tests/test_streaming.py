"""Host-streamed shards (capacity beyond HBM; copy/compute overlap on GPU)."""

import math

import pytest
import torch

from sparkagd_amd import (HostStreamedDenseShard, DenseShard, LogisticGradient,
                          SimpleUpdater, SquaredL2Updater,
                          generate_dense_problem, run, ops)


def _mk(n=1000, d=64, seed=5):
    shard, _ = generate_dense_problem(n, d, seed=seed, dtype=torch.float64)
    streamed = HostStreamedDenseShard(shard.features, shard.labels,
                                      device="cpu", chunk_rows=192)
    return shard, streamed


def test_streamed_eval_matches_dense():
    shard, streamed = _mk()
    w = torch.randn(64, dtype=torch.float64,
                    generator=torch.Generator().manual_seed(1))
    g1, lc1 = shard.eval(w, ops.LOSS_LOGISTIC)
    g2, lc2 = streamed.eval(w, ops.LOSS_LOGISTIC)
    torch.testing.assert_close(g1, g2, rtol=1e-12, atol=1e-12)
    torch.testing.assert_close(lc1, lc2, rtol=1e-12, atol=1e-12)
    # margins + from-margins round trip
    z1 = shard.margins(w)
    z2 = streamed.margins(w)
    torch.testing.assert_close(z1, z2)
    g3, lc3 = streamed.eval_from_margins(z2, ops.LOSS_LOGISTIC)
    torch.testing.assert_close(g3, g1, rtol=1e-12, atol=1e-12)
    torch.testing.assert_close(lc3, lc1, rtol=1e-12, atol=1e-12)
    # loss-only
    gn, lc4 = streamed.eval(w, ops.LOSS_LOGISTIC, need_grad=False)
    assert gn is None
    torch.testing.assert_close(lc4, lc1)


def test_streamed_agd_matches_dense():
    """Full AGD trajectory identical on the streamed and in-memory shards
    (tracking engages on both — streamed margins/eval_from_margins exist)."""
    shard, streamed = _mk(n=2000, d=40)
    w0 = torch.zeros(40, dtype=torch.float64)
    args = (LogisticGradient(), SquaredL2Updater(), 1e-10, 25, 0.01, w0,
            1.0, math.inf, 0.5, 0.9, True)
    w1, h1 = run(shard, *args)
    w2, h2 = run(streamed, *args)
    assert len(h1) == len(h2)
    for a, b in zip(h1, h2):
        assert abs(a - b) < 1e-12 * max(1.0, abs(b))
    torch.testing.assert_close(w1, w2, rtol=1e-12, atol=1e-14)


def test_streamed_with_mask_and_weights():
    shard, _ = generate_dense_problem(500, 16, seed=6, dtype=torch.float64)
    sw = torch.rand(500, generator=torch.Generator().manual_seed(2)).double()
    dense = DenseShard(shard.features, shard.labels, sample_weight=sw)
    streamed = HostStreamedDenseShard(shard.features, shard.labels,
                                      device="cpu", chunk_rows=128,
                                      sample_weight=sw)
    mask = (torch.rand(500, generator=torch.Generator().manual_seed(3)) < 0.5).to(torch.uint8)
    w = torch.randn(16, dtype=torch.float64,
                    generator=torch.Generator().manual_seed(4))
    g1, lc1 = dense.eval(w, ops.LOSS_LOGISTIC, mask=mask)
    g2, lc2 = streamed.eval(w, ops.LOSS_LOGISTIC, mask=mask)
    torch.testing.assert_close(g1, g2, rtol=1e-12, atol=1e-12)
    torch.testing.assert_close(lc1, lc2)


def test_streamed_minibatch_matches_dense():
    """run_mini_batch on a streamed shard == on the in-memory shard (the
    per-chunk mask slicing path)."""
    from sparkagd_amd import run_mini_batch

    shard, streamed = _mk(n=1500, d=24)
    w0 = torch.zeros(24, dtype=torch.float64)
    w1, h1 = run_mini_batch(shard, LogisticGradient(), SimpleUpdater(),
                            1.0, 8, 0.0, 0.5, w0)
    w2, h2 = run_mini_batch(streamed, LogisticGradient(), SimpleUpdater(),
                            1.0, 8, 0.0, 0.5, w0)
    torch.testing.assert_close(w1, w2, rtol=1e-12, atol=1e-14)
    for a, b in zip(h1, h2):
        assert abs(a - b) < 1e-12 * max(1.0, abs(b))


@pytest.mark.gpu
def test_streamed_gpu_matches_hbm():
    """Streamed (pinned-host + double-buffered H2D) equals the in-HBM shard
    bitwise — same kernels, same chunk row ranges as grad row-blocks differ,
    so compare at fp32-accumulation tolerance."""
    dev = torch.device("cuda")
    n, d = 30000, 4096
    g = torch.Generator().manual_seed(9)
    feats_cpu = torch.randn((n, d), generator=g).to(torch.bfloat16)
    labels = (torch.rand(n, generator=g) < 0.5).float()
    hbm = DenseShard(feats_cpu.to(dev), labels.to(dev))
    streamed = HostStreamedDenseShard(feats_cpu, labels, device=dev,
                                      chunk_rows=4096)
    w = (torch.randn(d, generator=g) / math.sqrt(d)).float().to(dev)
    g1, lc1 = hbm.eval(w, ops.LOSS_LOGISTIC)
    g2, lc2 = streamed.eval(w, ops.LOSS_LOGISTIC)
    torch.testing.assert_close(lc1, lc2, rtol=1e-9, atol=1e-9)
    num = float(torch.norm(g1 - g2)); den = float(torch.norm(g1)) + 1e-30
    assert num / den < 1e-5
    z1 = hbm.margins(w)
    z2 = streamed.margins(w)
    torch.testing.assert_close(z1, z2, rtol=1e-6, atol=1e-6)
    # e2e: a few AGD iterations on the streamed shard converge
    w0 = torch.zeros(d, device=dev)
    wr, h = run(streamed, LogisticGradient(), SimpleUpdater(), 1e-10, 5, 0.0,
                w0, 1.0, math.inf, 0.5, 0.9, True)
    assert h[-1] < h[0] and all(math.isfinite(x) for x in h)


@pytest.mark.gpu
def test_streamed_back_to_back_passes_no_corruption():
    """Regression for the advisor-flagged double-buffer race: two
    back-to-back feature passes with NO intervening host sync (margins(x)
    then margins(z), exactly the optimizer's refresh pattern) must not let
    the second pass's H2D copies overwrite chunks the first pass's tail
    kernels still read. Repeat several times — the race was timing
    dependent."""
    import torch as _t

    from sparkagd_amd.data import DenseShard
    from sparkagd_amd.streaming import HostStreamedDenseShard

    g = _t.Generator().manual_seed(41)
    n, d = 16384, 2048
    feats = _t.randn((n, d), generator=g)
    labels = (_t.rand(n, generator=g) < 0.5).float()
    dev_shard = DenseShard(feats.to("cuda:0"), labels.to("cuda:0"))
    st = HostStreamedDenseShard(feats, labels, device="cuda:0",
                                chunk_rows=1024)
    x = _t.randn(d, generator=g).to("cuda:0")
    z = _t.randn(d, generator=g).to("cuda:0")
    ref_x = dev_shard.margins(x)
    ref_z = dev_shard.margins(z)
    for _ in range(6):
        mx = st.margins(x)   # no torch.cuda.synchronize between these
        mz = st.margins(z)
        _t.testing.assert_close(mx, ref_x, rtol=1e-4, atol=1e-4)
        _t.testing.assert_close(mz, ref_z, rtol=1e-4, atol=1e-4)
