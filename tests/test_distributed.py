"""Multi-process distributed tests on CPU via gloo, world_size=2.

The analog of the reference's `local-cluster[2,1,512]` tier
(``AcceleratedGradientDescentSuite.scala:242-260``): real separate processes,
real collectives, no GPU needed — the exact code path that runs over RCCL on
an 8-GPU node, exercised with the gloo backend.
"""

import math
import os

import pytest
import torch
import torch.multiprocessing as mp

from sparkagd_amd import (
    Communicator,
    LogisticGradient,
    SimpleUpdater,
    SquaredL2Updater,
    generate_logistic_data,
    run,
    run_mini_batch,
)
from sparkagd_amd.data import DenseShard, shard_range

N = 4000


def _init_file() -> str:
    """Fresh FileStore rendezvous path. File-based init removes the
    bind-release-rebind race an ephemeral TCP port helper has (another
    process can grab the port between release and rebind)."""
    import tempfile

    fd, path = tempfile.mkstemp(prefix="sparkagd_dist_", suffix=".init")
    os.close(fd)
    os.unlink(path)  # init_method="file://" wants a nonexistent path
    return path


def _pool_worker(rank, world, in_q, out_q, init_file):
    """Persistent gloo worker: init the process group ONCE, then execute
    picklable (rank, world) callables from the queue until the None
    sentinel. Reusing workers across tests cuts the tier's dominant cost
    (a process spawn + torch import + group init per test)."""
    import traceback

    torch.distributed.init_process_group(
        "gloo", init_method=f"file://{init_file}", rank=rank, world_size=world)
    try:
        while True:
            fn = in_q.get()
            if fn is None:
                break
            try:
                out_q.put((rank, ("ok", fn(rank, world))))
            except Exception:  # noqa: BLE001 — report, stay alive
                out_q.put((rank, ("err", traceback.format_exc())))
    finally:
        torch.distributed.destroy_process_group()


class _DistPool:
    def __init__(self, world):
        ctx = mp.get_context("spawn")
        self.world = world
        self.in_qs = [ctx.Queue() for _ in range(world)]
        self.out_q = ctx.Queue()
        init_file = _init_file()
        self.procs = [
            ctx.Process(target=_pool_worker,
                        args=(r, world, self.in_qs[r], self.out_q, init_file),
                        daemon=True)
            for r in range(world)
        ]
        for p in self.procs:
            p.start()

    def run(self, fn):
        for q in self.in_qs:
            q.put(fn)
        results = {}
        for _ in range(self.world):
            rank, (status, res) = self.out_q.get(timeout=300)
            assert status == "ok", f"rank {rank} failed:\n{res}"
            results[rank] = res
        return results

    def close(self):
        for q in self.in_qs:
            q.put(None)
        for p in self.procs:
            p.join(timeout=60)


_pools = {}


def _run_dist(fn, world=2):
    pool = _pools.get(world)
    if pool is None or any(not p.is_alive() for p in pool.procs):
        pool = _pools[world] = _DistPool(world)
    return pool.run(fn)


def teardown_module(_m):
    for pool in _pools.values():
        pool.close()
    _pools.clear()


def _make_shard(rank, world):
    full = generate_logistic_data(2.0, -1.5, N, seed=42)
    lo, hi = shard_range(N, rank, world)
    return DenseShard(full.features[lo:hi], full.labels[lo:hi]), full


def _dist_agd(rank, world):
    shard, _ = _make_shard(rank, world)
    comm = Communicator()
    w0 = torch.tensor([0.3, 0.12], dtype=torch.float64)
    w, hist = run(shard, LogisticGradient(), SquaredL2Updater(), 1e-12, 8, 0.2,
                  w0, 1.0, math.inf, 0.5, 0.9, True, comm=comm)
    return w.numpy().tolist(), hist


def _dist_minibatch(rank, world):
    shard, _ = _make_shard(rank, world)
    comm = Communicator()
    w0 = torch.tensor([0.3, 0.12], dtype=torch.float64)
    w, hist = run_mini_batch(shard, LogisticGradient(), SimpleUpdater(), 1.0, 10,
                             0.0, 0.5, w0, comm=comm, seed=7)
    return w.numpy().tolist(), hist


def test_sharded_agd_matches_single_process():
    """Row-sharded 2-process AGD == single-process AGD on the same data
    (the replicated-update determinism the design relies on)."""
    results = _run_dist(_dist_agd, world=2)
    full = generate_logistic_data(2.0, -1.5, N, seed=42)
    w0 = torch.tensor([0.3, 0.12], dtype=torch.float64)
    w_ref, hist_ref = run(full, LogisticGradient(), SquaredL2Updater(), 1e-12, 8,
                          0.2, w0, 1.0, math.inf, 0.5, 0.9, True)
    for rank in (0, 1):
        w_r, hist_r = results[rank]
        torch.testing.assert_close(
            torch.tensor(w_r, dtype=torch.float64), w_ref, rtol=1e-9, atol=1e-12
        )
        assert len(hist_r) == len(hist_ref)
        for a, b in zip(hist_r, hist_ref):
            assert abs(a - b) < 1e-9 * max(1.0, abs(b))
    # Replicated update: both ranks produced bit-identical state
    assert results[0][0] == results[1][0]


def test_sharded_minibatch_runs_and_replicates():
    results = _run_dist(_dist_minibatch, world=2)
    assert results[0][0] == results[1][0]
    assert len(results[0][1]) == 10
    # loss decreased overall
    assert results[0][1][-1] < results[0][1][0]


def _dist_comm_primitives(rank, world):
    comm = Communicator()
    t = torch.full((4,), float(rank + 1), dtype=torch.float64)
    comm.allreduce_(t)
    ok_allreduce = bool(torch.all(t == 3.0))
    b = torch.full((3,), float(rank), dtype=torch.float64)
    comm.broadcast_(b, src=0)
    ok_bcast = bool(torch.all(b == 0.0))
    rep = torch.ones(5) * 2.0
    ok_rep = comm.check_replicated(rep)
    div = torch.ones(5) * float(rank)
    ok_div = not comm.check_replicated(div)
    return ok_allreduce, ok_bcast, ok_rep, ok_div


def test_comm_primitives():
    results = _run_dist(_dist_comm_primitives, world=2)
    for rank in (0, 1):
        assert all(results[rank]), results[rank]


def _dist_gram_uneven(rank, world):
    # odd global row count -> uneven shards (exercises the padded all_gather)
    full = generate_logistic_data(2.0, -1.5, N + 1, seed=43)
    lo, hi = shard_range(N + 1, rank, world)
    shard = DenseShard(full.features[lo:hi], full.labels[lo:hi])
    comm = Communicator()
    w0 = torch.tensor([0.3, 0.12], dtype=torch.float64)
    w, hist = run(shard, LogisticGradient(), SquaredL2Updater(), 1e-12, 5, 0.2,
                  w0, 1.0, math.inf, 0.5, 0.9, True, comm=comm, solver="gram")
    return w.numpy().tolist(), hist


def test_sharded_gram_uneven_rows():
    results = _run_dist(_dist_gram_uneven, world=2)
    full = generate_logistic_data(2.0, -1.5, N + 1, seed=43)
    w0 = torch.tensor([0.3, 0.12], dtype=torch.float64)
    w_ref, hist_ref = run(full, LogisticGradient(), SquaredL2Updater(), 1e-12, 5,
                          0.2, w0, 1.0, math.inf, 0.5, 0.9, True)
    for rank in (0, 1):
        torch.testing.assert_close(
            torch.tensor(results[rank][0], dtype=torch.float64), w_ref,
            rtol=1e-6, atol=1e-8)


def _dist_multiclass(rank, world):
    from sparkagd_amd import MultinomialLogisticGradient
    from sparkagd_amd.data import generate_multiclass_problem

    full, _ = generate_multiclass_problem(1200, 12, 4, seed=44,
                                          dtype=torch.float64, label_noise=0.1)
    lo, hi = shard_range(1200, rank, world)
    shard = DenseShard(full.features[lo:hi], full.labels[lo:hi])
    comm = Communicator()
    w0 = torch.zeros(48, dtype=torch.float64)
    w, hist = run(shard, MultinomialLogisticGradient(4), SquaredL2Updater(),
                  1e-12, 10, 0.01, w0, 1.0, math.inf, 0.5, 0.9, True,
                  comm=comm)
    return w.numpy().tolist(), hist


def test_sharded_multiclass_matches_single_process():
    """2-rank multinomial softmax (grad [d*K] + f64 loss_count all-reduce)
    equals the single-process run bit-for-bit at f64."""
    from sparkagd_amd import MultinomialLogisticGradient
    from sparkagd_amd.data import generate_multiclass_problem

    results = _run_dist(_dist_multiclass, world=2)
    full, _ = generate_multiclass_problem(1200, 12, 4, seed=44,
                                          dtype=torch.float64, label_noise=0.1)
    w0 = torch.zeros(48, dtype=torch.float64)
    w_ref, hist_ref = run(full, MultinomialLogisticGradient(4),
                          SquaredL2Updater(), 1e-12, 10, 0.01, w0,
                          1.0, math.inf, 0.5, 0.9, True)
    for rank in (0, 1):
        torch.testing.assert_close(
            torch.tensor(results[rank][0], dtype=torch.float64), w_ref,
            rtol=1e-9, atol=1e-12)
        for a, b in zip(results[rank][1], hist_ref):
            assert abs(a - b) < 1e-9 * max(1.0, abs(b))


def _dist_gram_multiclass(rank, world):
    from sparkagd_amd import MultinomialLogisticGradient
    from sparkagd_amd.data import generate_multiclass_problem

    full, _ = generate_multiclass_problem(901, 10, 3, seed=46,
                                          dtype=torch.float64, label_noise=0.2)
    lo, hi = shard_range(901, rank, world)
    shard = DenseShard(full.features[lo:hi], full.labels[lo:hi])
    comm = Communicator()
    w0 = torch.zeros(30, dtype=torch.float64)
    w, hist = run(shard, MultinomialLogisticGradient(3), SquaredL2Updater(),
                  1e-12, 8, 0.01, w0, 1.0, math.inf, 0.5, 0.9, True,
                  comm=comm, solver="gram")
    return w.numpy().tolist(), hist


def test_sharded_gram_multiclass_matches_single():
    """2-rank Gram solver on a multiclass problem (uneven shards -> padded
    class-column all_gather) == single-process direct solver."""
    from sparkagd_amd import MultinomialLogisticGradient
    from sparkagd_amd.data import generate_multiclass_problem

    results = _run_dist(_dist_gram_multiclass, world=2)
    full, _ = generate_multiclass_problem(901, 10, 3, seed=46,
                                          dtype=torch.float64, label_noise=0.2)
    w0 = torch.zeros(30, dtype=torch.float64)
    w_ref, hist_ref = run(full, MultinomialLogisticGradient(3),
                          SquaredL2Updater(), 1e-12, 8, 0.01, w0,
                          1.0, math.inf, 0.5, 0.9, True)
    for rank in (0, 1):
        torch.testing.assert_close(
            torch.tensor(results[rank][0], dtype=torch.float64), w_ref,
            rtol=1e-6, atol=1e-8)
        for a, b in zip(results[rank][1], hist_ref):
            assert abs(a - b) < 1e-7 * max(1.0, abs(b))


def test_sharded_gram_world3():
    """Odd world size (3 ranks, uneven rows) — exercises the padded
    all_gather and cross-rank K blocks off the power-of-two path."""
    results = _run_dist(_dist_gram_uneven, world=3)
    full = generate_logistic_data(2.0, -1.5, N + 1, seed=43)
    w0 = torch.tensor([0.3, 0.12], dtype=torch.float64)
    w_ref, _ = run(full, LogisticGradient(), SquaredL2Updater(), 1e-12, 5,
                   0.2, w0, 1.0, math.inf, 0.5, 0.9, True)
    for rank in (0, 1, 2):
        torch.testing.assert_close(
            torch.tensor(results[rank][0], dtype=torch.float64), w_ref,
            rtol=1e-6, atol=1e-8)


def _dist_gram(rank, world):
    shard, _ = _make_shard(rank, world)
    comm = Communicator()
    w0 = torch.tensor([0.3, 0.12], dtype=torch.float64)
    w, hist = run(shard, LogisticGradient(), SquaredL2Updater(), 1e-12, 8, 0.2,
                  w0, 1.0, math.inf, 0.5, 0.9, True, comm=comm, solver="gram")
    return w.numpy().tolist(), hist


def test_sharded_gram_matches_single_process():
    """2-process Gram solver (cross-rank K blocks via chunked broadcast) ==
    single-process direct solver on the same data."""
    results = _run_dist(_dist_gram, world=2)
    full = generate_logistic_data(2.0, -1.5, N, seed=42)
    w0 = torch.tensor([0.3, 0.12], dtype=torch.float64)
    w_ref, hist_ref = run(full, LogisticGradient(), SquaredL2Updater(), 1e-12, 8,
                          0.2, w0, 1.0, math.inf, 0.5, 0.9, True)
    for rank in (0, 1):
        w_r, hist_r = results[rank]
        torch.testing.assert_close(
            torch.tensor(w_r, dtype=torch.float64), w_ref, rtol=1e-6, atol=1e-8
        )
        assert len(hist_r) == len(hist_ref)
        for a, b in zip(hist_r, hist_ref):
            assert abs(a - b) < 1e-7 * max(1.0, abs(b))
    assert results[0][0] == results[1][0]


# ---------------------------------------------------------------------------
# Fault injection (SURVEY.md §5 'Failure detection'): kill a rank mid-run,
# assert the survivor aborts cleanly (collective error, not a hang), and that
# training resumes from the last checkpoint bit-identically with the math of
# an uninterrupted run at the same iteration count.
# ---------------------------------------------------------------------------

def _fault_worker(rank, world, ckpt_path, out_q, init_file):
    import datetime

    torch.distributed.init_process_group(
        "gloo", init_method=f"file://{init_file}", rank=rank, world_size=world,
        timeout=datetime.timedelta(seconds=20),
    )
    shard, _ = _make_shard(rank, world)
    comm = Communicator()
    w0 = torch.tensor([0.3, 0.12], dtype=torch.float64)

    def hook(n_iter):
        if rank == 1 and n_iter == 3:
            os._exit(17)  # simulated hard crash mid-run
        return None

    try:
        run(shard, LogisticGradient(), SquaredL2Updater(), 0.0, 10, 0.2, w0,
            1.0, math.inf, 0.5, 0.9, True, comm=comm,
            checkpoint_path=ckpt_path, checkpoint_every=1, iteration_hook=hook)
        out_q.put((rank, "completed"))
    except Exception as e:  # noqa: BLE001
        out_q.put((rank, f"error:{type(e).__name__}"))


def test_rank_failure_aborts_cleanly_and_resumes(tmp_path):
    ckpt = str(tmp_path / "fault.safetensors")
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    init_file = _init_file()
    procs = [
        ctx.Process(target=_fault_worker, args=(r, 2, ckpt, q, init_file))
        for r in range(2)
    ]
    for p in procs:
        p.start()
    # rank 1 hard-exits before reporting, so only rank 0's result ever
    # arrives — stop as soon as we have it (waiting a second full timeout
    # for the dead rank was the CPU tier's 2-minute hot spot)
    rank0_result = None
    for _ in range(2):
        try:
            rank, res = q.get(timeout=120)
        except Exception:  # noqa: BLE001 - rank 1 died without reporting
            break
        if rank == 0:
            rank0_result = res
            break
    for p in procs:
        p.join(timeout=60)
    assert procs[1].exitcode == 17  # the injected crash
    # survivor must abort with a collective error, not hang or "complete"
    assert rank0_result is not None and rank0_result.startswith("error:"), rank0_result

    # recovery: resume single-process from the last checkpoint and finish;
    # the result must equal an uninterrupted single-process run.
    assert os.path.exists(ckpt)
    full = generate_logistic_data(2.0, -1.5, N, seed=42)
    w0 = torch.tensor([0.3, 0.12], dtype=torch.float64)
    w_res, hist_res = run(full, LogisticGradient(), SquaredL2Updater(), 0.0, 10,
                          0.2, w0, 1.0, math.inf, 0.5, 0.9, True, resume_from=ckpt)
    w_ref, hist_ref = run(full, LogisticGradient(), SquaredL2Updater(), 0.0, 10,
                          0.2, w0, 1.0, math.inf, 0.5, 0.9, True)
    torch.testing.assert_close(w_res, w_ref, rtol=1e-9, atol=1e-12)
    assert len(hist_res) == len(hist_ref)


# ---------------------------------------------------------------------------
# init_from_env: the torchrun rendezvous path bench.py runs under
# (RANK/WORLD_SIZE/MASTER_* env vars -> init_process_group -> Communicator).
# Exercised here with the gloo backend; on a GPU node the same code selects
# nccl (=RCCL) and binds the device from LOCAL_RANK.
# ---------------------------------------------------------------------------

def _env_worker(rank, world, port, out_q):
    from sparkagd_amd.parallel.comm import init_from_env

    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    comm = init_from_env()
    try:
        t = torch.full((4,), float(rank + 1), dtype=torch.float64)
        comm.allreduce_(t)
        out_q.put((rank, (comm.rank, comm.world_size, t.numpy().tolist())))
    finally:
        torch.distributed.destroy_process_group()


def test_init_from_env_rendezvous():
    import socket

    ctx = mp.get_context("spawn")
    last_err = None
    for _attempt in range(3):  # env:// needs a real port; retry on collision
        with socket.socket() as sock:
            sock.bind(("127.0.0.1", 0))
            port = sock.getsockname()[1]
        q = ctx.Queue()
        procs = [ctx.Process(target=_env_worker, args=(r, 2, port, q))
                 for r in range(2)]
        for p in procs:
            p.start()
        try:
            results = {}
            for _ in range(2):
                rank, res = q.get(timeout=120)
                results[rank] = res
            for p in procs:
                p.join(timeout=60)
                assert p.exitcode == 0
            for rank in (0, 1):
                got_rank, got_world, reduced = results[rank]
                assert got_rank == rank and got_world == 2
                assert reduced == [3.0, 3.0, 3.0, 3.0]
            return
        except Exception as e:  # noqa: BLE001 - port stolen between probes
            last_err = e
            for p in procs:
                if p.is_alive():
                    p.terminate()
                p.join(timeout=30)
    raise AssertionError(f"init_from_env rendezvous failed 3x: {last_err}")


def _dist_csr(rank, world):
    from sparkagd_amd.data import generate_csr_problem

    full, _ = generate_csr_problem(3001, 400, 10, seed=61)
    lo, hi = shard_range(3001, rank, world)
    rp = full.rowptr[lo:hi + 1].to(torch.int64)
    k_lo, k_hi = int(rp[0]), int(rp[-1])
    from sparkagd_amd.data import CSRShard

    shard = CSRShard(rp - k_lo, full.col[k_lo:k_hi], full.val[k_lo:k_hi],
                     full.labels[lo:hi], full.d)
    comm = Communicator()
    w0 = torch.zeros(400, dtype=torch.float32)
    w, hist = run(shard, LogisticGradient(), SquaredL2Updater(), 1e-12, 6,
                  0.05, w0, 1.0, math.inf, 0.5, 0.9, True, comm=comm)
    return w.numpy().tolist(), hist


def test_sharded_csr_matches_single_process():
    """Row-sharded CSR shards across 2 ranks == single-process run
    (exercises the deterministic CSC gather + all-reduce on sparse)."""
    from sparkagd_amd.data import generate_csr_problem

    results = _run_dist(_dist_csr, world=2)
    full, _ = generate_csr_problem(3001, 400, 10, seed=61)
    w0 = torch.zeros(400, dtype=torch.float32)
    w_ref, hist_ref = run(full, LogisticGradient(), SquaredL2Updater(), 1e-12,
                          6, 0.05, w0, 1.0, math.inf, 0.5, 0.9, True)
    for rank in (0, 1):
        torch.testing.assert_close(
            torch.tensor(results[rank][0]), w_ref, rtol=1e-4, atol=1e-5)
    assert results[0][0] == results[1][0]


def _dist_mixed(rank, world):
    from sparkagd_amd.data import CSRShard, MixedShard, generate_csr_problem

    full, _ = generate_csr_problem(2000, 300, 8, seed=62)
    lo, hi = shard_range(2000, rank, world)
    rp = full.rowptr[lo:hi + 1].to(torch.int64)
    k_lo, k_hi = int(rp[0]), int(rp[-1])
    csr_part = CSRShard(rp - k_lo, full.col[k_lo:k_hi], full.val[k_lo:k_hi],
                        full.labels[lo:hi], full.d)
    # one dense block per rank with its own synthetic rows
    from sparkagd_amd.data import generate_dense_problem

    dense_part, _ = generate_dense_problem(500, 300, seed=63 + rank,
                                           dtype=torch.float32)
    shard = MixedShard([dense_part, csr_part])
    comm = Communicator()
    w0 = torch.zeros(300, dtype=torch.float32)
    w, hist = run(shard, LogisticGradient(), SquaredL2Updater(), 1e-12, 6,
                  0.05, w0, 1.0, math.inf, 0.5, 0.9, True, comm=comm)
    return w.numpy().tolist(), hist


def test_sharded_mixed_replicates():
    """Heterogeneous (dense+CSR) shards across ranks: ranks stay
    bit-identical under the replicated update."""
    results = _run_dist(_dist_mixed, world=2)
    assert results[0][0] == results[1][0]
    assert results[0][1][-1] < results[0][1][0]


def test_sharded_agd_world4():
    """4 ranks, uneven shards — the widest CPU-tier rank layout (the driver
    scales 1/2/4/8 on hardware)."""
    results = _run_dist(_dist_agd, world=4)
    full = generate_logistic_data(2.0, -1.5, N, seed=42)
    w0 = torch.tensor([0.3, 0.12], dtype=torch.float64)
    w_ref, _ = run(full, LogisticGradient(), SquaredL2Updater(), 1e-12, 8,
                   0.2, w0, 1.0, math.inf, 0.5, 0.9, True)
    for rank in range(4):
        torch.testing.assert_close(
            torch.tensor(results[rank][0], dtype=torch.float64), w_ref,
            rtol=1e-9, atol=1e-12)
    assert results[0][0] == results[3][0]
