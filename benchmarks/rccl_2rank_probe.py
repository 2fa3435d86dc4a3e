"""Probe: can RCCL run 2 ranks on ONE MI355X (both ranks bound to cuda:0)?

VERDICT r01's fallback ask when no multi-GPU lease exists: "validate
init_from_env + nccl-backend collectives with 2 ranks on one device (or
document precisely why that's unsupported)". NCCL/RCCL communicators
historically reject duplicate devices; this probe records the actual RCCL
behavior on ROCm 7.2 either way, exercising exactly the collectives the
optimizer hot path issues per evaluation:

  * all_reduce SUM fp32 [d]      (the gradient reduction, C1)
  * all_reduce SUM fp64 [2]      (the (loss, count) pair)
  * all_reduce MAX fp64 [1]      (bench.py's elapsed-max)
  * barrier(device_ids=[...])    (the timing bracket)
  * broadcast fp32               (init weights / divergence checks)

Launch:  python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
             --master-addr 127.0.0.1 benchmarks/rccl_2rank_probe.py
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402
import torch.distributed as dist  # noqa: E402

from sparkagd_amd.parallel.comm import init_from_env  # noqa: E402


def main() -> int:
    comm = init_from_env()
    rank, world = comm.rank, comm.world_size
    dev = torch.device("cuda", torch.cuda.current_device())
    print(f"[rank {rank}] world={world} device={dev} "
          f"backend={dist.get_backend() if dist.is_initialized() else 'none'}",
          flush=True)

    g = torch.full((1_000_000,), float(rank + 1), device=dev)
    comm.allreduce_(g)
    expect = world * (world + 1) / 2.0
    ok_grad = bool(torch.all(g == expect))

    lc = torch.tensor([1.5 * (rank + 1), 100.0], dtype=torch.float64, device=dev)
    comm.allreduce_(lc)
    ok_lc = abs(float(lc[0]) - 1.5 * expect) < 1e-12 and float(lc[1]) == 100.0 * world

    el = torch.tensor([float(rank)], dtype=torch.float64, device=dev)
    dist.all_reduce(el, op=dist.ReduceOp.MAX)
    ok_max = float(el[0]) == float(world - 1)

    comm.barrier()

    b = torch.full((8,), float(rank), device=dev)
    comm.broadcast_(b, src=0)
    ok_bc = bool(torch.all(b == 0.0))

    ok_rep = comm.check_replicated(torch.ones(16, device=dev) * 3.0)

    print(f"[rank {rank}] grad_allreduce={ok_grad} loss_count={ok_lc} "
          f"max={ok_max} broadcast={ok_bc} replicated={ok_rep}", flush=True)
    ok = all([ok_grad, ok_lc, ok_max, ok_bc, ok_rep])
    comm.barrier()
    if rank == 0:
        print("RCCL_PROBE_" + ("OK" if ok else "FAIL"), flush=True)
    dist.destroy_process_group()
    return 0 if ok else 1


if __name__ == "__main__":
    sys.exit(main())
