#!/usr/bin/env python3
"""RCCL-over-xGMI all-reduce microbenchmark (rccl-tests style).

Measures the per-iteration collective the framework actually issues — a
float32 sum all-reduce of the gradient plus a 2-element float64 all-reduce —
across message sizes, so bucket/size decisions are grounded in measurement
(SURVEY.md §5 'Distributed communication backend': 7 point-to-point xGMI
links x ~153 GB/s per GPU; small-d reductions are latency-dominated).

Launch (driver-style):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 benchmarks/allreduce_bench.py

Rank 0 prints one JSON line per size: {bytes, us_per_call, algbw_GBs, busbw_GBs}.
busbw = algbw * 2(N-1)/N (ring accounting, comparable to rccl-tests).
"""

import json
import os
import sys
import time

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from sparkagd_amd.parallel.comm import init_from_env  # noqa: E402


def main():
    comm = init_from_env()
    world = comm.world_size
    dev = torch.device("cuda", torch.cuda.current_device()) if torch.cuda.is_available() else torch.device("cpu")

    sizes = [1 << k for k in range(10, 31)]  # 1 KiB .. 1 GiB (element counts: /4)
    results = []
    for nbytes in sizes:
        n = nbytes // 4
        t = torch.ones(n, dtype=torch.float32, device=dev)
        # warmup
        for _ in range(5):
            comm.allreduce_(t)
        if dev.type == "cuda":
            torch.cuda.synchronize()
        comm.barrier()
        iters = max(3, min(50, (1 << 28) // max(nbytes, 1)))
        t0 = time.perf_counter()
        for _ in range(iters):
            comm.allreduce_(t)
        if dev.type == "cuda":
            torch.cuda.synchronize()
        el = torch.tensor([time.perf_counter() - t0], dtype=torch.float64, device=dev)
        if world > 1:
            dist.all_reduce(el, op=dist.ReduceOp.MAX)
        sec = float(el[0]) / iters
        algbw = nbytes / sec / 1e9
        busbw = algbw * 2 * (world - 1) / world if world > 1 else 0.0
        row = {"bytes": nbytes, "world": world, "iters": iters,
               "us_per_call": round(sec * 1e6, 2),
               "algbw_GBs": round(algbw, 2), "busbw_GBs": round(busbw, 2)}
        results.append(row)
        if comm.rank == 0:
            print(json.dumps(row), flush=True)
    return 0


if __name__ == "__main__":
    sys.exit(main())
