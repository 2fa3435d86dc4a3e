"""Mixed-workload endurance: cycles of dense / CSR / multiclass / Gram solves
in ONE process, tracking device memory and per-cycle wall time — catches
leaks, allocator fragmentation and cross-path interference that single-run
benches cannot."""
import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from sparkagd_amd import (GramOperator, LogisticGradient,
                          MultinomialLogisticGradient, SimpleUpdater,
                          SquaredL2Updater, generate_dense_problem,
                          generate_multiclass_problem, run)
from sparkagd_amd.data import generate_csr_problem

dev = torch.device("cuda")
dense, _ = generate_dense_problem(16384, 1_000_000, seed=1, device=dev,
                                  dtype=torch.bfloat16)
csr, _ = generate_csr_problem(1_000_000, 10_000_000, 64, seed=2, device=dev)
multi, _ = generate_multiclass_problem(200_000, 10240, 16, seed=3, device=dev,
                                       dtype=torch.bfloat16)
torch.cuda.synchronize()
base_mem = torch.cuda.memory_allocated() / 2**30
print(f"shards resident: {base_mem:.2f} GiB allocated")

args = (1e-12, 12, 1e-3)
mems, times = [], []
t_end = time.time() + float(os.environ.get("SOAK_SECONDS", "300"))
cycle = 0
from sparkagd_amd import Communicator
gop = GramOperator(dense, Communicator())
while time.time() < t_end:
    t0 = time.perf_counter()
    w0 = torch.zeros(1_000_000, device=dev)
    run(dense, LogisticGradient(), SquaredL2Updater(), *args, w0,
        1.0, math.inf, 0.5, 0.9, True)
    run(csr, LogisticGradient(), SimpleUpdater(), 1e-12, 8, 0.0,
        torch.zeros(10_000_000, device=dev), 1.0, math.inf, 0.5, 0.9, True)
    run(multi, MultinomialLogisticGradient(16), SquaredL2Updater(), *args,
        torch.zeros(10240 * 16, device=dev), 1.0, math.inf, 0.5, 0.9, True)
    run(dense, LogisticGradient(), SquaredL2Updater(), 1e-12, 40, 1e-3, w0,
        1.0, math.inf, 0.5, 0.9, True, solver="gram", gram_op=gop)
    torch.cuda.synchronize()
    times.append(time.perf_counter() - t0)
    mems.append(torch.cuda.memory_allocated() / 2**30)
    cycle += 1
print(f"cycles: {cycle}")
print(f"cycle wall s: first {times[0]:.3f}  last {times[-1]:.3f}  "
      f"min {min(times):.3f}  max {max(times):.3f}")
print(f"allocated GiB after cycle: first {mems[0]:.3f}  last {mems[-1]:.3f}  "
      f"max {max(mems):.3f}")
assert mems[-1] <= mems[0] + 0.01, "memory growth across cycles (leak)"
assert times[-1] <= times[0] * 1.3 + 0.2, "cycle time degraded"
print("ENDURANCE OK")
