"""iters-to-ε at bench scale: AGD vs mini-batch GD (the reference's core
claim — Suite.scala:60-90 shows AGD(10) ~ GD(50) at toy scale — measured on
the MI355X headline config)."""
import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from sparkagd_amd import (LogisticGradient, SquaredL2Updater,
                          generate_dense_problem, run, run_mini_batch)

dev = torch.device("cuda" if torch.cuda.is_available() else "cpu")
n, d = (16384, 1_000_000) if dev.type == "cuda" else (2048, 256)
shard, _ = generate_dense_problem(n, d, seed=5, device=dev,
                                  dtype=torch.bfloat16 if dev.type == "cuda" else torch.float64)
wdt = torch.float32 if dev.type == "cuda" else torch.float64
w0 = torch.zeros(d, device=dev, dtype=wdt)
REG = 1e-3

t0 = time.perf_counter()
_, h_agd = run(shard, LogisticGradient(), SquaredL2Updater(), 0.0, 100, REG,
               w0, 1.0, math.inf, 0.5, 0.9, True, loss_history_mode="backtrack")
if dev.type == "cuda":
    torch.cuda.synchronize()
t_agd = time.perf_counter() - t0

t0 = time.perf_counter()
_, h_gd = run_mini_batch(shard, LogisticGradient(), SquaredL2Updater(), 1.0,
                         400, REG, 1.0, w0, step_schedule="constant")
if dev.type == "cuda":
    torch.cuda.synchronize()
t_gd = time.perf_counter() - t0

from sparkagd_amd.utils.metrics import iters_to_eps  # noqa: E402 (shared definition with bench.py)

lstar = min(min(h_agd), min(h_gd))


print(f"config: dense logistic d={d} n={n} {'bf16' if dev.type=='cuda' else 'f64'}, L2 {REG}")
print(f"AGD: {len(h_agd)} iters in {t_agd:.2f}s ({1e3*t_agd/len(h_agd):.1f} ms/it), final {h_agd[-1]:.8f}")
print(f"GD : {len(h_gd)} iters in {t_gd:.2f}s ({1e3*t_gd/len(h_gd):.1f} ms/it), final {h_gd[-1]:.8f}")
print(f"L* = {lstar:.8f}")
print(f"{'eps':>8} {'AGD iters':>10} {'GD iters':>10}")
for eps in (0.5, 0.2, 0.1, 0.05, 0.02, 0.01, 0.001):
    ta = iters_to_eps(h_agd, eps, loss_star=lstar)
    tg = iters_to_eps(h_gd, eps, loss_star=lstar)
    print(f"{eps:>8} {str(ta):>10} {str(tg):>10}")
