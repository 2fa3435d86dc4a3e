"""Checkpoint overhead at bench scale: same 200-iteration run with and
without periodic checkpointing (every 10 iterations, rank-0 safetensors)."""
import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from sparkagd_amd import (LogisticGradient, SquaredL2Updater,
                          generate_dense_problem, run)

dev = torch.device("cuda" if torch.cuda.is_available() else "cpu")
n, d = (16384, 1_000_000) if dev.type == "cuda" else (2048, 256)
shard, _ = generate_dense_problem(n, d, seed=6, device=dev,
                                  dtype=torch.bfloat16 if dev.type == "cuda" else torch.float64)
w0 = torch.zeros(d, device=dev,
                 dtype=torch.float32 if dev.type == "cuda" else torch.float64)
args = (LogisticGradient(), SquaredL2Updater(), 0.0, 200, 1e-3, w0,
        1.0, math.inf, 0.5, 0.9, True)


def timed(**kw):
    if dev.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    _, h = run(shard, *args, loss_history_mode="backtrack", **kw)
    if dev.type == "cuda":
        torch.cuda.synchronize()
    return (time.perf_counter() - t0) / len(h) * 1e3, len(h)


base_ms, it0 = timed()
ck_ms, it1 = timed(checkpoint_path="gpurun_out/ck_overhead.safetensors",
                   checkpoint_every=10)
print(f"no checkpoint:        {base_ms:.3f} ms/step ({it0} iters)")
print(f"checkpoint every 10:  {ck_ms:.3f} ms/step ({it1} iters)  "
      f"overhead {100 * (ck_ms - base_ms) / base_ms:+.1f}%")
