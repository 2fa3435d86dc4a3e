#!/usr/bin/env python3
"""Linear SVM (hinge loss, L2) on a CSR-sparse shard, with mini-batch SGD
under the strong-convexity step schedule and an AGD run for comparison."""

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

from sparkagd_amd import (  # noqa: E402
    HingeGradient,
    SquaredL2Updater,
    run,
    run_mini_batch,
)
from sparkagd_amd.data import generate_csr_problem  # noqa: E402
from sparkagd_amd import ops  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--n", type=int, default=200000)
    p.add_argument("--d", type=int, default=100000)
    p.add_argument("--nnz-per-row", type=int, default=32)
    p.add_argument("--reg", type=float, default=1e-4)
    p.add_argument("--iters", type=int, default=40)
    args = p.parse_args()

    device = "cuda" if torch.cuda.is_available() else "cpu"
    shard, _ = generate_csr_problem(
        args.n, args.d, args.nnz_per_row, seed=3, loss_type=ops.LOSS_HINGE,
        device=device,
    )
    w0 = torch.zeros(args.d, device=shard.device, dtype=torch.float32 if device == "cuda" else torch.float64)

    import math

    w_agd, hist_agd = run(
        shard, HingeGradient(), SquaredL2Updater(), 1e-8, args.iters, args.reg,
        w0, 1.0, math.inf, 0.5, 0.9, True, loss_history_mode="backtrack",
    )
    w_sgd, hist_sgd = run_mini_batch(
        shard, HingeGradient(), SquaredL2Updater(), 1.0, args.iters, args.reg,
        0.25, w0, step_schedule="linear",
    )
    print(f"AGD   : {len(hist_agd)} iters, loss {hist_agd[-1]:.6f}")
    print(f"SGD   : {len(hist_sgd)} iters, loss {hist_sgd[-1]:.6f} "
          f"(1/4 mini-batches, 1/(lambda*t) schedule)")


if __name__ == "__main__":
    main()
