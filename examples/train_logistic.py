#!/usr/bin/env python3
"""Train logistic regression with accelerated gradient descent.

Single process:
    python examples/train_logistic.py --n 100000 --d 1000
One process per GPU (row-sharded, RCCL):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 examples/train_logistic.py --n 1000000 --d 100000
"""

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

from sparkagd_amd import (  # noqa: E402
    AcceleratedGradientDescent,
    AGDConfig,
    LogisticGradient,
    SquaredL2Updater,
    generate_dense_problem,
)
from sparkagd_amd.parallel.comm import init_from_env  # noqa: E402
from sparkagd_amd.utils.metrics import JsonlMetrics  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--n", type=int, default=100000, help="rows per rank")
    p.add_argument("--d", type=int, default=1000)
    p.add_argument("--iters", type=int, default=50)
    p.add_argument("--reg", type=float, default=0.01)
    p.add_argument("--tol", type=float, default=1e-6)
    p.add_argument("--checkpoint", type=str, default=None)
    p.add_argument("--resume", type=str, default=None)
    p.add_argument("--metrics", type=str, default=None, help="JSONL metrics path")
    args = p.parse_args()

    comm = init_from_env()
    if torch.cuda.is_available():
        device, dtype, wdtype = "cuda", torch.bfloat16, torch.float32
    else:
        device, dtype, wdtype = "cpu", torch.float64, torch.float64

    shard, w_true = generate_dense_problem(
        args.n, args.d, seed=7 + comm.rank, device=device, dtype=dtype
    )

    cfg = AGDConfig(
        convergence_tol=args.tol,
        num_iterations=args.iters,
        reg_param=args.reg,
        loss_history_mode="backtrack",
    )
    opt = AcceleratedGradientDescent(LogisticGradient(), SquaredL2Updater(), cfg, comm)
    opt.metrics = JsonlMetrics(args.metrics, rank=comm.rank)
    if args.checkpoint:
        opt.checkpoint_path = args.checkpoint
        opt.checkpoint_every = 10
    opt.resume_from = args.resume

    w0 = torch.zeros(args.d, device=shard.device, dtype=wdtype)
    w = opt.optimize(shard, w0)

    if comm.rank == 0:
        hist = opt.loss_history
        acc_w = torch.nn.functional.cosine_similarity(
            w.float().cpu(), w_true.float().cpu(), dim=0
        )
        print(f"iterations: {len(hist)}")
        print(f"loss: {hist[0]:.6f} -> {hist[-1]:.6f}")
        print(f"cosine(w, w_true): {float(acc_w):.4f}")
    opt.metrics.close()


if __name__ == "__main__":
    main()
