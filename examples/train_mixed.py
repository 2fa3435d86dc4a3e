#!/usr/bin/env python3
"""Heterogeneous dense + CSR-sparse training with a MixedShard.

MLlib accepts dense or sparse vectors per example within one RDD (the
reference invokes Gradient.compute on whatever representation each example
carries, AGD.scala:198). Here examples are grouped by representation into a
dense block and a sparse block evaluated by their own fused kernels — this
example builds a problem whose first half is dense measurements and second
half is sparse (e.g. one-hot-ish) features over the SAME feature space.
"""

import argparse
import math
import os
import sys

import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

from sparkagd_amd import (  # noqa: E402
    LogisticGradient,
    MixedShard,
    SquaredL2Updater,
    run,
)
from sparkagd_amd.data import (  # noqa: E402
    DenseShard,
    generate_csr_problem,
    generate_dense_problem,
)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--n-dense", type=int, default=20000)
    p.add_argument("--n-sparse", type=int, default=80000)
    p.add_argument("--d", type=int, default=5000)
    p.add_argument("--nnz-per-row", type=int, default=24)
    p.add_argument("--reg", type=float, default=1e-3)
    p.add_argument("--iters", type=int, default=30)
    args = p.parse_args()

    dev = "cuda:0" if torch.cuda.is_available() else "cpu"
    dtype = torch.bfloat16 if dev != "cpu" else torch.float32

    dense, _ = generate_dense_problem(args.n_dense, args.d, seed=1,
                                      device=dev, dtype=dtype)
    sparse, _ = generate_csr_problem(args.n_sparse, args.d, args.nnz_per_row,
                                     seed=2, device=dev)
    # CSR kernels are f32; a mixed shard can hold bf16 dense + f32 sparse
    # blocks, but the weight vector is shared, so use f32 weights
    if dtype == torch.bfloat16:
        dense = DenseShard(dense.features.to(torch.float32), dense.labels)
    shard = MixedShard([dense, sparse])
    print(f"mixed shard: {shard.n} rows ({dense.n} dense + {sparse.n} CSR), "
          f"d={shard.d}, {shard.nbytes / 2**20:.1f} MiB on {dev}")

    w0 = torch.zeros(args.d, device=dev,
                     dtype=torch.float32 if dev != "cpu" else torch.float64)
    w, hist = run(shard, LogisticGradient(), SquaredL2Updater(), 1e-10,
                  args.iters, args.reg, w0, 1.0, math.inf, 0.5, 0.9, True,
                  loss_history_mode="backtrack")
    print(f"{len(hist)} iterations, loss {hist[0]:.4f} -> {hist[-1]:.4f}")
    assert hist[-1] < hist[0]


if __name__ == "__main__":
    main()
