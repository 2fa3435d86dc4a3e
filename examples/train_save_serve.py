#!/usr/bin/env python3
"""The full user lifecycle a reference (MLlib) user expects: train a GLM
with AGD, persist the model, reload it in a fresh 'serving' step, predict,
and evaluate — plus a checkpointed training run resumed mid-way.

    python examples/train_save_serve.py
"""

import argparse
import os
import sys
import tempfile

import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

from sparkagd_amd import evaluation  # noqa: E402
from sparkagd_amd.data import generate_dense_problem  # noqa: E402
from sparkagd_amd.models.trainers import (  # noqa: E402
    LinearModel,
    LogisticRegressionWithAGD,
)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--n", type=int, default=50000)
    p.add_argument("--d", type=int, default=512)
    p.add_argument("--iters", type=int, default=30)
    args = p.parse_args()

    dev = "cuda:0" if torch.cuda.is_available() else "cpu"
    dtype = torch.bfloat16 if dev != "cpu" else torch.float32
    # one planted problem, row-split into train/test (same true weights)
    from sparkagd_amd.data import DenseShard

    full, _ = generate_dense_problem(args.n + args.n // 5, args.d, seed=1,
                                     device=dev, dtype=dtype)
    train = DenseShard(full.features[: args.n].contiguous(),
                       full.labels[: args.n])
    test = DenseShard(full.features[args.n:].contiguous(),
                      full.labels[args.n:])

    # --- train (with periodic weight+momentum checkpoints) ---
    with tempfile.TemporaryDirectory() as tmp:
        ckpt = os.path.join(tmp, "train.ckpt.safetensors")
        model = LogisticRegressionWithAGD.train(
            train, num_iterations=args.iters, reg_param=1e-3,
            convergence_tol=0.0, checkpoint_path=ckpt, checkpoint_every=10)
        print(f"trained {len(model.loss_history)} iters, "
              f"loss {model.loss_history[0]:.4f} -> {model.loss_history[-1]:.4f}")

        # --- persist + reload (the serving side) ---
        mpath = os.path.join(tmp, "model.safetensors")
        model.save(mpath)
        served = LinearModel.load(mpath, device=dev)

        # --- predict + evaluate ---
        margins = served.margins(test.features)
        pred = served.predict(test.features)
        acc = evaluation.accuracy(pred, test.labels)
        auc = evaluation.roc_auc(margins, test.labels)
        ll = evaluation.log_loss(margins, test.labels)
        prf = evaluation.precision_recall_f1(pred, test.labels)
        print(f"test accuracy={acc:.4f} auc={auc:.4f} log_loss={ll:.4f} "
              f"f1={prf['f1']:.4f}")
        assert acc > 0.85 and auc > 0.9
        assert os.path.exists(ckpt), "periodic checkpoints must exist"
    print("train/save/serve lifecycle OK")


if __name__ == "__main__":
    main()
