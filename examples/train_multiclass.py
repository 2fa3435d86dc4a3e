"""Multinomial softmax regression end-to-end: train, evaluate, persist.

Runs on CPU (oracle kernels) or on an MI355X (hipBLASLt GEMM margins/grad)
with the same code. On GPU use a bf16 shard for the matrix-core path.
"""

import sys
import os
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from sparkagd_amd import (SoftmaxRegressionWithAGD, MultinomialModel,
                          generate_multiclass_problem, evaluation)

dev = "cuda" if torch.cuda.is_available() else "cpu"
dtype = torch.bfloat16 if dev == "cuda" else torch.float64
K = 8

shard, _ = generate_multiclass_problem(20000, 256, K, seed=7, device=dev,
                                       dtype=dtype, label_noise=0.2)
model = SoftmaxRegressionWithAGD.train(shard, num_classes=K,
                                       num_iterations=60, reg_param=1e-3)

pred = model.predict(shard.features)
acc = evaluation.accuracy(pred, shard.labels)
cm = evaluation.confusion_matrix(pred, shard.labels, K)
print(f"iterations: {len(model.loss_history)}")
print(f"loss: {model.loss_history[0]:.4f} -> {model.loss_history[-1]:.4f}")
print(f"train accuracy: {acc:.3f} (chance {1.0 / K:.3f})")
print(f"confusion diagonal fraction: {float(cm.diag().sum()) / float(cm.sum()):.3f}")

with tempfile.TemporaryDirectory() as d:
    path = os.path.join(d, "softmax.safetensors")
    model.save(path)
    reloaded = MultinomialModel.load(path, device=dev)
    assert torch.equal(model.predict(shard.features).cpu(),
                       reloaded.predict(shard.features).cpu())
print("model save/load round-trip: ok")
