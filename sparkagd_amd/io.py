"""Shard ingestion and persistence.

The reference's data plane starts from ``sc.parallelize(...)`` /
``LabeledPoint`` collections (``Suite.scala:51``); the equivalents here build
device-resident shards from host arrays, scipy CSR matrices, svmlight files
(via scikit-learn when available), or safetensors files written by
``save_shard``. Row sharding for multi-rank loading uses
``data.shard_range``.
"""

from __future__ import annotations

import os
from typing import Optional, Union

import numpy as np
import torch

from .data import CSRShard, DenseShard, shard_range


def dense_from_arrays(
    features,
    labels,
    device: Union[str, torch.device] = "cpu",
    dtype: Optional[torch.dtype] = None,
    rank: int = 0,
    world_size: int = 1,
) -> DenseShard:
    """Build this rank's DenseShard from host arrays (numpy or torch),
    taking the rank's balanced row range."""
    f = torch.as_tensor(features)
    l = torch.as_tensor(labels)
    lo, hi = shard_range(f.shape[0], rank, world_size)
    f = f[lo:hi]
    l = l[lo:hi]
    if dtype is not None:
        f = f.to(dtype)
    return DenseShard(f.to(device).contiguous(), l.to(device))


def csr_from_scipy(
    mat,
    labels,
    device: Union[str, torch.device] = "cpu",
    rank: int = 0,
    world_size: int = 1,
    deterministic: bool = True,
) -> CSRShard:
    """Build this rank's CSRShard from a scipy.sparse CSR matrix."""
    import scipy.sparse as sp

    if not sp.isspmatrix_csr(mat):
        mat = mat.tocsr()
    n = mat.shape[0]
    lo, hi = shard_range(n, rank, world_size)
    sub = mat[lo:hi]
    dev = torch.device(device)
    return CSRShard(
        torch.from_numpy(np.asarray(sub.indptr, dtype=np.int32)).to(dev),
        torch.from_numpy(np.asarray(sub.indices, dtype=np.int32)).to(dev),
        torch.from_numpy(np.asarray(sub.data, dtype=np.float32)).to(dev),
        torch.as_tensor(np.asarray(labels[lo:hi]), dtype=torch.float32).to(dev),
        d=mat.shape[1],
        deterministic=deterministic,
    )


def load_svmlight(
    path: str,
    d: Optional[int] = None,
    device: Union[str, torch.device] = "cpu",
    rank: int = 0,
    world_size: int = 1,
) -> CSRShard:
    """Load an svmlight/libsvm file into a CSRShard (requires scikit-learn)."""
    from sklearn.datasets import load_svmlight_file

    X, y = load_svmlight_file(path, n_features=d)
    return csr_from_scipy(X, y, device=device, rank=rank, world_size=world_size)


def save_shard(path: str, shard: Union[DenseShard, CSRShard]) -> None:
    """Persist a shard as a single safetensors file (host copy)."""
    from safetensors.torch import save_file

    if shard.kind == "dense":
        tensors = {
            "features": shard.features.cpu().contiguous(),
            "labels": shard.labels.cpu().contiguous(),
        }
        meta = {"kind": "dense"}
    else:
        tensors = {
            "rowptr": shard.rowptr.cpu().contiguous(),
            "col": shard.col.cpu().contiguous(),
            "val": shard.val.cpu().contiguous(),
            "labels": shard.labels.cpu().contiguous(),
        }
        meta = {"kind": "csr", "d": str(shard.d)}
    os.makedirs(os.path.dirname(os.path.abspath(path)) or ".", exist_ok=True)
    save_file(tensors, path, metadata=meta)


def load_shard(
    path: str,
    device: Union[str, torch.device] = "cpu",
    dtype: Optional[torch.dtype] = None,
    rank: int = 0,
    world_size: int = 1,
) -> Union[DenseShard, CSRShard]:
    """Load a shard saved by save_shard, taking this rank's row range."""
    from safetensors import safe_open
    from safetensors.torch import load_file

    with safe_open(path, framework="pt", device="cpu") as f:
        meta = f.metadata() or {}
    t = load_file(path)
    if meta.get("kind") == "dense":
        return dense_from_arrays(t["features"], t["labels"], device=device,
                                 dtype=dtype, rank=rank, world_size=world_size)
    d = int(meta["d"])
    n = t["rowptr"].numel() - 1
    lo, hi = shard_range(n, rank, world_size)
    rp = t["rowptr"][lo: hi + 1].clone()
    k_lo, k_hi = int(rp[0]), int(rp[-1])
    rp -= k_lo
    dev = torch.device(device)
    return CSRShard(rp.to(dev), t["col"][k_lo:k_hi].to(dev),
                    t["val"][k_lo:k_hi].to(dev), t["labels"][lo:hi].to(dev), d=d)
