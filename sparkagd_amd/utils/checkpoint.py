"""Weight + momentum checkpointing.

The reference has NO checkpoint capability (``run()`` holds all state in
locals, ``AcceleratedGradientDescent.scala:224-235``, and returns only
(x, lossHistory)). This module adds the weight+momentum checkpoint format the
north star mandates: the AT momentum is carried by the (x, z) pair (SURVEY.md
§2.5 — y_k = x_k + beta (x_k - x_{k-1}) is the equivalent Nesterov form), so a
checkpoint is exactly {x, z} plus the scalar loop state
{theta, L, iter, backtrack_simple} and the loss history.

Format: single safetensors file written by rank 0, bitwise round-trip (tensors
stored in their exact dtype; scalars in JSON string metadata).
"""

from __future__ import annotations

import json
import math
import os
import tempfile
from typing import Any, Dict, List, Optional

import torch
from safetensors.torch import load_file, save_file

FORMAT_VERSION = 1


def save_checkpoint(
    path: str,
    *,
    x: torch.Tensor,
    z: torch.Tensor,
    theta: float,
    L: float,
    iter: int,  # noqa: A002
    backtrack_simple: bool,
    loss_history: List[float],
    extra: Optional[Dict[str, Any]] = None,
) -> None:
    meta = {
        "format_version": str(FORMAT_VERSION),
        "theta": "inf" if math.isinf(theta) else repr(float(theta)),
        "L": repr(float(L)),
        "iter": str(int(iter)),
        "backtrack_simple": "1" if backtrack_simple else "0",
        "loss_history": json.dumps(loss_history),
    }
    if extra:
        meta["extra"] = json.dumps(extra)
    # clone(): safetensors rejects aliased/shared storage (z is x after a restart)
    tensors = {"x": x.detach().to("cpu").contiguous().clone(), "z": z.detach().to("cpu").contiguous().clone()}
    # Atomic write: temp file in the same directory, then rename.
    d = os.path.dirname(os.path.abspath(path)) or "."
    os.makedirs(d, exist_ok=True)
    fd, tmp = tempfile.mkstemp(dir=d, suffix=".tmp")
    os.close(fd)
    try:
        save_file(tensors, tmp, metadata=meta)
        os.replace(tmp, path)
    finally:
        if os.path.exists(tmp):
            os.unlink(tmp)


def load_checkpoint(
    path: str,
    device: Optional[torch.device] = None,
    dtype: Optional[torch.dtype] = None,
) -> Dict[str, Any]:
    from safetensors import safe_open

    with safe_open(path, framework="pt", device="cpu") as f:
        meta = f.metadata() or {}
    tensors = load_file(path)
    if int(meta.get("format_version", "0")) != FORMAT_VERSION:
        raise ValueError(f"unsupported checkpoint format_version {meta.get('format_version')}")
    x, z = tensors["x"], tensors["z"]
    if device is not None:
        x, z = x.to(device), z.to(device)
    if dtype is not None:
        x, z = x.to(dtype), z.to(dtype)
    theta_s = meta["theta"]
    out: Dict[str, Any] = {
        "x": x,
        "z": z,
        "theta": math.inf if theta_s == "inf" else float(theta_s),
        "L": float(meta["L"]),
        "iter": int(meta["iter"]),
        "backtrack_simple": meta["backtrack_simple"] == "1",
        "loss_history": json.loads(meta["loss_history"]),
    }
    if "extra" in meta:
        out["extra"] = json.loads(meta["extra"])
    return out
