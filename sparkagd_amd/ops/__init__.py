"""Compute-op dispatch layer.

Exactly two execution paths, chosen by tensor placement:

* **CUDA (= ROCm/HIP) tensors** -> the hand-written CDNA4 HIP kernels in
  ``sparkagd_amd/csrc`` via the ctypes binding in ``hiplib.py``. If the
  extension is missing on a GPU machine this raises — there is deliberately
  NO silent eager/PyTorch fallback on GPU (the HIP path must be the path
  that runs).
* **CPU tensors** -> the plain-PyTorch oracles in ``reference.py`` (test/CI
  path; also the ground truth the kernels are validated against).

No Triton, no multi-backend dispatch beyond this placement switch.
"""

from __future__ import annotations

import os
from typing import Optional, Tuple

import torch

from . import reference
from .reference import (
    LOSS_LOGISTIC,
    LOSS_LEAST_SQUARES,
    LOSS_HINGE,
    LOSS_SMOOTH_HINGE,
    PROX_SIMPLE,
    PROX_L1,
    PROX_SQUARED_L2,
    PROX_ELASTIC_NET,
)

_hip = None
_hip_err: Optional[str] = None


def _get_hip():
    """Load the HIP extension binding, raising loudly on GPU if unavailable."""
    global _hip, _hip_err
    if _hip is None and _hip_err is None:
        try:
            from . import hiplib

            hiplib.load()
            _hip = hiplib
        except Exception as e:  # noqa: BLE001 - record and re-raise at use site
            _hip_err = f"{type(e).__name__}: {e}"
    if _hip is None:
        raise RuntimeError(
            "sparkagd_amd HIP extension (libagd_hip.so) is required for GPU "
            "tensors but could not be loaded. Build it with "
            "`python -c 'import __graft_entry__; __graft_entry__.build()'` "
            f"from the repo root. Underlying error: {_hip_err}"
        )
    return _hip


def hip_available() -> bool:
    try:
        _get_hip()
        return True
    except RuntimeError:
        return False


def _use_hip(t: torch.Tensor) -> bool:
    if t.is_cuda:
        if os.environ.get("SPARKAGD_FORCE_REFERENCE") == "1":
            return False
        return True
    return False


def dense_eval(
    features: torch.Tensor,
    labels: torch.Tensor,
    w: torch.Tensor,
    loss_type: int,
    mask: Optional[torch.Tensor] = None,
    need_grad: bool = True,
    sample_weight: Optional[torch.Tensor] = None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    if _use_hip(features):
        return _get_hip().dense_eval(features, labels, w, loss_type, mask,
                                     need_grad, sample_weight)
    return reference.dense_eval(features, labels, w, loss_type, mask,
                                need_grad, sample_weight)


def csr_eval(
    rowptr: torch.Tensor,
    col: torch.Tensor,
    val: torch.Tensor,
    labels: torch.Tensor,
    w: torch.Tensor,
    loss_type: int,
    mask: Optional[torch.Tensor] = None,
    d: Optional[int] = None,
    csc=None,
    need_grad: bool = True,
    sample_weight: Optional[torch.Tensor] = None,
    csc_heavy=None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    if _use_hip(val):
        return _get_hip().csr_eval(rowptr, col, val, labels, w, loss_type, mask,
                                   d, csc, need_grad, sample_weight, csc_heavy)
    # the torch reference (sparse_csr @ / .t() @) is already deterministic
    return reference.csr_eval(rowptr, col, val, labels, w, loss_type, mask, d,
                              need_grad, sample_weight)


def dense_margins(features: torch.Tensor, v: torch.Tensor) -> torch.Tensor:
    if _use_hip(features):
        return _get_hip().dense_margins(features, v)
    return reference.dense_margins(features, v)


def dense_eval_from_margins(features, margins, labels, loss_type, mask=None,
                            need_grad=True, sample_weight=None):
    if _use_hip(features):
        return _get_hip().dense_eval_from_margins(features, margins, labels,
                                                  loss_type, mask, need_grad,
                                                  sample_weight)
    return reference.dense_eval_from_margins(features, margins, labels,
                                             loss_type, mask, need_grad,
                                             sample_weight)


def dense_multiplier_loss(features, margins, labels, loss_type, mask=None,
                          sample_weight=None):
    if _use_hip(features):
        return _get_hip().dense_multiplier_loss(features, margins, labels,
                                                loss_type, mask, sample_weight)
    return reference.dense_multiplier_loss(features, margins, labels,
                                           loss_type, mask, sample_weight)


def dense_grad_from_mult(features, mult):
    if _use_hip(features):
        return _get_hip().dense_grad_from_mult(features, mult)
    return reference.dense_grad_from_mult(features, mult)


def csr_margins(rowptr, col, val, v):
    if _use_hip(val):
        return _get_hip().csr_margins(rowptr, col, val, v)
    return reference.csr_margins(rowptr, col, val, v)


def csr_eval_from_margins(rowptr, col, val, margins, labels, loss_type,
                          mask=None, d=None, csc=None, need_grad=True,
                          sample_weight=None, csc_heavy=None):
    if _use_hip(val):
        return _get_hip().csr_eval_from_margins(rowptr, col, val, margins, labels,
                                                loss_type, mask, d, csc,
                                                need_grad, sample_weight,
                                                csc_heavy)
    return reference.csr_eval_from_margins(rowptr, col, val, margins, labels,
                                           loss_type, mask, d, csc, need_grad,
                                           sample_weight)


def at_margin_update(zm_old, xm_old, gm, pz: float, pg: float, theta: float):
    """Fused AT margin update: zm' = pz*zm + pg*gm; xm' = (1-th)*xm + th*zm'
    — one pass on GPU, the axpby composition on CPU (identical algebra)."""
    if _use_hip(zm_old):
        return _get_hip().at_margin_update(zm_old, xm_old, gm, pz, pg, theta)
    zm_new = pz * zm_old + pg * gm
    xm_new = (1.0 - theta) * xm_old + theta * zm_new
    return zm_new, xm_new


def prox(
    kind: int, w: torch.Tensor, g: torch.Tensor, step: float, lam: float,
    lam2: float = 0.0,
) -> Tuple[torch.Tensor, torch.Tensor]:
    if _use_hip(w):
        return _get_hip().prox(kind, w, g, step, lam, lam2)
    return reference.prox(kind, w, g, step, lam, lam2)


def axpby(
    a: float, x: torch.Tensor, b: float, y: torch.Tensor, out: Optional[torch.Tensor] = None
) -> torch.Tensor:
    if _use_hip(x):
        return _get_hip().axpby(a, x, b, y, out)
    return reference.axpby(a, x, b, y, out)


def fused_scalars(
    x: torch.Tensor, y: torch.Tensor, g_y: torch.Tensor, x_old: torch.Tensor
) -> torch.Tensor:
    if _use_hip(x):
        return _get_hip().fused_scalars(x, y, g_y, x_old)
    return reference.fused_scalars(x, y, g_y, x_old)


def dot_diff(
    x: torch.Tensor, y: torch.Tensor, g_x: torch.Tensor, g_y: torch.Tensor
) -> torch.Tensor:
    if _use_hip(x):
        return _get_hip().dot_diff(x, y, g_x, g_y)
    return reference.dot_diff(x, y, g_x, g_y)


__all__ = [
    "LOSS_LOGISTIC",
    "LOSS_LEAST_SQUARES",
    "LOSS_HINGE",
    "LOSS_SMOOTH_HINGE",
    "PROX_SIMPLE",
    "PROX_L1",
    "PROX_SQUARED_L2",
    "PROX_ELASTIC_NET",
    "dense_eval",
    "csr_eval",
    "dense_margins",
    "dense_multiplier_loss",
    "dense_grad_from_mult",
    "dense_eval_from_margins",
    "csr_margins",
    "csr_eval_from_margins",
    "prox",
    "axpby",
    "fused_scalars",
    "dot_diff",
    "hip_available",
    "reference",
]
