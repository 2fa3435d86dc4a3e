"""Plain-PyTorch reference implementations of every compute primitive.

These are the *oracles*: the CPU execution path for tests/CI (this container has
no GPU) and the ground truth every HIP kernel is validated against
(``tests/test_gpu_kernels.py``). They intentionally use nothing but dense torch
ops, mirroring the role Breeze/netlib BLAS plays under the reference
(``AcceleratedGradientDescent.scala:23,196-207``) and the per-example
``Gradient.compute`` semantics of MLlib 1.3 (usage site ``AGD.scala:198``).

Loss/multiplier conventions (z = <w, x_i> is the *dot product*, NOT MLlib's
negated "margin"; algebra normalized so all three losses share one structure):

  logistic     : mult = sigmoid(z) - y            loss = y>0 ? softplus(-z) : softplus(z)
  least squares: mult = 2 (z - y)                 loss = (z - y)^2
  hinge        : s = 2y - 1
                 mult = (s*z < 1) ? -s : 0        loss = max(0, 1 - s*z)

  grad_sum = A^T mult ; loss_sum = sum loss ; count = #rows (masked rows excluded)

These are algebraically identical to MLlib 1.3's LogisticGradient /
LeastSquaresGradient / HingeGradient (margin = -z substitution).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

LOSS_LOGISTIC = 0
LOSS_LEAST_SQUARES = 1
LOSS_HINGE = 2
LOSS_SMOOTH_HINGE = 3

PROX_SIMPLE = 0
PROX_L1 = 1
PROX_SQUARED_L2 = 2
PROX_ELASTIC_NET = 3


def _multiplier_and_loss(
    z: torch.Tensor, labels: torch.Tensor, loss_type: int
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Elementwise multiplier m_i and per-example loss l_i from dots z_i."""
    z = z.to(torch.float32) if z.dtype == torch.bfloat16 else z
    y = labels.to(z.dtype)
    if loss_type == LOSS_LOGISTIC:
        mult = torch.sigmoid(z) - y
        loss = torch.where(y > 0, torch.nn.functional.softplus(-z), torch.nn.functional.softplus(z))
    elif loss_type == LOSS_LEAST_SQUARES:
        diff = z - y
        mult = 2.0 * diff
        loss = diff * diff
    elif loss_type == LOSS_HINGE:
        s = 2.0 * y - 1.0
        viol = s * z < 1.0
        mult = torch.where(viol, -s, torch.zeros_like(z))
        loss = torch.clamp(1.0 - s * z, min=0.0)
    elif loss_type == LOSS_SMOOTH_HINGE:
        # Rennie's quadratically smoothed hinge: differentiable, so the
        # accelerated method's smoothness assumptions hold (plain hinge is
        # nonsmooth and is kept for MLlib parity).
        s = 2.0 * y - 1.0
        sz = s * z
        mult = torch.where(sz >= 1.0, torch.zeros_like(z),
                           torch.where(sz > 0.0, -s * (1.0 - sz), -s))
        loss = torch.where(sz >= 1.0, torch.zeros_like(z),
                           torch.where(sz > 0.0, 0.5 * (1.0 - sz) ** 2, 0.5 - sz))
    else:
        raise ValueError(f"unknown loss_type {loss_type}")
    return mult, loss


def _apply_mask_weight(mult, loss, mask, sample_weight, n, device):
    """Apply the mini-batch mask and/or per-example weights; count is the
    number of unmasked examples (sum of their weights when weighted)."""
    if mask is not None:
        m = mask.to(mult.dtype)
        mult = mult * m
        loss = loss * m
    if sample_weight is not None:
        sw = sample_weight.to(mult.dtype)
        mult = mult * sw
        loss = loss * sw
        if mask is not None:
            count = (mask.to(torch.float64) * sample_weight.to(torch.float64)).sum()
        else:
            count = sample_weight.to(torch.float64).sum()
    elif mask is not None:
        count = mask.sum().to(torch.float64)
    else:
        count = torch.tensor(float(n), dtype=torch.float64, device=device)
    return mult, loss, count


def dense_eval(
    features: torch.Tensor,
    labels: torch.Tensor,
    w: torch.Tensor,
    loss_type: int,
    mask: Optional[torch.Tensor] = None,
    need_grad: bool = True,
    sample_weight: Optional[torch.Tensor] = None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Batched loss/gradient over a dense shard.

    Equivalent of one partition's seqOp fold in the reference's
    ``applySmooth`` (``AGD.scala:196-200``), batched: one GEMV pair instead of
    n per-example axpy/dot calls.

    Returns ``(grad_sum, loss_count)`` where grad_sum has w's dtype and shape
    [d], and loss_count is float64 [2] = (sum of losses, number of examples).
    """
    acc_dtype = torch.float32 if features.dtype in (torch.bfloat16, torch.float16, torch.float8_e4m3fn) else features.dtype
    wa = w.to(acc_dtype)
    z = (features.to(acc_dtype) @ wa) if features.dtype in (torch.bfloat16, torch.float16, torch.float8_e4m3fn) else (features @ wa)
    mult, loss = _multiplier_and_loss(z, labels, loss_type)
    mult, loss, count = _apply_mask_weight(mult, loss, mask, sample_weight,
                                           features.shape[0], features.device)
    loss_count = torch.stack([loss.to(torch.float64).sum(), count])
    if not need_grad:
        return None, loss_count
    grad_sum = (features.to(acc_dtype).T @ mult.to(acc_dtype)).to(w.dtype)
    return grad_sum, loss_count


def csr_eval(
    rowptr: torch.Tensor,
    col: torch.Tensor,
    val: torch.Tensor,
    labels: torch.Tensor,
    w: torch.Tensor,
    loss_type: int,
    mask: Optional[torch.Tensor] = None,
    d: Optional[int] = None,
    need_grad: bool = True,
    sample_weight: Optional[torch.Tensor] = None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Batched loss/gradient over a CSR shard (MLlib sparse-Vector path analog)."""
    n = rowptr.numel() - 1
    d = d if d is not None else w.numel()
    acc_dtype = torch.float32 if val.dtype in (torch.bfloat16, torch.float16) else val.dtype
    a = torch.sparse_csr_tensor(
        rowptr.to(torch.int64), col.to(torch.int64), val.to(acc_dtype), size=(n, d)
    )
    z = a @ w.to(acc_dtype)
    mult, loss = _multiplier_and_loss(z, labels, loss_type)
    mult, loss, count = _apply_mask_weight(mult, loss, mask, sample_weight, n, val.device)
    loss_count = torch.stack([loss.to(torch.float64).sum(), count])
    if not need_grad:
        return None, loss_count
    grad_sum = (a.t() @ mult.to(acc_dtype)).to(w.dtype)
    return grad_sum, loss_count


def dense_margins(features: torch.Tensor, v: torch.Tensor) -> torch.Tensor:
    """margins = A @ v (margin-state tracking support)."""
    acc = torch.float32 if features.dtype in (torch.bfloat16, torch.float16, torch.float8_e4m3fn) else features.dtype
    return features.to(acc) @ v.to(acc)


def dense_eval_from_margins(
    features: torch.Tensor,
    margins: torch.Tensor,
    labels: torch.Tensor,
    loss_type: int,
    mask: Optional[torch.Tensor] = None,
    need_grad: bool = True,
    sample_weight: Optional[torch.Tensor] = None,
) -> Tuple[Optional[torch.Tensor], torch.Tensor]:
    acc = torch.float32 if features.dtype in (torch.bfloat16, torch.float16, torch.float8_e4m3fn) else features.dtype
    mult, loss = _multiplier_and_loss(margins, labels, loss_type)
    mult, loss, count = _apply_mask_weight(mult, loss, mask, sample_weight,
                                           features.shape[0], features.device)
    loss_count = torch.stack([loss.to(torch.float64).sum(), count])
    if not need_grad:
        return None, loss_count
    grad_sum = (features.to(acc).T @ mult.to(acc))
    return grad_sum, loss_count


def dense_multiplier_loss(
    features: torch.Tensor,
    margins: torch.Tensor,
    labels: torch.Tensor,
    loss_type: int,
    mask: Optional[torch.Tensor] = None,
    sample_weight: Optional[torch.Tensor] = None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    mult, loss = _multiplier_and_loss(margins, labels, loss_type)
    mult, loss, count = _apply_mask_weight(mult, loss, mask, sample_weight,
                                           features.shape[0], features.device)
    return mult, torch.stack([loss.to(torch.float64).sum(), count])


def dense_grad_from_mult(features: torch.Tensor, mult: torch.Tensor) -> torch.Tensor:
    acc = torch.float32 if features.dtype in (torch.bfloat16, torch.float16, torch.float8_e4m3fn) else features.dtype
    return features.to(acc).T @ mult.to(acc)


def csr_margins(rowptr, col, val, v: torch.Tensor, n: Optional[int] = None) -> torch.Tensor:
    n = rowptr.numel() - 1
    a = torch.sparse_csr_tensor(rowptr.to(torch.int64), col.to(torch.int64),
                                val.to(torch.float32), size=(n, v.numel()))
    return a @ v.to(torch.float32)


def csr_eval_from_margins(rowptr, col, val, margins, labels, loss_type,
                          mask=None, d=None, csc=None, need_grad=True,
                          sample_weight=None):
    n = rowptr.numel() - 1
    mult, loss = _multiplier_and_loss(margins, labels, loss_type)
    mult, loss, count = _apply_mask_weight(mult, loss, mask, sample_weight, n, val.device)
    loss_count = torch.stack([loss.to(torch.float64).sum(), count])
    if not need_grad:
        return None, loss_count
    a = torch.sparse_csr_tensor(rowptr.to(torch.int64), col.to(torch.int64),
                                val.to(torch.float32), size=(n, d))
    grad_sum = a.t() @ mult.to(torch.float32)
    return grad_sum, loss_count


def prox(
    kind: int,
    w: torch.Tensor,
    g: torch.Tensor,
    step: float,
    lam: float,
    lam2: float = 0.0,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """One proximal-gradient step + regularization value.

    Semantics of MLlib 1.3's SimpleUpdater / L1Updater / SquaredL2Updater
    (invoked by the reference at ``AGD.scala:215-220``), with the
    stepSize/sqrt(iter) internal rescaling handled by the *caller*
    (models/updater.py), since AGD always passes iter=1 exactly to defeat it.

    Returns ``(w_new, reg_value)`` with reg_value a float64 scalar tensor.
    """
    if kind == PROX_SIMPLE:
        w_new = w - step * g
        reg = torch.zeros((), dtype=torch.float64, device=w.device)
    elif kind == PROX_L1:
        w1 = w - step * g
        shrink = lam * step
        w_new = torch.sign(w1) * torch.clamp(w1.abs() - shrink, min=0.0)
        reg = lam * w_new.abs().to(torch.float64).sum()
    elif kind == PROX_SQUARED_L2:
        w_new = w * (1.0 - step * lam) - step * g
        reg = 0.5 * lam * (w_new.to(torch.float64) ** 2).sum()
    elif kind == PROX_ELASTIC_NET:
        # prox of lam*|w|_1 + lam2/2*|w|^2: soft-threshold then shrink
        w1 = w - step * g
        wsoft = torch.sign(w1) * torch.clamp(w1.abs() - lam * step, min=0.0)
        w_new = wsoft / (1.0 + step * lam2)
        reg = (lam * w_new.abs().to(torch.float64).sum()
               + 0.5 * lam2 * (w_new.to(torch.float64) ** 2).sum())
    else:
        raise ValueError(f"unknown prox kind {kind}")
    return w_new, reg


def axpby(a: float, x: torch.Tensor, b: float, y: torch.Tensor, out: Optional[torch.Tensor] = None) -> torch.Tensor:
    """out = a*x + b*y (the reference's affine combinations, ``AGD.scala:249,255``)."""
    if out is None:
        out = torch.empty_like(x)
    torch.add(a * x, y, alpha=b, out=out)
    return out


def fused_scalars(
    x: torch.Tensor,
    y: torch.Tensor,
    g_y: torch.Tensor,
    x_old: torch.Tensor,
) -> torch.Tensor:
    """One pass producing the 5 iteration scalars as float64 [5]:

      [0] ||x - y||^2          (backtracking xy_sq,      AGD.scala:263-264)
      [1] <x - y, g_y>         (simple backtrack test,   AGD.scala:273)
      [2] ||x||^2              (convergence,             AGD.scala:315)
      [3] ||x - x_old||^2      (convergence,             AGD.scala:316)
      [4] <g_y, x - x_old>     (gradient-test restart,   AGD.scala:327)
    """
    xd = x.to(torch.float64)
    xy = xd - y.to(torch.float64)
    dx = xd - x_old.to(torch.float64)
    gy = g_y.to(torch.float64)
    return torch.stack(
        [
            (xy * xy).sum(),
            (xy * gy).sum(),
            (xd * xd).sum(),
            (dx * dx).sum(),
            (gy * dx).sum(),
        ]
    )


def dot_diff(x: torch.Tensor, y: torch.Tensor, g_x: torch.Tensor, g_y: torch.Tensor) -> torch.Tensor:
    """<x - y, g_x - g_y> as float64 scalar (alternate backtrack test, AGD.scala:278)."""
    xy = x.to(torch.float64) - y.to(torch.float64)
    dg = g_x.to(torch.float64) - g_y.to(torch.float64)
    return (xy * dg).sum()
