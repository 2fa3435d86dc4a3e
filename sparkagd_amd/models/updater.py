"""Updater (proximal-step) plug-ins.

Semantics of MLlib 1.3's ``SimpleUpdater`` / ``L1Updater`` /
``SquaredL2Updater`` (invoked by the reference at
``AcceleratedGradientDescent.scala:215-220``): one proximal gradient step plus
the regularization value at the new point. MLlib internally rescales the step
as stepSize/sqrt(iter); the AGD driver defeats that by passing iter=1
(``AGD.scala:218-219``) — we reproduce the rescaling here (host-side scalar)
so ``runMiniBatch`` gets the decaying schedule and AGD gets the raw step.

On GPU the step runs as one fused HIP prox kernel per updater type that also
emits the regularization scalar via a block reduction (no weight broadcast is
ever needed: every rank executes the identical update deterministically).
"""

from __future__ import annotations

import math
from typing import Tuple

import torch

from .. import ops


class Updater:
    """Updates weights given a gradient step; returns the new weights and the
    value of the regularization term at the new weights."""

    PROX_KIND: int = -1
    #: True when the prox step is AFFINE in (w, g) — then margins propagate
    #: algebraically (A·prox(w - s·g) is computable from A·w and A·g) and the
    #: optimizer can run in margin-state-tracking mode (optimizer.py).
    AFFINE_PROX: bool = False

    def prox_margins(self, wm: torch.Tensor, gm: torch.Tensor, step: float,
                     reg_param: float) -> torch.Tensor:
        """Margins of the prox output from margins of (w, g); only valid when
        AFFINE_PROX."""
        raise NotImplementedError(f"{type(self).__name__} has no affine prox")

    def prox_margin_coeffs(self, step: float, reg_param: float):
        """(pz, pg) with prox_margins(wm, gm) = pz*wm + pg*gm; only valid
        when AFFINE_PROX (used by the fused AT margin-update kernels)."""
        raise NotImplementedError(f"{type(self).__name__} has no affine prox")

    def compute(
        self,
        weights_old: torch.Tensor,
        gradient: torch.Tensor,
        step_size: float,
        iter: int,  # noqa: A002 - MLlib parameter name
        reg_param: float,
    ) -> Tuple[torch.Tensor, torch.Tensor]:
        """MLlib contract: step = step_size / sqrt(iter); returns (w_new, reg_value).

        ``reg_value`` is a float64 scalar tensor (device-resident on GPU).
        """
        this_step = step_size / math.sqrt(iter)
        return ops.prox(self.PROX_KIND, weights_old, gradient, this_step, reg_param)

    def reg_value(self, weights: torch.Tensor, reg_param: float) -> torch.Tensor:
        """Regularization value at ``weights`` (the reference's step=0 trick,
        ``AGD.scala:305``: applyProjector(x, g, 0) returns c_x)."""
        zero_grad = torch.zeros_like(weights)
        _, reg = ops.prox(self.PROX_KIND, weights, zero_grad, 0.0, reg_param)
        return reg


class SimpleUpdater(Updater):
    """w' = w - step*g; no regularization."""

    PROX_KIND = ops.PROX_SIMPLE
    AFFINE_PROX = True

    def prox_margins(self, wm, gm, step, reg_param):
        return ops.axpby(1.0, wm, -step, gm)

    def prox_margin_coeffs(self, step, reg_param):
        return 1.0, -step


class L1Updater(Updater):
    """Gradient step then soft-threshold prox: w'_i = sign(w1_i)*max(0, |w1_i| - step*lambda);
    reg = lambda * ||w'||_1."""

    PROX_KIND = ops.PROX_L1


class ElasticNetUpdater(Updater):
    """Elastic-net prox (beyond the reference's updater family):
    reg(w) = l1_ratio*lambda*||w||_1 + (1-l1_ratio)*lambda/2*||w||^2;
    prox = soft-threshold by step*lambda1, then shrink by 1/(1+step*lambda2).
    Non-affine (margin tracking / Gram solver fall back to the direct path)."""

    PROX_KIND = ops.PROX_ELASTIC_NET
    AFFINE_PROX = False

    def __init__(self, l1_ratio: float = 0.5):
        if not (0.0 <= l1_ratio <= 1.0):
            raise ValueError("l1_ratio must be in [0, 1]")
        self.l1_ratio = l1_ratio

    def compute(self, weights_old, gradient, step_size, iter, reg_param):  # noqa: A002
        this_step = step_size / math.sqrt(iter)
        return ops.prox(self.PROX_KIND, weights_old, gradient, this_step,
                        self.l1_ratio * reg_param,
                        (1.0 - self.l1_ratio) * reg_param)

    def reg_value(self, weights, reg_param):
        zero_grad = torch.zeros_like(weights)
        _, reg = ops.prox(self.PROX_KIND, weights, zero_grad, 0.0,
                          self.l1_ratio * reg_param,
                          (1.0 - self.l1_ratio) * reg_param)
        return reg


class SquaredL2Updater(Updater):
    """w' = w*(1 - step*lambda) - step*g; reg = (lambda/2) ||w'||^2."""

    PROX_KIND = ops.PROX_SQUARED_L2
    AFFINE_PROX = True

    def prox_margins(self, wm, gm, step, reg_param):
        return ops.axpby(1.0 - step * reg_param, wm, -step, gm)

    def prox_margin_coeffs(self, step, reg_param):
        return 1.0 - step * reg_param, -step
