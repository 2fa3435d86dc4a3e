"""Gradient (loss) plug-ins.

Batched equivalents of MLlib 1.3's ``LogisticGradient`` /
``LeastSquaresGradient`` / ``HingeGradient`` (invoked per example by the
reference at ``AcceleratedGradientDescent.scala:198``; constructed in
``AcceleratedGradientDescentSuite.scala:39,251``). Where MLlib folds one
example at a time into an accumulator, these evaluate a whole GPU shard with
one fused HIP kernel sequence (margins = A·w, elementwise multiplier,
grad = A^T·m, loss reduction).

A per-example ``compute`` with MLlib's exact signature is kept for API
parity and for documentation of the single-example math.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from .. import ops


class Gradient:
    """Computes summed loss and gradient of a loss function over a shard."""

    LOSS_TYPE: int = -1

    def eval(
        self,
        shard,
        w: torch.Tensor,
        mask: Optional[torch.Tensor] = None,
        need_grad: bool = True,
    ) -> Tuple[torch.Tensor, torch.Tensor]:
        """Return (grad_sum [d], loss_count float64 [2]) over the local shard.

        ``loss_count[0]`` is the sum of per-example losses; ``loss_count[1]``
        is the number of (unmasked) examples. Dividing by the globally
        all-reduced count happens in the optimizer, matching
        ``AGD.scala:206-207``. With ``need_grad=False`` only the loss side is
        computed (one data pass instead of two — used by the simple
        backtracking test, which needs f_x but not g_x).
        """
        return shard.eval(w, self.LOSS_TYPE, mask, need_grad)

    def margins(self, shard, v: torch.Tensor) -> torch.Tensor:
        """A @ v over the local shard (margin-state tracking)."""
        return shard.margins(v)

    def eval_from_margins(self, shard, margins: torch.Tensor,
                          mask: Optional[torch.Tensor] = None,
                          need_grad: bool = True):
        """Loss (+ gradient) from precomputed margins — saves the A·w pass."""
        return shard.eval_from_margins(margins, self.LOSS_TYPE, mask, need_grad)

    def multiplier_loss(self, shard, margins: torch.Tensor,
                        mask: Optional[torch.Tensor] = None):
        """(multiplier, loss_count) from precomputed margins (zero data
        passes) — the Gram solver's n-space evaluation."""
        return ops.dense_multiplier_loss(shard.features, margins, shard.labels,
                                         self.LOSS_TYPE, mask,
                                         getattr(shard, "sample_weight", None))

    # --- MLlib per-example API parity (reference Gradient.compute) ---
    def compute(
        self,
        features: torch.Tensor,
        label: float,
        weights: torch.Tensor,
        cum_gradient: torch.Tensor,
    ) -> float:
        """Add one example's gradient into ``cum_gradient`` in place; return its loss.

        Same contract as MLlib's ``Gradient.compute(data, label, weights,
        cumGradient): Double`` (reference usage ``AGD.scala:198``).
        """
        grad_sum, loss_count = ops.reference.dense_eval(
            features.reshape(1, -1),
            torch.tensor([label], dtype=torch.float64, device=features.device),
            weights,
            self.LOSS_TYPE,
        )
        cum_gradient += grad_sum.to(cum_gradient.dtype)
        return float(loss_count[0])


class LogisticGradient(Gradient):
    """Binary logistic loss; labels in {0, 1}.

    loss_i = y>0 ? log1p(exp(-z)) : log1p(exp(z)),  mult_i = sigmoid(z) - y,
    with z = <w, x_i> (MLlib writes margin = -z; the algebra is identical).
    """

    LOSS_TYPE = ops.LOSS_LOGISTIC


class LeastSquaresGradient(Gradient):
    """Squared loss: loss_i = (z - y)^2, mult_i = 2 (z - y)."""

    LOSS_TYPE = ops.LOSS_LEAST_SQUARES


class HingeGradient(Gradient):
    """Hinge loss (linear SVM); labels in {0, 1}, scaled to s = 2y-1 in {-1, 1}.

    loss_i = max(0, 1 - s z), mult_i = -s when s z < 1 else 0.
    NOTE: the hinge is nonsmooth; the accelerated method's backtracking
    assumes a Lipschitz gradient, so SmoothedHingeGradient usually converges
    much better under AGD. Plain hinge is kept for MLlib parity.
    """

    LOSS_TYPE = ops.LOSS_HINGE


class MultinomialLogisticGradient(Gradient):
    """Multinomial (softmax) logistic regression over K classes — a model
    family beyond the reference (MLlib 1.3's LogisticGradient is binary).

    Weights are the flattened [d, K] matrix (feature-major, classes
    contiguous); labels are class indices 0..K-1; the loss is the softmax
    cross-entropy; grad = Aᵀ(softmax(Z) − onehot). Works on dense and CSR
    shards (CSR gradients use the deterministic CSC gather).
    Composes with masks, sample weights, margin-state tracking AND the Gram
    (dual-space) solver (padded class columns ride the coefficient-space
    machinery; sparkagd_amd/gram.py)."""

    IS_MULTICLASS = True

    def __init__(self, num_classes: int):
        if num_classes < 2:
            raise ValueError("num_classes must be >= 2")
        self.num_classes = int(num_classes)

    def eval(self, shard, w, mask=None, need_grad=True):
        from ..ops import multiclass as mc

        if getattr(shard, "kind", None) == "csr":
            return mc.eval_multi_csr(shard, w, self.num_classes, mask,
                                     need_grad, shard.sample_weight)
        return mc.eval_multi(shard.features, shard.labels, w, self.num_classes,
                             mask, need_grad, shard.sample_weight)

    def margins(self, shard, v):
        from ..ops import multiclass as mc

        if getattr(shard, "kind", None) == "csr":
            return mc.csr_margins_multi(shard, v, self.num_classes)
        return mc.margins_multi(shard.features, v, self.num_classes)

    def eval_from_margins(self, shard, margins, mask=None, need_grad=True):
        from ..ops import multiclass as mc

        if getattr(shard, "kind", None) == "csr":
            return mc.eval_multi_csr_from_margins(shard, margins,
                                                  self.num_classes, mask,
                                                  need_grad,
                                                  shard.sample_weight)
        return mc.eval_multi_from_margins(shard.features, margins, shard.labels,
                                          self.num_classes, mask, need_grad,
                                          shard.sample_weight)

    def multiplier_loss(self, shard, margins, mask=None):
        """(padded multiplier flat [n*KC], loss_count) from padded margins —
        the Gram solver's n-space evaluation."""
        from ..ops import multiclass as mc

        return mc.multiplier_loss_multi(shard.labels, margins,
                                        self.num_classes, mask,
                                        getattr(shard, "sample_weight", None))


class SmoothedHingeGradient(Gradient):
    """Rennie's quadratically smoothed hinge (differentiable SVM loss):
    loss = 0 if sz>=1; (1-sz)^2/2 if 0<sz<1; 0.5-sz otherwise. Capability
    beyond the reference: a hinge-family loss that satisfies the smoothness
    assumptions of the accelerated method."""

    LOSS_TYPE = ops.LOSS_SMOOTH_HINGE
