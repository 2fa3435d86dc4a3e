"""Hyperparameter configuration for the AGD optimizer.

Knob names and defaults mirror the reference's fluent-setter state block
(``AcceleratedGradientDescent.scala:44-51``): convergenceTol=1e-4,
numIterations=100, regParam=0.0, L0=1.0, Lexact=+inf, beta=0.5, alpha=0.9,
mayRestart=true.
"""

from __future__ import annotations

import dataclasses
import json
import math


@dataclasses.dataclass
class AGDConfig:
    """Hyperparameters for accelerated proximal gradient descent.

    Attributes
    ----------
    convergence_tol:
        Relative tolerance on ||x - x_old|| / max(||x||, 1) at which
        optimization stops (reference ``AGD.scala:314-324``).
    num_iterations:
        Maximum number of outer iterations.
    reg_param:
        Regularization parameter passed to the Updater.
    L0:
        Initial Lipschitz estimate.
    Lexact:
        Known exact Lipschitz bound; backtracking never raises L above it.
    beta:
        Backtracking shrink factor; ``beta >= 1`` disables backtracking
        (single evaluation per iteration, reference ``AGD.scala:257-259``).
    alpha:
        Per-iteration optimistic Lipschitz decrease factor (``L *= alpha``).
    may_restart:
        Enable the O'Donoghue–Candes gradient-test adaptive restart
        (reference ``AGD.scala:326-331``).
    loss_history_mode:
        'exact'     — TFOCS-parity: one extra full-data pass per iteration to
                      record loss at x (reference ``AGD.scala:302-307``).
        'backtrack' — reuse the accepted backtracking evaluation's f_x (no
                      extra pass; falls back to f_y + c_y when beta >= 1).
        'none'      — record f_y + c_y (cheapest; still one entry/iteration,
                      because tests use len(loss_history) as iteration count).
    """

    convergence_tol: float = 1e-4
    num_iterations: int = 100
    reg_param: float = 0.0
    L0: float = 1.0
    Lexact: float = math.inf
    beta: float = 0.5
    alpha: float = 0.9
    may_restart: bool = True
    loss_history_mode: str = "exact"
    #: 'direct' streams the shard (2 passes/iteration with margin tracking);
    #: 'gram' precomputes K = A·A^T and iterates in O(n_local·n_global)
    #: (dense shards + affine prox only; see sparkagd_amd/gram.py).
    solver: str = "direct"
    #: margin-state tracking on the direct solver (optimizer.run docstring);
    #: True/'auto' enable when eligible, False forces full-evaluation passes
    #: (bitwise reproducibility vs checkpoints).
    track_margins: bool | str = "auto"
    #: recompute tracked margins from the weight vectors every k iterations
    #: (0 = never; drift is at fp32 rounding level).
    margin_refresh_every: int = 0

    # Numerical guard below which the simple backtracking test switches to the
    # alternate (cancellation-safe) test (reference ``AGD.scala:234-235,272-278``).
    backtrack_tol: float = 1e-10

    def to_json(self) -> str:
        d = dataclasses.asdict(self)
        if math.isinf(d["Lexact"]):
            d["Lexact"] = "inf"
        return json.dumps(d)

    @classmethod
    def from_json(cls, s: str) -> "AGDConfig":
        d = json.loads(s)
        if d.get("Lexact") == "inf":
            d["Lexact"] = math.inf
        return cls(**d)

    def validate(self) -> None:
        if self.num_iterations < 0:
            raise ValueError("num_iterations must be >= 0")
        if self.L0 <= 0:
            raise ValueError("L0 must be > 0")
        if not (0.0 < self.alpha <= 1.0):
            raise ValueError("alpha must be in (0, 1]")
        if self.beta <= 0:
            raise ValueError("beta must be > 0")
        if self.loss_history_mode not in ("exact", "backtrack", "none"):
            raise ValueError("loss_history_mode must be exact|backtrack|none")
        if self.solver not in ("direct", "gram"):
            raise ValueError("solver must be direct|gram")
