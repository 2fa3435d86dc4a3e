"""Accelerated (proximal) gradient descent — the algorithm core.

A faithful re-implementation of the reference's driver loop
(``AcceleratedGradientDescent.scala:177-338``): the TFOCS Auslender–Teboulle
two-sequence accelerated proximal gradient method (Becker, Candès & Grant
2010) with backtracking line search on the local Lipschitz estimate L and the
O'Donoghue–Candès gradient-test adaptive restart — plus the MLlib-style
mini-batch SGD golden baseline (``GradientDescent.runMiniBatchSGD``, used as
the oracle by the reference suite at ``Suite.scala:78-86``).

MI355X-native restatement: scalar state (theta, L, f) lives on
the host in float64; vector state (x, z, y, g) is device-resident and
**replicated identically on every rank** — each evaluation is one fused HIP
kernel sequence per GPU plus one RCCL all-reduce, and the prox/Nesterov
update runs on every rank with bit-identical inputs so no weight broadcast
ever happens (the reference broadcasts the weights every evaluation,
``AGD.scala:193``).
"""

from __future__ import annotations

import logging
import math
import os
import time
from typing import List, Optional, Tuple

import torch

from . import ops
from .config import AGDConfig
from .models.gradient import Gradient
from .models.updater import Updater
from .parallel.comm import Communicator

logger = logging.getLogger(__name__)

#: cumulative all-reduce wall time when SPARKAGD_TIMING_DETAIL=1
_COMM_SECONDS = [0.0]


class Optimizer:
    """Solves an optimization problem over a (sharded) dataset.

    API parity with MLlib's ``Optimizer`` trait (reference usage
    ``AGD.scala:42``): single method ``optimize(data, initial_weights)``.
    """

    def optimize(self, data, initial_weights: torch.Tensor) -> torch.Tensor:
        raise NotImplementedError


def _apply_smooth(
    data, gradient: Gradient, comm: Communicator, v: torch.Tensor,
    mask: Optional[torch.Tensor] = None, need_grad: bool = True,
) -> Tuple[float, Optional[torch.Tensor], float]:
    """The distributed loss/gradient pass (reference ``applySmooth``,
    ``AGD.scala:192-208``): one fused kernel sequence on the local shard +
    one all-reduce; returns (mean loss, mean gradient [device], count).

    ``need_grad=False`` evaluates the loss side only (one data pass instead
    of two): used for the simple-backtracking f_x trials, which the reference
    pays a full gradient evaluation for (AGD.scala:269) without ever using
    g_x in the simple test.

    SPARKAGD_TIMING_DETAIL=1 splits compute vs all-reduce wall time into
    _COMM_SECONDS (adds one device sync per eval — diagnostics only)."""
    grad_sum, loss_count = gradient.eval(data, v, mask, need_grad)
    detail = os.environ.get("SPARKAGD_TIMING_DETAIL") == "1"
    if detail:
        if v.is_cuda:
            torch.cuda.synchronize(v.device)
        t_comm = time.perf_counter()
    if grad_sum is not None:
        comm.allreduce_(grad_sum)
    comm.allreduce_(loss_count)
    if detail:
        if v.is_cuda:
            torch.cuda.synchronize(v.device)
        _COMM_SECONDS[0] += time.perf_counter() - t_comm
    lc = loss_count.to("cpu", non_blocking=False)  # single host sync per eval
    loss_sum, count = float(lc[0]), float(lc[1])
    if count > 0 and grad_sum is not None:
        grad_sum.div_(count)
    return (loss_sum / count if count > 0 else float("nan")), grad_sum, count


def run(
    data,
    gradient: Gradient,
    updater: Updater,
    convergence_tol: float,
    num_iterations: int,
    reg_param: float,
    initial_weights: torch.Tensor,
    L0: float = 1.0,
    Lexact: float = math.inf,
    beta: float = 0.5,
    alpha: float = 0.9,
    may_restart: bool = True,
    *,
    loss_history_mode: str = "exact",
    comm: Optional[Communicator] = None,
    metrics=None,
    checkpoint_path: Optional[str] = None,
    checkpoint_every: int = 0,
    resume_from: Optional[str] = None,
    iteration_hook=None,
    check_replication_every: int = 0,
    track_margins: str | bool = "auto",
    margin_refresh_every: int = 0,
    solver: str = "direct",
    gram_op=None,
    backtrack_tol: float = 1e-10,
) -> Tuple[torch.Tensor, List[float]]:
    """Run accelerated proximal gradient descent.

    Parameter-for-parameter equivalent of the reference's 12-parameter static
    ``AcceleratedGradientDescent.run`` (``AGD.scala:177-189``), returning
    ``(weights, loss_history)``. Keyword-only extras are new capabilities
    (metrics, checkpoint/resume, loss-history mode — SURVEY.md §5).
    ``iteration_hook(n_iter)`` runs at the end of each completed iteration;
    returning the string "stop" ends the loop (used by bench.py to bracket
    exactly K timed steps).

    **Margin-state tracking** (``track_margins``, default auto): the same
    linear-operator caching TFOCS itself performs. Margins are linear in the
    weights, and for AFFINE prox operators (Simple/SquaredL2) the margins of
    the AT iterates propagate algebraically:

        A·y  = (1-θ)·A·x_old + θ·A·z_old            (free)
        A·z' = prox-margins(A·z_old, A·g_y, step)   (free given A·g_y)
        A·x' = (1-θ)·A·x_old + θ·A·z'               (free)

    so the only data passes per accepted backtracking trial are the A^T·m
    gradient pass and one margins pass over g_y — 2 instead of 3 (and a
    rejected trial costs 2 instead of 3; loss-only f_x checks cost ZERO
    passes). The math is identical up to fp accumulation order;
    ``margin_refresh_every`` > 0 recomputes the tracked margins from the
    weight vectors every k iterations to bound drift. Disabled automatically
    for non-affine updaters (L1) and when ``resume_from``/bitwise
    reproducibility against the non-tracking path is required
    (``track_margins=False``).
    """
    if solver not in ("direct", "gram"):
        raise ValueError("solver must be 'direct' or 'gram'")
    if solver == "gram":
        # Dual-space solver (gram.py): O(n_local * n_global) iterations for
        # the n << d regime. Checkpoint/resume and mini-batching stay on the
        # direct path.
        from .gram import run_gram

        if resume_from is not None or checkpoint_path is not None:
            raise ValueError("checkpoint/resume requires the direct solver")
        return run_gram(
            data, gradient, updater, convergence_tol, num_iterations,
            reg_param, initial_weights, L0, Lexact, beta, alpha, may_restart,
            loss_history_mode=loss_history_mode, comm=comm, metrics=metrics,
            iteration_hook=iteration_hook, gram_op=gram_op,
            backtrack_tol=backtrack_tol,
        )

    comm = comm or Communicator()

    x = initial_weights.clone()
    z = x.clone()
    theta = math.inf
    L = L0
    backtrack_simple = True
    loss_history: List[float] = []
    start_iter = 1

    if resume_from is not None:
        from .utils.checkpoint import load_checkpoint

        state = load_checkpoint(resume_from, device=x.device, dtype=x.dtype)
        x, z = state["x"], state["z"]
        theta, L = state["theta"], state["L"]
        backtrack_simple = state["backtrack_simple"]
        loss_history = list(state["loss_history"])
        start_iter = state["iter"] + 1

    eval_state = {"n": 0, "seconds": 0.0}

    def apply_smooth(v, mask=None, need_grad=True):
        t0 = time.perf_counter()
        out = _apply_smooth(data, gradient, comm, v, mask, need_grad)
        # _apply_smooth ends with the host fetch of (loss, count), so this
        # wall segment covers the kernels + all-reduce for the evaluation.
        eval_state["n"] += 1
        eval_state["seconds"] += time.perf_counter() - t0
        return out

    def apply_smooth_margins(vm, need_grad=True):
        """applySmooth from tracked margins: multiplier/loss (+ A^T·m)."""
        t0 = time.perf_counter()
        grad_sum, loss_count = gradient.eval_from_margins(data, vm, need_grad=need_grad)
        detail = os.environ.get("SPARKAGD_TIMING_DETAIL") == "1"
        if detail:
            if vm.is_cuda:
                torch.cuda.synchronize(vm.device)
            t_comm = time.perf_counter()
        if grad_sum is not None:
            comm.allreduce_(grad_sum)
        comm.allreduce_(loss_count)
        if detail:
            if vm.is_cuda:
                torch.cuda.synchronize(vm.device)
            _COMM_SECONDS[0] += time.perf_counter() - t_comm
        lc = loss_count.to("cpu")
        loss_sum, count = float(lc[0]), float(lc[1])
        if count > 0 and grad_sum is not None:
            grad_sum.div_(count)
        eval_state["n"] += 1
        eval_state["seconds"] += time.perf_counter() - t0
        return (loss_sum / count if count > 0 else float("nan")), grad_sum, count

    # Margin-state tracking eligibility (see docstring).
    tracking = bool(track_margins) and getattr(updater, "AFFINE_PROX", False) \
        and hasattr(data, "margins") and hasattr(data, "eval_from_margins")
    xm = zm = None
    if tracking:
        xm = gradient.margins(data, x)
        zm = xm.clone() if torch.equal(x, z) else gradient.margins(data, z)

    broke = False
    for n_iter in range(start_iter, num_iterations + 1):
        t_iter0 = time.perf_counter()
        eval_n0, eval_s0 = eval_state["n"], eval_state["seconds"]
        comm_s0 = _COMM_SECONDS[0]
        if tracking and margin_refresh_every > 0 and n_iter % margin_refresh_every == 0:
            xm = gradient.margins(data, x)
            zm = gradient.margins(data, z)
        # Auslender and Teboulle's accelerated method (AGD.scala:237-255).
        x_old, z_old = x, z
        xm_old, zm_old = xm, zm
        L_old = L
        L = L * alpha
        theta_old = theta

        f_y = 0.0
        g_y: Optional[torch.Tensor] = None
        f_x_bt: Optional[float] = None  # accepted backtracking f_x, for loss history reuse
        scal = None  # the 5 fused iteration scalars (float64, host)
        n_backtracks = 0

        while True:
            # theta recurrence; theta_old = inf on the first iteration /
            # after a restart gives theta = 1 (AGD.scala:248).
            theta = 2.0 / (1.0 + math.sqrt(1.0 + 4.0 * (L / L_old) / (theta_old * theta_old)))
            y = ops.axpby(1.0 - theta, x_old, theta, z_old)
            if tracking:
                ym = ops.axpby(1.0 - theta, xm_old, theta, zm_old)
                f_y, g_y, _count = apply_smooth_margins(ym)  # A^T·m pass only
            else:
                f_y, g_y, _count = apply_smooth(y)
            step = 1.0 / (theta * L)
            z, _ = updater.compute(z_old, g_y, step, 1, reg_param)
            x = ops.axpby(1.0 - theta, x_old, theta, z)
            if tracking:
                gm = gradient.margins(data, g_y)  # the only other data pass
                try:
                    pz, pg = updater.prox_margin_coeffs(step, reg_param)
                except NotImplementedError:
                    pz = None
                if pz is not None:
                    # fused: zm' and xm' in one pass (k_at_margin_update) —
                    # one launch + one fewer read stream than the
                    # prox-margins + axpby pair; identical algebra
                    zm, xm = ops.at_margin_update(zm_old, xm_old, gm,
                                                  pz, pg, theta)
                else:
                    zm = updater.prox_margins(zm_old, gm, step, reg_param)
                    xm = ops.axpby(1.0 - theta, xm_old, theta, zm)

            if beta >= 1.0:
                scal = None  # computed after the loop for the convergence test
                break

            # Backtracking (AGD.scala:261-292). One fused pass produces all 5
            # iteration scalars (backtracking + convergence + restart).
            scal = ops.fused_scalars(x, y, g_y, x_old).to("cpu")
            xy_sq = float(scal[0])
            if xy_sq == 0.0:
                break

            if backtrack_simple:
                # the simple test needs f_x only: loss-only evaluation
                # (tracking: ZERO data passes — f_x comes from xm)
                if tracking:
                    f_x, _gx_none, _ = apply_smooth_margins(xm, need_grad=False)
                else:
                    f_x, _gx_none, _ = apply_smooth(x, need_grad=False)
                f_x_bt = f_x
                q_x = f_y + float(scal[1]) + 0.5 * L * xy_sq
                localL = L + 2.0 * max(f_x - q_x, 0.0) / xy_sq
                backtrack_simple = abs(f_y - f_x) >= backtrack_tol * max(abs(f_x), abs(f_y))
            else:
                if tracking:
                    f_x, g_x, _ = apply_smooth_margins(xm)
                else:
                    f_x, g_x, _ = apply_smooth(x)
                f_x_bt = f_x
                localL = 2.0 * float(ops.dot_diff(x, y, g_x, g_y)) / xy_sq

            if localL <= L or L >= Lexact:
                break

            n_backtracks += 1
            if not math.isinf(localL):
                L = min(Lexact, localL)
            else:
                localL = L
            L = min(Lexact, max(localL, L / beta))

        if scal is None:
            scal = ops.fused_scalars(x, y, g_y, x_old).to("cpu")

        # Loss history (AGD.scala:296-307). 'exact' reproduces the reference's
        # extra full-data pass at x (TFOCS validation); 'backtrack' reuses the
        # accepted backtracking evaluation; 'none' records f_y + c_y.
        if loss_history_mode == "exact":
            if tracking:
                f_x2, _g_x2, _ = apply_smooth_margins(xm, need_grad=False)
            else:
                f_x2, _g_x2, _ = apply_smooth(x, need_grad=False)
            c_x = float(updater.reg_value(x, reg_param))
            loss_history.append(f_x2 + c_x)
        elif loss_history_mode == "backtrack" and f_x_bt is not None:
            c_x = float(updater.reg_value(x, reg_param))
            loss_history.append(f_x_bt + c_x)
        else:
            c_y = float(updater.reg_value(y, reg_param))
            loss_history.append(f_y + c_y)

        if math.isnan(f_y) or math.isinf(f_y):
            logger.warning("Unable to compute loss function.")
            broke = True

        # Convergence (AGD.scala:314-324).
        norm_x = math.sqrt(max(float(scal[2]), 0.0))
        norm_dx = math.sqrt(max(float(scal[3]), 0.0))
        restarted = False
        if not broke:
            if norm_dx == 0.0 and n_iter > 1:
                broke = True
            elif norm_dx < convergence_tol * max(norm_x, 1.0):
                broke = True

        # Gradient-test restart (O'Donoghue & Candès 2013; AGD.scala:326-331).
        if not broke and may_restart and float(scal[4]) > 0.0:
            z = x.clone()
            if tracking:
                zm = xm.clone()
            theta = math.inf
            backtrack_simple = True
            restarted = True

        if (
            check_replication_every > 0
            and n_iter % check_replication_every == 0
            and not comm.check_replicated(x)
        ):
            raise RuntimeError(
                f"rank divergence detected at iteration {n_iter}: the "
                "replicated weight state is no longer identical across ranks"
            )

        if metrics is not None:
            metrics.log(
                iter=n_iter,
                loss=loss_history[-1],
                f_y=f_y,
                L=L,
                theta=theta,
                n_backtracks=n_backtracks,
                restarted=restarted,
                norm_dx=norm_dx,
                iter_seconds=time.perf_counter() - t_iter0,
                n_evals=eval_state["n"] - eval_n0,
                eval_seconds=eval_state["seconds"] - eval_s0,
                **(
                    {"comm_seconds": _COMM_SECONDS[0] - comm_s0}
                    if os.environ.get("SPARKAGD_TIMING_DETAIL") == "1" else {}
                ),
            )

        if (
            checkpoint_path is not None
            and checkpoint_every > 0
            and (n_iter % checkpoint_every == 0 or broke or n_iter == num_iterations)
            and comm.rank == 0
        ):
            from .utils.checkpoint import save_checkpoint

            save_checkpoint(
                checkpoint_path,
                x=x, z=z, theta=theta, L=L, iter=n_iter,
                backtrack_simple=backtrack_simple, loss_history=loss_history,
            )

        if broke:
            break
        if iteration_hook is not None and iteration_hook(n_iter) == "stop":
            break

    logger.info(
        "AcceleratedGradientDescent.run finished. Last 10 losses %s",
        ", ".join(f"{v:.6g}" for v in loss_history[-10:]),
    )
    return x, loss_history


def run_mini_batch(
    data,
    gradient: Gradient,
    updater: Updater,
    step_size: float,
    num_iterations: int,
    reg_param: float,
    mini_batch_fraction: float,
    initial_weights: torch.Tensor,
    *,
    comm: Optional[Communicator] = None,
    seed: int = 42,
    metrics=None,
    step_schedule: str = "sqrt",
) -> Tuple[torch.Tensor, List[float]]:
    """Mini-batch (S)GD — the golden-baseline optimizer.

    Semantics of MLlib 1.3's ``GradientDescent.runMiniBatchSGD`` (reference
    usage ``Suite.scala:78-86``): per iteration i, sample a
    Bernoulli(fraction) row subset with seed ``seed + i``, reduce
    (grad_sum, loss_sum, batch_size), record loss_sum/batch_size + regVal
    (regVal from the *previous* update), then one updater step with the real
    iteration number so the internal step decays as step_size/sqrt(i).

    On GPU the sampling is a seeded per-shard mask pushed into the gradient
    kernels (masked rows contribute neither loss nor gradient).

    ``step_schedule``: 'sqrt' is MLlib's step_size/sqrt(i) (the default and
    the parity mode); 'constant' uses step_size; 'linear' is the
    strong-convexity (Pegasos-style) schedule 1/(reg_param * i) for
    L2-regularized objectives (BASELINE.md hinge/SVM config).
    """
    if step_schedule not in ("sqrt", "constant", "linear"):
        raise ValueError("step_schedule must be sqrt|constant|linear")
    if step_schedule == "linear" and reg_param <= 0:
        raise ValueError("linear (strong-convexity) schedule needs reg_param > 0")
    comm = comm or Communicator()
    w = initial_weights.clone()
    history: List[float] = []
    reg_val = float(updater.compute(w, torch.zeros_like(w), 0.0, 1, reg_param)[1])

    device = w.device
    n_local = data.n
    use_mask = mini_batch_fraction < 1.0

    for i in range(1, num_iterations + 1):
        mask = None
        if use_mask:
            gen = torch.Generator(device=device)
            gen.manual_seed(((seed + i) * 1000003 + comm.rank * 7919) % (2**63 - 1))
            mask = (
                torch.rand(n_local, generator=gen, device=device) < mini_batch_fraction
            ).to(torch.uint8)
        grad_sum, loss_count = gradient.eval(data, w, mask)
        comm.allreduce_eval_(grad_sum, loss_count)
        lc = loss_count.to("cpu")
        loss_sum, batch_size = float(lc[0]), float(lc[1])
        if batch_size > 0:
            history.append(loss_sum / batch_size + reg_val)
            grad_sum.div_(batch_size)
            if step_schedule == "sqrt":
                w, reg_t = updater.compute(w, grad_sum, step_size, i, reg_param)
            else:
                s = step_size if step_schedule == "constant" else 1.0 / (reg_param * i)
                w, reg_t = updater.compute(w, grad_sum, s, 1, reg_param)
            reg_val = float(reg_t)
            if metrics is not None:
                metrics.log(iter=i, loss=history[-1], batch_size=batch_size)
        else:
            logger.warning("Iteration (%d/%d). The size of sampled batch is zero", i, num_iterations)
    return w, history


#: MLlib-parity alias (the API name BASELINE.json requires).
runMiniBatch = run_mini_batch


class AcceleratedGradientDescent(Optimizer):
    """Configurable optimizer with the reference's fluent-setter surface
    (``AGD.scala:41-144``): construct with (gradient, updater) delegates, chain
    setters, then call ``optimize(data, initial_weights)``."""

    def __init__(self, gradient: Gradient, updater: Updater,
                 config: Optional[AGDConfig] = None, comm: Optional[Communicator] = None):
        self.gradient = gradient
        self.updater = updater
        self.config = config or AGDConfig()
        self.comm = comm
        self.metrics = None
        self.checkpoint_path: Optional[str] = None
        self.checkpoint_every: int = 0
        self.resume_from: Optional[str] = None
        self.loss_history: List[float] = []

    # --- fluent setters (camelCase = reference parity, AGD.scala:57-120) ---
    def setConvergenceTol(self, tol: float) -> "AcceleratedGradientDescent":
        self.config.convergence_tol = tol
        return self

    def setNumIterations(self, iters: int) -> "AcceleratedGradientDescent":
        self.config.num_iterations = iters
        return self

    def setRegParam(self, reg_param: float) -> "AcceleratedGradientDescent":
        self.config.reg_param = reg_param
        return self

    def setL0(self, L0: float) -> "AcceleratedGradientDescent":
        self.config.L0 = L0
        return self

    def setLexact(self, Lexact: float) -> "AcceleratedGradientDescent":
        self.config.Lexact = Lexact
        return self

    def setBeta(self, beta: float) -> "AcceleratedGradientDescent":
        self.config.beta = beta
        return self

    def setAlpha(self, alpha: float) -> "AcceleratedGradientDescent":
        self.config.alpha = alpha
        return self

    def setMayRestart(self, may_restart: bool) -> "AcceleratedGradientDescent":
        self.config.may_restart = may_restart
        return self

    def setGradient(self, gradient: Gradient) -> "AcceleratedGradientDescent":
        self.gradient = gradient
        return self

    def setUpdater(self, updater: Updater) -> "AcceleratedGradientDescent":
        self.updater = updater
        return self

    # pythonic aliases
    set_convergence_tol = setConvergenceTol
    set_num_iterations = setNumIterations
    set_reg_param = setRegParam
    set_gradient = setGradient
    set_updater = setUpdater

    def optimize(self, data, initial_weights: torch.Tensor) -> torch.Tensor:
        """Run AGD; returns the solution vector (reference ``AGD.scala:128-144``).
        The per-iteration loss history is kept on ``self.loss_history``."""
        c = self.config
        c.validate()
        weights, self.loss_history = run(
            data,
            self.gradient,
            self.updater,
            c.convergence_tol,
            c.num_iterations,
            c.reg_param,
            initial_weights,
            c.L0,
            c.Lexact,
            c.beta,
            c.alpha,
            c.may_restart,
            loss_history_mode=c.loss_history_mode,
            comm=self.comm,
            metrics=self.metrics,
            checkpoint_path=self.checkpoint_path,
            checkpoint_every=self.checkpoint_every,
            resume_from=self.resume_from,
            solver=c.solver,
            track_margins=c.track_margins,
            margin_refresh_every=c.margin_refresh_every,
            backtrack_tol=c.backtrack_tol,
        )
        return weights


class GradientDescent(Optimizer):
    """Fluent mini-batch SGD optimizer — the analog of MLlib 1.3's
    ``GradientDescent`` class (the ``Optimizer`` the reference suite's golden
    baseline ``runMiniBatchSGD`` belongs to, ``Suite.scala:78-86``), with the
    same setter surface. ``optimize`` delegates to :func:`run_mini_batch`."""

    def __init__(self, gradient: Gradient, updater: Updater,
                 comm: Optional[Communicator] = None):
        self.gradient = gradient
        self.updater = updater
        self.comm = comm
        self.step_size = 1.0
        self.num_iterations = 100
        self.reg_param = 0.0
        self.mini_batch_fraction = 1.0
        self.step_schedule = "sqrt"
        self.seed = 42
        self.loss_history: List[float] = []

    def setStepSize(self, step: float) -> "GradientDescent":
        self.step_size = step
        return self

    def setNumIterations(self, iters: int) -> "GradientDescent":
        self.num_iterations = iters
        return self

    def setRegParam(self, reg_param: float) -> "GradientDescent":
        self.reg_param = reg_param
        return self

    def setMiniBatchFraction(self, fraction: float) -> "GradientDescent":
        self.mini_batch_fraction = fraction
        return self

    def setStepSchedule(self, schedule: str) -> "GradientDescent":
        self.step_schedule = schedule
        return self

    def setGradient(self, gradient: Gradient) -> "GradientDescent":
        self.gradient = gradient
        return self

    def setUpdater(self, updater: Updater) -> "GradientDescent":
        self.updater = updater
        return self

    def optimize(self, data, initial_weights: torch.Tensor) -> torch.Tensor:
        weights, self.loss_history = run_mini_batch(
            data, self.gradient, self.updater, self.step_size,
            self.num_iterations, self.reg_param, self.mini_batch_fraction,
            initial_weights, comm=self.comm, seed=self.seed,
            step_schedule=self.step_schedule,
        )
        return weights
