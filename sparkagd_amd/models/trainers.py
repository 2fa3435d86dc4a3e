"""User-layer model trainers (the reference's L6).

In the reference this layer is MLlib's trainer classes (e.g.
``LogisticRegressionWithSGD``-style wrappers) that construct an optimizer and
call ``optimize()`` (in-repo example
``AcceleratedGradientDescentSuite.scala:217-222``). These are the equivalent
conveniences: a GLM trainer per loss family, returning a fitted linear model
with ``predict``.
"""

from __future__ import annotations

from typing import Optional

import torch

from ..config import AGDConfig
from ..optimizer import AcceleratedGradientDescent
from ..parallel.comm import Communicator
from .gradient import HingeGradient, LeastSquaresGradient, LogisticGradient
from .updater import SimpleUpdater, SquaredL2Updater, Updater


def _save_model(path: str, weights: torch.Tensor, meta: dict) -> None:
    import json
    import os as _os
    import tempfile

    from safetensors.torch import save_file

    d = _os.path.dirname(_os.path.abspath(path)) or "."
    _os.makedirs(d, exist_ok=True)
    fd, tmp = tempfile.mkstemp(dir=d, suffix=".tmp")
    _os.close(fd)
    try:
        save_file({"weights": weights.detach().cpu().contiguous().clone()},
                  tmp, metadata={k: json.dumps(v) for k, v in meta.items()})
        _os.replace(tmp, path)
    finally:
        if _os.path.exists(tmp):
            _os.unlink(tmp)


def _load_model(path: str, device=None):
    import json

    from safetensors import safe_open
    from safetensors.torch import load_file

    with safe_open(path, framework="pt", device="cpu") as f:
        meta = {k: json.loads(v) for k, v in (f.metadata() or {}).items()}
    w = load_file(path)["weights"]
    if device is not None:
        w = w.to(device)
    return w, meta


class LinearModel:
    """A fitted generalized linear model: weights [d] (+ loss history).
    MLlib-style model persistence via ``save``/``load`` (safetensors)."""

    def __init__(self, weights: torch.Tensor, loss_history, link: str):
        self.weights = weights
        self.loss_history = list(loss_history)
        self.link = link

    def save(self, path: str) -> None:
        _save_model(path, self.weights, {"kind": "linear", "link": self.link,
                                         "loss_history": self.loss_history})

    @classmethod
    def load(cls, path: str, device=None) -> "LinearModel":
        w, meta = _load_model(path, device)
        if meta.get("kind") != "linear":
            raise ValueError(f"not a LinearModel checkpoint: {meta.get('kind')}")
        return cls(w, meta.get("loss_history", []), meta["link"])

    def margins(self, features: torch.Tensor) -> torch.Tensor:
        acc = torch.float32 if features.dtype in (torch.bfloat16, torch.float16) else features.dtype
        return features.to(acc) @ self.weights.to(acc)

    def predict(self, features: torch.Tensor,
                threshold: Optional[float] = 0.5) -> torch.Tensor:
        """Class predictions. ``threshold`` is the probability cut for
        logistic models (MLlib's ``setThreshold`` analog; margin cut
        logit(t)); ``None`` returns the raw margin (``clearThreshold``
        semantics). Identity-link models always return the raw prediction."""
        import math as _math

        z = self.margins(features)
        if self.link == "identity":
            return z
        if threshold is None:
            return z
        cut = 0.0
        if self.link == "logistic" and threshold != 0.5:
            cut = _math.log(threshold / (1.0 - threshold))
        return (z > cut).to(torch.float32)

    def predict_proba(self, features: torch.Tensor) -> torch.Tensor:
        if self.link != "logistic":
            raise ValueError("predict_proba only for logistic models")
        return torch.sigmoid(self.margins(features))


class _GLMTrainer:
    GRADIENT_CLS = LogisticGradient
    LINK = "logistic"

    @classmethod
    def train(
        cls,
        data,
        num_iterations: int = 100,
        reg_param: float = 0.0,
        convergence_tol: float = 1e-4,
        updater: Optional[Updater] = None,
        initial_weights: Optional[torch.Tensor] = None,
        comm: Optional[Communicator] = None,
        config: Optional[AGDConfig] = None,
        checkpoint_path: Optional[str] = None,
        checkpoint_every: int = 0,
        resume_from: Optional[str] = None,
    ) -> LinearModel:
        cfg = config or AGDConfig()
        cfg.num_iterations = num_iterations
        cfg.reg_param = reg_param
        cfg.convergence_tol = convergence_tol
        if updater is None:
            updater = SquaredL2Updater() if reg_param > 0 else SimpleUpdater()
        opt = AcceleratedGradientDescent(cls.GRADIENT_CLS(), updater, cfg, comm)
        opt.checkpoint_path = checkpoint_path
        opt.checkpoint_every = checkpoint_every
        opt.resume_from = resume_from
        if initial_weights is None:
            wdtype = torch.float64 if data.device.type == "cpu" else torch.float32
            initial_weights = torch.zeros(data.d, device=data.device, dtype=wdtype)
        w = opt.optimize(data, initial_weights)
        return LinearModel(w, opt.loss_history, cls.LINK)


def regularization_path(
    data,
    lambdas,
    gradient=None,
    num_iterations: int = 100,
    convergence_tol: float = 1e-6,
    solver: str = "auto",
    comm: Optional[Communicator] = None,
    warm_start: bool = True,
):
    """Solve an L2 regularization path (one model per lambda), reusing the
    Gram operator across solves when eligible — hyperparameter sweeps then
    cost O(n_local·n_global) per solve instead of full shard passes
    (sparkagd_amd/gram.py). Returns a list of LinearModel, one per lambda.

    solver: 'auto' uses the Gram solver for dense shards (falling back to
    direct if K exceeds the memory budget), 'direct'/'gram' force a path.
    """
    from ..gram import GramOperator
    from ..optimizer import run

    gradient = gradient or LogisticGradient()
    updater = SquaredL2Updater()
    is_multi = getattr(gradient, "IS_MULTICLASS", False)
    link = {0: "logistic", 1: "identity", 2: "hinge", 3: "hinge"}.get(
        gradient.LOSS_TYPE, "logistic")
    comm = comm or Communicator()
    op = None
    if solver in ("auto", "gram") and getattr(data, "kind", None) == "dense":
        try:
            op = GramOperator(data, comm)
        except MemoryError:
            if solver == "gram":
                raise
            op = None
    wdtype = torch.float64 if data.device.type == "cpu" else torch.float32
    dim = data.d * gradient.num_classes if is_multi else data.d
    w = torch.zeros(dim, device=data.device, dtype=wdtype)
    models = []
    for lam in lambdas:
        w0 = w if warm_start else torch.zeros_like(w)
        import math as _math

        w, hist = run(
            data, gradient, updater, convergence_tol, num_iterations,
            float(lam), w0, 1.0, _math.inf, 0.5, 0.9, True,
            loss_history_mode="backtrack", comm=comm,
            solver="gram" if op is not None else "direct", gram_op=op,
        )
        if is_multi:
            models.append(MultinomialModel(w.clone(), hist,
                                           gradient.num_classes))
        else:
            models.append(LinearModel(w.clone(), hist, link))
    return models


class MultinomialModel:
    """A fitted softmax (multinomial logistic) model: weights [d*K]
    feature-major. predict returns class indices; predict_proba softmax
    probabilities [n, K]."""

    def __init__(self, weights: torch.Tensor, loss_history, num_classes: int):
        self.weights = weights
        self.loss_history = list(loss_history)
        self.num_classes = int(num_classes)

    def margins(self, features: torch.Tensor) -> torch.Tensor:
        acc = torch.float32 if features.dtype in (torch.bfloat16, torch.float16) else features.dtype
        w = self.weights.reshape(features.shape[1], self.num_classes).to(acc)
        return features.to(acc) @ w  # [n, K]

    def predict(self, features: torch.Tensor) -> torch.Tensor:
        return self.margins(features).argmax(dim=1).to(torch.float32)

    def predict_proba(self, features: torch.Tensor) -> torch.Tensor:
        return torch.softmax(self.margins(features), dim=1)

    def save(self, path: str) -> None:
        _save_model(path, self.weights,
                    {"kind": "multinomial", "num_classes": self.num_classes,
                     "loss_history": self.loss_history})

    @classmethod
    def load(cls, path: str, device=None) -> "MultinomialModel":
        w, meta = _load_model(path, device)
        if meta.get("kind") != "multinomial":
            raise ValueError(f"not a MultinomialModel checkpoint: {meta.get('kind')}")
        return cls(w, meta.get("loss_history", []), meta["num_classes"])


class SoftmaxRegressionWithAGD:
    """Multinomial (softmax) logistic regression trainer — a model family
    beyond the reference (MLlib 1.3's LogisticGradient is binary-only)."""

    @classmethod
    def train(
        cls,
        data,
        num_classes: int,
        num_iterations: int = 100,
        reg_param: float = 0.0,
        convergence_tol: float = 1e-4,
        updater: Optional[Updater] = None,
        initial_weights: Optional[torch.Tensor] = None,
        comm: Optional[Communicator] = None,
        config: Optional[AGDConfig] = None,
    ) -> MultinomialModel:
        from .gradient import MultinomialLogisticGradient

        cfg = config or AGDConfig()
        cfg.num_iterations = num_iterations
        cfg.reg_param = reg_param
        cfg.convergence_tol = convergence_tol
        if updater is None:
            updater = SquaredL2Updater() if reg_param > 0 else SimpleUpdater()
        opt = AcceleratedGradientDescent(
            MultinomialLogisticGradient(num_classes), updater, cfg, comm)
        if initial_weights is None:
            wdtype = torch.float64 if data.device.type == "cpu" else torch.float32
            initial_weights = torch.zeros(data.d * num_classes,
                                          device=data.device, dtype=wdtype)
        w = opt.optimize(data, initial_weights)
        return MultinomialModel(w, opt.loss_history, num_classes)


class _SGDTrainer:
    """MLlib 1.3's primary entry points were ``LogisticRegressionWithSGD``
    etc. (``GradientDescent``-backed); these are the equivalents, returning
    the same fitted ``LinearModel`` as the AGD trainers."""

    GRADIENT_CLS = LogisticGradient
    LINK = "logistic"

    @classmethod
    def train(
        cls,
        data,
        num_iterations: int = 100,
        step_size: float = 1.0,
        mini_batch_fraction: float = 1.0,
        reg_param: float = 0.0,
        updater: Optional[Updater] = None,
        initial_weights: Optional[torch.Tensor] = None,
        comm: Optional[Communicator] = None,
        step_schedule: str = "sqrt",
    ) -> LinearModel:
        from ..optimizer import GradientDescent

        if updater is None:
            updater = SquaredL2Updater() if reg_param > 0 else SimpleUpdater()
        opt = (GradientDescent(cls.GRADIENT_CLS(), updater, comm)
               .setStepSize(step_size).setNumIterations(num_iterations)
               .setRegParam(reg_param).setMiniBatchFraction(mini_batch_fraction)
               .setStepSchedule(step_schedule))
        if initial_weights is None:
            wdtype = torch.float64 if data.device.type == "cpu" else torch.float32
            initial_weights = torch.zeros(data.d, device=data.device, dtype=wdtype)
        w = opt.optimize(data, initial_weights)
        return LinearModel(w, opt.loss_history, cls.LINK)


class LogisticRegressionWithSGD(_SGDTrainer):
    GRADIENT_CLS = LogisticGradient
    LINK = "logistic"


class LinearRegressionWithSGD(_SGDTrainer):
    GRADIENT_CLS = LeastSquaresGradient
    LINK = "identity"


class SVMWithSGD(_SGDTrainer):
    GRADIENT_CLS = HingeGradient
    LINK = "hinge"


class LogisticRegressionWithAGD(_GLMTrainer):
    GRADIENT_CLS = LogisticGradient
    LINK = "logistic"


class LinearRegressionWithAGD(_GLMTrainer):
    GRADIENT_CLS = LeastSquaresGradient
    LINK = "identity"


class SVMWithAGD(_GLMTrainer):
    GRADIENT_CLS = HingeGradient
    LINK = "hinge"
