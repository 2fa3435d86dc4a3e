"""Per-iteration metrics / observability.

The reference's observability is the returned ``lossHistory`` array plus
log4j lines (``AcceleratedGradientDescent.scala:25,227,334-335``) and Spark's
implicit per-job instrumentation. Here: the same returned loss history (it is
load-bearing API — the reference tests count iterations with it,
``Suite.scala:165,181``), Python logging, and a JSONL per-iteration metrics
stream (loss, L, theta, backtrack count, restarts, iteration wall time) that
the benchmark harness consumes. Kernel-level timing belongs to rocprofv3
(profiles/), not this layer.
"""

from __future__ import annotations

import json
import time
from typing import IO, Optional


class NullMetrics:
    def log(self, **kw) -> None:  # noqa: D102
        pass

    def close(self) -> None:  # noqa: D102
        pass


class JsonlMetrics:
    """Append one JSON object per event to ``path`` (rank 0 only by default)."""

    def __init__(self, path: Optional[str], rank: int = 0, enabled_rank: int = 0):
        self._fh: Optional[IO[str]] = None
        if path is not None and rank == enabled_rank:
            self._fh = open(path, "a", buffering=1)
        self._t0 = time.time()

    def log(self, **kw) -> None:
        if self._fh is None:
            return
        kw.setdefault("t", round(time.time() - self._t0, 6))
        self._fh.write(json.dumps(kw) + "\n")

    def close(self) -> None:
        if self._fh is not None:
            self._fh.close()
            self._fh = None

    def __enter__(self) -> "JsonlMetrics":
        return self

    def __exit__(self, *a) -> None:
        self.close()


def iters_to_eps(history, eps: float, loss_star: Optional[float] = None):
    """Iterations to ε-accuracy: the first (1-based) iteration whose recorded
    objective is ≤ (1 + eps)·L*, the distance-to-optimum definition of the
    BASELINE metric's second half (the at-scale version of the reference's
    AGD(10) ≈ GD(50) iteration-advantage contract, ``Suite.scala:60-90``).

    ``loss_star`` defaults to min(history) — callers should pass a history
    long enough to have converged (bench.py extends the run past the timed
    window for exactly this). Returns None if the target is never reached.
    The single shared definition for bench.py and benchmarks/iters_to_eps.py.
    """
    if not history:
        return None
    lstar = min(history) if loss_star is None else loss_star
    target = lstar * (1.0 + eps) + 1e-15  # absolute guard for L* == 0
    for i, v in enumerate(history):
        if v <= target:
            return i + 1
    return None
