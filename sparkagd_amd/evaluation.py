"""Model evaluation metrics (MLlib's BinaryClassificationMetrics /
MulticlassMetrics analog, GPU-resident).

The reference ships no metrics in-repo, but an MLlib 1.3 user evaluates the
trained GLMs with ``mllib.evaluation``; these are the equivalents a user
switching frameworks expects. All functions take plain tensors (margins /
scores / labels) and run on whatever device they live on.
"""

from __future__ import annotations

from typing import Dict

import torch


def accuracy(pred: torch.Tensor, labels: torch.Tensor) -> float:
    """Fraction of exact prediction/label matches."""
    return float((pred.to(torch.float32) == labels.to(torch.float32))
                 .to(torch.float64).mean())


def log_loss(margins: torch.Tensor, labels: torch.Tensor) -> float:
    """Mean binary logistic loss from margins z = <w, x> (labels in {0,1})."""
    z = margins.to(torch.float64)
    y = labels.to(torch.float64)
    # log(1+e^-z) for y=1, log(1+e^z) for y=0, computed stably
    return float((torch.nn.functional.softplus(-z) * y
                  + torch.nn.functional.softplus(z) * (1.0 - y)).mean())


def roc_auc(scores: torch.Tensor, labels: torch.Tensor) -> float:
    """Area under the ROC curve via the rank statistic
    AUC = (U - n_pos(n_pos+1)/2) / (n_pos * n_neg), with midrank tie
    handling — equivalent to trapezoidal integration over all thresholds."""
    s = scores.to(torch.float64).flatten()
    y = (labels.to(torch.float64).flatten() > 0.5)
    n_pos = int(y.sum())
    n_neg = y.numel() - n_pos
    if n_pos == 0 or n_neg == 0:
        raise ValueError("roc_auc needs both classes present")
    order = torch.argsort(s)
    sorted_s = s[order]
    ranks = torch.empty_like(s)
    # midranks for ties
    uniq, inv, counts = torch.unique(sorted_s, return_inverse=True,
                                     return_counts=True)
    cum = torch.cumsum(counts.to(torch.float64), 0)
    mid = cum - (counts.to(torch.float64) - 1) / 2.0
    ranks[order] = mid[inv]
    u = float(ranks[y].sum())
    return (u - n_pos * (n_pos + 1) / 2.0) / (n_pos * n_neg)


def precision_recall_f1(pred: torch.Tensor, labels: torch.Tensor) -> Dict[str, float]:
    """Binary precision/recall/F1 (positive class = 1)."""
    p = pred.to(torch.bool)
    y = labels.to(torch.float32) > 0.5
    tp = float((p & y).sum())
    fp = float((p & ~y).sum())
    fn = float((~p & y).sum())
    prec = tp / (tp + fp) if tp + fp > 0 else 0.0
    rec = tp / (tp + fn) if tp + fn > 0 else 0.0
    f1 = 2 * prec * rec / (prec + rec) if prec + rec > 0 else 0.0
    return {"precision": prec, "recall": rec, "f1": f1}


def confusion_matrix(pred: torch.Tensor, labels: torch.Tensor,
                     num_classes: int) -> torch.Tensor:
    """[K, K] counts; rows = true class, cols = predicted."""
    idx = (labels.to(torch.int64) * num_classes + pred.to(torch.int64))
    return torch.bincount(idx, minlength=num_classes * num_classes).reshape(
        num_classes, num_classes)
