"""Quality-metric formulas identical to the reference.

The reference computes sklearn subset accuracy, Hamming loss and fbeta
(beta=0.5 in train/eval loops — biGRU_model.py:215-222, 274-281). These
implementations reproduce the sklearn formulas exactly for binary multilabel
indicator input, with a torch fast path that can run on-GPU (no host sync in
the training hot loop); `test_metrics.py` validates them against sklearn.
"""
from typing import Tuple

import torch


def subset_accuracy(target: torch.Tensor, pred: torch.Tensor) -> torch.Tensor:
    """sklearn.metrics.accuracy_score on multilabel data: exact-match ratio."""
    t = target.to(torch.bool)
    p = pred.to(torch.bool)
    return (t == p).all(dim=1).float().mean()


def hamming(target: torch.Tensor, pred: torch.Tensor) -> torch.Tensor:
    """sklearn.metrics.hamming_loss: fraction of wrong labels."""
    t = target.to(torch.bool)
    p = pred.to(torch.bool)
    return (t != p).float().mean()


def fbeta_per_class(target: torch.Tensor, pred: torch.Tensor,
                    beta: float = 0.5) -> torch.Tensor:
    """sklearn.metrics.fbeta_score(average=None) on binary indicator columns.

    sklearn returns 0 for a class where precision and recall are both
    undefined or the denominator is 0 (zero_division default).
    """
    t = target.to(torch.float32)
    p = pred.to(torch.float32)
    tp = (t * p).sum(dim=0)
    fp = ((1 - t) * p).sum(dim=0)
    fn = (t * (1 - p)).sum(dim=0)
    b2 = beta * beta
    denom = (1 + b2) * tp + b2 * fn + fp
    fbeta = torch.where(denom > 0, (1 + b2) * tp / torch.clamp(denom, min=1e-38),
                        torch.zeros_like(denom))
    return fbeta


def batch_metrics(target: torch.Tensor, pred: torch.Tensor,
                  beta: float = 0.5) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    return (subset_accuracy(target, pred), hamming(target, pred),
            fbeta_per_class(target, pred, beta))


def multilabel_confusion(target: torch.Tensor,
                         pred: torch.Tensor) -> torch.Tensor:
    """sklearn.metrics.multilabel_confusion_matrix: per-class 2x2
    [[tn, fp], [fn, tp]] (the notebook's per-class confusion matrices,
    reference training notebook cells 31/37)."""
    t = target.to(torch.float32)
    p = pred.to(torch.float32)
    tp = (t * p).sum(dim=0)
    fp = ((1 - t) * p).sum(dim=0)
    fn = (t * (1 - p)).sum(dim=0)
    tn = ((1 - t) * (1 - p)).sum(dim=0)
    return torch.stack([tn, fp, fn, tp], dim=1).reshape(-1, 2, 2).long()


def three_class_accuracy(target: torch.Tensor,
                         pred: torch.Tensor) -> torch.Tensor:
    """Derived up/down/stall accuracy over the 4-label head.

    The reference README frames the task as 3-class up/down/stall while the
    implementation is 4-label multilabel (up1, up2, down1, down2 — reference
    create_database.py:166-190); "stall" is the implicit all-zeros labeling.
    This collapses the 4 logits-thresholded labels to {up, down, stall}:
    any up* set -> up, any down* set -> down, both or neither -> stall
    (conflicting signals carry no direction), and scores exact agreement.
    """
    def collapse(x: torch.Tensor) -> torch.Tensor:
        b = x.to(torch.bool)
        up = b[:, 0] | b[:, 1]
        down = b[:, 2] | b[:, 3]
        # 0 = stall (neither, or conflicting up&down), 1 = up, 2 = down
        return torch.where(up & ~down, torch.ones_like(up, dtype=torch.long),
                           torch.where(down & ~up,
                                       torch.full_like(up, 2, dtype=torch.long),
                                       torch.zeros_like(up, dtype=torch.long)))

    return (collapse(target) == collapse(pred)).float().mean()
