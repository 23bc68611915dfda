"""Training-report plots (the reference notebook's visual outputs).

The training notebook renders learning curves (cell 30) and per-class
confusion-matrix heatmaps (cells 31/37). These reproduce them headlessly
(Agg backend) from the JSONL history / confusion lists the training driver
already produces, so `python -m fmda_amd.train --plots DIR` leaves the same
artifacts a notebook run would.
"""
from typing import Dict, List, Optional, Sequence

import matplotlib

matplotlib.use("Agg")
import matplotlib.pyplot as plt  # noqa: E402

from .features import TARGET_NAMES  # noqa: E402


def learning_curves(history: List[Dict], path: str) -> None:
    """Train/val subset accuracy + train loss over epochs (notebook
    cell 30's two-panel figure)."""
    epochs = [r["epoch"] for r in history if "epoch" in r]
    if not epochs:
        return
    tr_acc = [r["train_acc"] for r in history if "epoch" in r]
    va_acc = [r["val_acc"] for r in history if "epoch" in r]
    tr_loss = [r["train_loss"] for r in history if "epoch" in r]

    fig, (ax1, ax2) = plt.subplots(1, 2, figsize=(10, 4))
    ax1.plot(epochs, tr_acc, label="train")
    ax1.plot(epochs, va_acc, label="val")
    ax1.set_xlabel("epoch")
    ax1.set_ylabel("subset accuracy")
    ax1.legend()
    ax1.set_title("accuracy")
    ax2.plot(epochs, tr_loss, color="tab:red")
    ax2.set_xlabel("epoch")
    ax2.set_ylabel("train loss")
    ax2.set_title("loss")
    fig.tight_layout()
    fig.savefig(path, dpi=110)
    plt.close(fig)


def confusion_heatmaps(confusion: Sequence[Sequence[Sequence[int]]],
                       path: str,
                       labels: Optional[List[str]] = None) -> None:
    """Per-class 2x2 confusion heatmaps ([[tn, fp], [fn, tp]] per class,
    the notebook cells 31/37 layout)."""
    labels = labels or list(TARGET_NAMES)
    n = len(confusion)
    fig, axes = plt.subplots(1, n, figsize=(3 * n, 3))
    if n == 1:
        axes = [axes]
    for ax, cm, name in zip(axes, confusion, labels):
        ax.imshow(cm, cmap="Blues")
        for i in range(2):
            for j in range(2):
                ax.text(j, i, str(cm[i][j]), ha="center", va="center")
        ax.set_title(name)
        ax.set_xticks([0, 1], ["pred 0", "pred 1"])
        ax.set_yticks([0, 1], ["true 0", "true 1"])
    fig.tight_layout()
    fig.savefig(path, dpi=110)
    plt.close(fig)
