"""Binary classification metrics (torchmetrics-equivalent subset).

The reference uses torchmetrics {Accuracy, Precision, Recall, F1Score} with
threshold 0.5 plus PR curves and sklearn classification_report
(base_module.py:34-68, 325-383). This is a device-resident counter
implementation whose state is 4 scalars, so DDP aggregation is a single
all_reduce (collective C3 of SURVEY.md §2.6).
"""

from __future__ import annotations

from typing import Dict, Optional

import torch


class BinaryStats:
    """Accumulates tp/fp/tn/fn at a fixed probability threshold."""

    def __init__(self, threshold: float = 0.5, device: Optional[torch.device] = None):
        self.threshold = threshold
        self.counts = torch.zeros(4, dtype=torch.int64, device=device)  # tp, fp, tn, fn

    def to(self, device):
        self.counts = self.counts.to(device)
        return self

    def update(
        self,
        probs: torch.Tensor,
        labels: torch.Tensor,
        mask: Optional[torch.Tensor] = None,
    ) -> None:
        """mask (same shape, truthy = count) excludes e.g. the dummy graphs a
        capture-padded batch appends (graph/pad.py)."""
        pred = (probs >= self.threshold).to(torch.int64)
        lab = labels.to(torch.int64)
        if mask is not None:
            m = mask.to(torch.bool)
            tp = ((pred == 1) & (lab == 1) & m).sum()
            fp = ((pred == 1) & (lab == 0) & m).sum()
            tn = ((pred == 0) & (lab == 0) & m).sum()
            fn = ((pred == 0) & (lab == 1) & m).sum()
        else:
            tp = ((pred == 1) & (lab == 1)).sum()
            fp = ((pred == 1) & (lab == 0)).sum()
            tn = ((pred == 0) & (lab == 0)).sum()
            fn = ((pred == 0) & (lab == 1)).sum()
        self.counts += torch.stack([tp, fp, tn, fn])

    def reset(self) -> None:
        self.counts.zero_()

    def all_reduce(self) -> None:
        import torch.distributed as dist

        if dist.is_available() and dist.is_initialized():
            dist.all_reduce(self.counts)

    def compute(self) -> Dict[str, float]:
        tp, fp, tn, fn = [float(x) for x in self.counts.tolist()]
        total = tp + fp + tn + fn
        acc = (tp + tn) / total if total else 0.0
        prec = tp / (tp + fp) if (tp + fp) else 0.0
        rec = tp / (tp + fn) if (tp + fn) else 0.0
        f1 = 2 * prec * rec / (prec + rec) if (prec + rec) else 0.0
        return {
            "accuracy": acc,
            "precision": prec,
            "recall": rec,
            "f1": f1,
            "tp": tp,
            "fp": fp,
            "tn": tn,
            "fn": fn,
        }


def pr_curve(probs: torch.Tensor, labels: torch.Tensor, num_thresholds: int = 101):
    """Precision/recall over a threshold sweep (base_module.py:358-361 PR csv
    export equivalent). Returns (thresholds, precision, recall) lists."""
    ths = torch.linspace(0, 1, num_thresholds)
    lab = labels.to(torch.bool)
    precs, recs = [], []
    pos = lab.sum().item()
    for t in ths.tolist():
        pred = probs >= t
        tp = (pred & lab).sum().item()
        fp = (pred & ~lab).sum().item()
        precs.append(tp / (tp + fp) if (tp + fp) else 1.0)
        recs.append(tp / pos if pos else 0.0)
    return ths.tolist(), precs, recs


def classification_report_dict(probs: torch.Tensor, labels: torch.Tensor, threshold=0.5):
    """sklearn.metrics.classification_report equivalent as a dict."""
    out = {}
    pred = (probs >= threshold).to(torch.int64)
    lab = labels.to(torch.int64)
    for cls in (0, 1):
        p_mask = pred == cls
        l_mask = lab == cls
        tp = (p_mask & l_mask).sum().item()
        prec = tp / p_mask.sum().item() if p_mask.sum() else 0.0
        rec = tp / l_mask.sum().item() if l_mask.sum() else 0.0
        f1 = 2 * prec * rec / (prec + rec) if (prec + rec) else 0.0
        out[str(cls)] = {
            "precision": prec,
            "recall": rec,
            "f1-score": f1,
            "support": int(l_mask.sum().item()),
        }
    out["accuracy"] = (pred == lab).float().mean().item()
    return out
