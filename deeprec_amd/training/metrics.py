"""Streaming eval metrics (reference: tf.metrics.accuracy / tf.metrics.auc
with num_thresholds buckets, used by every modelzoo train.py)."""
from __future__ import annotations

import torch


class StreamingAUC:
    """Histogram-bucketed AUC, equivalent to tf.metrics.auc(
    num_thresholds=N) up to bucketing resolution."""

    def __init__(self, num_thresholds: int = 1000):
        self.n = num_thresholds
        self.pos = torch.zeros(num_thresholds)
        self.neg = torch.zeros(num_thresholds)

    def update(self, probs: torch.Tensor, labels: torch.Tensor):
        probs = probs.detach().float().cpu().clamp(0, 1 - 1e-9)
        labels = labels.detach().float().cpu()
        idx = (probs * self.n).long()
        self.pos.index_add_(0, idx, labels)
        self.neg.index_add_(0, idx, 1.0 - labels)

    def result(self) -> float:
        # TPR/FPR over descending thresholds; trapezoid integration
        pos_rev = torch.flip(self.pos, [0]).cumsum(0)
        neg_rev = torch.flip(self.neg, [0]).cumsum(0)
        p_total, n_total = self.pos.sum(), self.neg.sum()
        if p_total == 0 or n_total == 0:
            return 0.5
        tpr = torch.cat([torch.zeros(1), pos_rev / p_total])
        fpr = torch.cat([torch.zeros(1), neg_rev / n_total])
        return float(torch.trapz(tpr, fpr))


class StreamingAccuracy:
    def __init__(self, threshold: float = 0.5):
        self.threshold = threshold
        self.correct = 0
        self.total = 0

    def update(self, probs: torch.Tensor, labels: torch.Tensor):
        preds = (probs.detach().float() > self.threshold).cpu()
        self.correct += int((preds == labels.cpu().bool()).sum())
        self.total += labels.numel()

    def result(self) -> float:
        return self.correct / max(self.total, 1)
