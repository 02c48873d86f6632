"""GLUE metrics.

Reference: ppfleetx/models/language_model/metrics.py — AccuracyAndF1 :31,
Mcc :180, PearsonAndSpearman :305, MultiLabelsMetric :445. Same streaming
accumulate/compute contract (update with batches, name() lists metric
names, accumulate() returns the tuple).
"""

from __future__ import annotations

import math
from typing import List, Tuple

import torch

__all__ = ["Accuracy", "AccuracyAndF1", "Mcc", "PearsonAndSpearman",
           "MultiLabelsMetric"]


class Metric:
    def reset(self):
        raise NotImplementedError

    def update(self, preds, labels):
        raise NotImplementedError

    def accumulate(self):
        raise NotImplementedError

    def name(self):
        raise NotImplementedError


class Accuracy(Metric):
    def __init__(self):
        self.reset()

    def reset(self):
        self.correct = 0
        self.total = 0

    def update(self, preds: torch.Tensor, labels: torch.Tensor):
        if preds.ndim > 1:
            preds = preds.argmax(dim=-1)
        self.correct += int((preds == labels).sum())
        self.total += labels.numel()

    def accumulate(self) -> float:
        return self.correct / max(1, self.total)

    def name(self):
        return "acc"


class AccuracyAndF1(Metric):
    """acc, precision, recall, f1, (acc+f1)/2 (metrics.py:31-178)."""

    def __init__(self, pos_label: int = 1):
        self.pos_label = pos_label
        self.reset()

    def reset(self):
        self.tp = self.fp = self.fn = self.tn = 0

    def update(self, preds: torch.Tensor, labels: torch.Tensor):
        if preds.ndim > 1:
            preds = preds.argmax(dim=-1)
        p = preds == self.pos_label
        l = labels == self.pos_label
        self.tp += int((p & l).sum())
        self.fp += int((p & ~l).sum())
        self.fn += int((~p & l).sum())
        self.tn += int((~p & ~l).sum())

    def accumulate(self) -> Tuple[float, float, float, float, float]:
        total = self.tp + self.fp + self.fn + self.tn
        acc = (self.tp + self.tn) / max(1, total)
        precision = self.tp / max(1, self.tp + self.fp)
        recall = self.tp / max(1, self.tp + self.fn)
        f1 = 2 * precision * recall / max(1e-12, precision + recall)
        return acc, precision, recall, f1, (acc + f1) / 2

    def name(self):
        return "acc", "precision", "recall", "f1", "acc and f1"


class Mcc(Metric):
    """Matthews correlation (metrics.py:180-303)."""

    def __init__(self):
        self.reset()

    def reset(self):
        self.tp = self.fp = self.fn = self.tn = 0

    def update(self, preds: torch.Tensor, labels: torch.Tensor):
        if preds.ndim > 1:
            preds = preds.argmax(dim=-1)
        p = preds == 1
        l = labels == 1
        self.tp += int((p & l).sum())
        self.fp += int((p & ~l).sum())
        self.fn += int((~p & l).sum())
        self.tn += int((~p & ~l).sum())

    def accumulate(self) -> Tuple[float]:
        num = self.tp * self.tn - self.fp * self.fn
        den = math.sqrt(max(1e-12,
                            (self.tp + self.fp) * (self.tp + self.fn) *
                            (self.tn + self.fp) * (self.tn + self.fn)))
        return (num / den,)

    def name(self):
        return ("mcc",)


def _rank(x: List[float]) -> List[float]:
    order = sorted(range(len(x)), key=lambda i: x[i])
    ranks = [0.0] * len(x)
    i = 0
    while i < len(order):
        j = i
        while j + 1 < len(order) and x[order[j + 1]] == x[order[i]]:
            j += 1
        avg = (i + j) / 2.0 + 1.0
        for k2 in range(i, j + 1):
            ranks[order[k2]] = avg
        i = j + 1
    return ranks


def _pearson(a: List[float], b: List[float]) -> float:
    n = len(a)
    ma = sum(a) / n
    mb = sum(b) / n
    cov = sum((x - ma) * (y - mb) for x, y in zip(a, b))
    va = math.sqrt(sum((x - ma) ** 2 for x in a))
    vb = math.sqrt(sum((y - mb) ** 2 for y in b))
    return cov / max(1e-12, va * vb)


class PearsonAndSpearman(Metric):
    """pearson, spearman, mean (metrics.py:305-443) — STS-B regression."""

    def __init__(self):
        self.reset()

    def reset(self):
        self.preds: List[float] = []
        self.labels: List[float] = []

    def update(self, preds: torch.Tensor, labels: torch.Tensor):
        self.preds.extend(preds.reshape(-1).float().tolist())
        self.labels.extend(labels.reshape(-1).float().tolist())

    def accumulate(self) -> Tuple[float, float, float]:
        p = _pearson(self.preds, self.labels)
        s = _pearson(_rank(self.preds), _rank(self.labels))
        return p, s, (p + s) / 2

    def name(self):
        return "pearson", "spearman", "pearson and spearman"


class MultiLabelsMetric(Metric):
    """Per-label precision/recall/F1 with None/binary/micro/macro/weighted
    averaging over streaming per-label confusion matrices
    (metrics.py:445-620 semantics)."""

    def __init__(self, num_labels: int):
        assert num_labels > 1, "num_labels must be > 1"
        self.num_labels = num_labels
        self.reset()

    def reset(self):
        # [label][truth][pred] 2x2 one-vs-rest confusion counts
        self.cm = torch.zeros(self.num_labels, 2, 2, dtype=torch.long)

    def update(self, preds: torch.Tensor, labels: torch.Tensor):
        if preds.ndim > 1:
            preds = preds.argmax(dim=-1)
        preds = preds.reshape(-1)
        labels = labels.reshape(-1)
        for c in range(self.num_labels):
            p = preds == c
            t = labels == c
            self.cm[c, 1, 1] += int((p & t).sum())
            self.cm[c, 1, 0] += int((~p & t).sum())
            self.cm[c, 0, 1] += int((p & ~t).sum())
            self.cm[c, 0, 0] += int((~p & ~t).sum())

    @staticmethod
    def _prf(tp, fp, fn):
        precision = tp / (tp + fp) if tp + fp > 0 else 0.0
        recall = tp / (tp + fn) if tp + fn > 0 else 0.0
        f1 = 2 * precision * recall / (precision + recall) \
            if precision + recall > 0 else 0.0
        return precision, recall, f1

    def accumulate(self, average=None, pos_label: int = 1):
        tp = self.cm[:, 1, 1].float()
        fp = self.cm[:, 0, 1].float()
        fn = self.cm[:, 1, 0].float()
        if average == "binary":
            return self._prf(float(tp[pos_label]), float(fp[pos_label]),
                             float(fn[pos_label]))
        if average == "micro":
            return self._prf(float(tp.sum()), float(fp.sum()),
                             float(fn.sum()))
        per = [self._prf(float(tp[c]), float(fp[c]), float(fn[c]))
               for c in range(self.num_labels)]
        if average == "macro":
            n = self.num_labels
            return (sum(p for p, _, _ in per) / n,
                    sum(r for _, r, _ in per) / n,
                    sum(f for _, _, f in per) / n)
        if average == "weighted":
            support = (tp + fn)
            tot = float(support.sum()) or 1.0
            w = [float(s) / tot for s in support]
            return (sum(wi * p for wi, (p, _, _) in zip(w, per)),
                    sum(wi * r for wi, (_, r, _) in zip(w, per)),
                    sum(wi * f for wi, (_, _, f) in zip(w, per)))
        # average=None: per-label arrays
        return ([p for p, _, _ in per], [r for _, r, _ in per],
                [f for _, _, f in per])

    def name(self):
        return "precision", "recall", "f1"
