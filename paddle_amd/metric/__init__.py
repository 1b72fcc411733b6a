"""paddle.metric parity (reference: python/paddle/metric/metrics.py)."""
from __future__ import annotations

import numpy as np
import torch


class Metric:
    def reset(self):  # pragma: no cover - abstract
        raise NotImplementedError

    def update(self, *args):  # pragma: no cover - abstract
        raise NotImplementedError

    def accumulate(self):  # pragma: no cover - abstract
        raise NotImplementedError

    def name(self):
        return self.__class__.__name__.lower()


class Accuracy(Metric):
    def __init__(self, topk=(1,), name=None):
        self.topk = topk if isinstance(topk, (list, tuple)) else (topk,)
        self._name = name or "acc"
        self.reset()

    def reset(self):
        self.total = [0.0] * len(self.topk)
        self.count = [0] * len(self.topk)

    def compute(self, pred, label):
        maxk = max(self.topk)
        if label.dim() == pred.dim():
            label = label.argmax(-1)
        _, idx = torch.topk(pred, maxk, dim=-1)
        correct = idx == label.unsqueeze(-1)
        return correct.float()

    def update(self, correct):
        if isinstance(correct, torch.Tensor):
            correct = correct.detach().cpu().numpy()
        n = correct.shape[0]
        res = []
        for i, k in enumerate(self.topk):
            c = correct[..., :k].sum()
            self.total[i] += float(c)
            self.count[i] += n
            res.append(float(c) / n)
        return res[0] if len(res) == 1 else res

    def accumulate(self):
        res = [t / c if c else 0.0 for t, c in zip(self.total, self.count)]
        return res[0] if len(res) == 1 else res

    def name(self):
        return self._name


class Precision(Metric):
    def __init__(self, name="precision"):
        self._name = name
        self.reset()

    def reset(self):
        self.tp = 0
        self.fp = 0

    def update(self, preds, labels):
        p = (np.asarray(preds) > 0.5).astype(int).flatten()
        l = np.asarray(labels).astype(int).flatten()
        self.tp += int(((p == 1) & (l == 1)).sum())
        self.fp += int(((p == 1) & (l == 0)).sum())

    def accumulate(self):
        d = self.tp + self.fp
        return self.tp / d if d else 0.0

    def name(self):
        return self._name


class Recall(Metric):
    def __init__(self, name="recall"):
        self._name = name
        self.reset()

    def reset(self):
        self.tp = 0
        self.fn = 0

    def update(self, preds, labels):
        p = (np.asarray(preds) > 0.5).astype(int).flatten()
        l = np.asarray(labels).astype(int).flatten()
        self.tp += int(((p == 1) & (l == 1)).sum())
        self.fn += int(((p == 0) & (l == 1)).sum())

    def accumulate(self):
        d = self.tp + self.fn
        return self.tp / d if d else 0.0

    def name(self):
        return self._name


class Auc(Metric):
    def __init__(self, curve="ROC", num_thresholds=4095, name="auc"):
        self._name = name
        self.num_thresholds = num_thresholds
        self.reset()

    def reset(self):
        self._stat_pos = np.zeros(self.num_thresholds + 1)
        self._stat_neg = np.zeros(self.num_thresholds + 1)

    def update(self, preds, labels):
        preds = np.asarray(preds)
        labels = np.asarray(labels).flatten()
        pos_prob = preds[:, 1] if preds.ndim == 2 and preds.shape[1] == 2 else preds.flatten()
        bins = (pos_prob * self.num_thresholds).astype(int).clip(0, self.num_thresholds)
        for b, l in zip(bins, labels):
            if l:
                self._stat_pos[b] += 1
            else:
                self._stat_neg[b] += 1

    def accumulate(self):
        tot_pos = tot_neg = auc = 0.0
        for i in range(self.num_thresholds, -1, -1):
            auc += self._stat_neg[i] * (tot_pos + self._stat_pos[i] / 2)
            tot_pos += self._stat_pos[i]
            tot_neg += self._stat_neg[i]
        return auc / (tot_pos * tot_neg) if tot_pos and tot_neg else 0.0

    def name(self):
        return self._name


def accuracy(input, label, k=1, correct=None, total=None, name=None):
    if label.dim() == 2 and label.shape[-1] == 1:
        label = label.squeeze(-1)
    _, idx = torch.topk(input, k, dim=-1)
    correct_ = (idx == label.unsqueeze(-1)).any(-1).float()
    return correct_.mean()
