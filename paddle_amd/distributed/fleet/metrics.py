"""fleet.metrics (reference: distributed/fleet/metrics/metric.py --
cross-rank metric reduction helpers)."""
from __future__ import annotations

import torch

from .. import collective as C


def _to_tensor(v):
    return v if isinstance(v, torch.Tensor) else torch.tensor(float(v))


def sum(input, scope=None, util=None):
    t = _to_tensor(input).clone()
    if C.is_initialized():
        C.all_reduce(t)
    return t


def max(input, scope=None, util=None):
    t = _to_tensor(input).clone()
    if C.is_initialized():
        C.all_reduce(t, op=C.ReduceOp.MAX)
    return t


def min(input, scope=None, util=None):
    t = _to_tensor(input).clone()
    if C.is_initialized():
        C.all_reduce(t, op=C.ReduceOp.MIN)
    return t


def mean(input, scope=None, util=None):
    import torch.distributed as dist
    t = _to_tensor(input).clone()
    if C.is_initialized():
        C.all_reduce(t)
        t = t / dist.get_world_size()
    return t


def auc(stat_pos, stat_neg, scope=None, util=None):
    """Distributed AUC from per-rank positive/negative histograms."""
    sp = _to_tensor(stat_pos).clone()
    sn = _to_tensor(stat_neg).clone()
    if C.is_initialized():
        C.all_reduce(sp)
        C.all_reduce(sn)
    pos = sp.flip(0).cumsum(0)
    neg = sn.flip(0).cumsum(0)
    tot_pos, tot_neg = pos[-1], neg[-1]
    area = ((neg[1:] - neg[:-1]) * (pos[1:] + pos[:-1]) / 2).sum()
    denom = (tot_pos * tot_neg).clamp(min=1e-10)
    return (area / denom).item()
