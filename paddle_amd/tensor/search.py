"""Search/indexing ops (reference: python/paddle/tensor/search.py)."""
from __future__ import annotations

import torch


def nonzero(x, as_tuple=False):
    return torch.nonzero(x, as_tuple=as_tuple)


def index_sample(x, index):
    return torch.gather(x, 1, index.long())


def masked_fill(x, mask, value, name=None):
    return x.masked_fill(mask, value)


def searchsorted(sorted_sequence, values, out_int32=False, right=False, name=None):
    return torch.searchsorted(sorted_sequence, values, out_int32=out_int32, right=right)
