"""Random sampling ops (reference: python/paddle/tensor/random.py)."""
from __future__ import annotations

import torch


def multinomial(x, num_samples=1, replacement=False, name=None):
    return torch.multinomial(x, num_samples, replacement)


def bernoulli(x, name=None):
    return torch.bernoulli(x)


def poisson(x, name=None):
    return torch.poisson(x)
