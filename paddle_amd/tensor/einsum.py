"""einsum (reference: python/paddle/tensor/einsum.py) -- torch backend."""
from __future__ import annotations

import torch


def einsum(equation, *operands):
    if len(operands) == 1 and isinstance(operands[0], (list, tuple)):
        operands = tuple(operands[0])
    return torch.einsum(equation, *operands)
