"""Tensor creation ops (reference: python/paddle/tensor/creation.py).

Paddle semantics: created tensors default to stop_gradient=True
(requires_grad=False) and live on the current default place.
"""
from __future__ import annotations

import numpy as np
import torch

from .. import framework


def _dev(place=None):
    return framework._place_from_any(place)


def _dt(dtype, default=None):
    if dtype is None:
        return default
    return framework.convert_dtype(dtype)


def to_tensor(data, dtype=None, place=None, stop_gradient=True):
    dev = _dev(place)
    if isinstance(data, torch.Tensor):
        t = data.to(device=dev)
        if dtype is not None:
            t = t.to(framework.convert_dtype(dtype))
        t = t.clone().detach()
    else:
        if isinstance(data, (list, tuple)) or np.isscalar(data):
            data = np.asarray(data)
        if isinstance(data, np.ndarray) and data.dtype == np.float64 and dtype is None:
            # paddle default float is float32
            data = data.astype(np.float32)
        t = torch.as_tensor(data, device=dev)
        if dtype is not None:
            t = t.to(framework.convert_dtype(dtype))
    t.requires_grad_(not stop_gradient and t.is_floating_point())
    return t


def zeros(shape, dtype=None, name=None):
    return torch.zeros(list(shape), dtype=_dt(dtype, torch.float32), device=_dev())


def ones(shape, dtype=None, name=None):
    return torch.ones(list(shape), dtype=_dt(dtype, torch.float32), device=_dev())


def full(shape, fill_value, dtype=None, name=None):
    if isinstance(fill_value, torch.Tensor):
        fill_value = fill_value.item()
    return torch.full(list(shape), fill_value, dtype=_dt(dtype, torch.float32), device=_dev())


def empty(shape, dtype=None, name=None):
    return torch.empty(list(shape), dtype=_dt(dtype, torch.float32), device=_dev())


def zeros_like(x, dtype=None, name=None):
    return torch.zeros_like(x, dtype=_dt(dtype))


def ones_like(x, dtype=None, name=None):
    return torch.ones_like(x, dtype=_dt(dtype))


def full_like(x, fill_value, dtype=None, name=None):
    return torch.full_like(x, fill_value, dtype=_dt(dtype))


def empty_like(x, dtype=None, name=None):
    return torch.empty_like(x, dtype=_dt(dtype))


def arange(start=0, end=None, step=1, dtype=None, name=None):
    if end is None:
        start, end = 0, start
    for v in (start, end, step):
        if isinstance(v, float):
            dtype = dtype or "float32"
    return torch.arange(start, end, step, dtype=_dt(dtype), device=_dev())


def linspace(start, stop, num, dtype=None, name=None):
    return torch.linspace(start, stop, int(num), dtype=_dt(dtype, torch.float32), device=_dev())


def eye(num_rows, num_columns=None, dtype=None, name=None):
    return torch.eye(num_rows, num_columns if num_columns is not None else num_rows,
                     dtype=_dt(dtype, torch.float32), device=_dev())


def diag(x, offset=0, padding_value=0, name=None):
    if padding_value != 0 and x.dim() == 1:
        n = x.numel() + abs(offset)
        out = torch.full((n, n), padding_value, dtype=x.dtype, device=x.device)
        out.diagonal(offset).copy_(x)
        return out
    return torch.diag(x, offset)


def tril(x, diagonal=0, name=None):
    return torch.tril(x, diagonal)


def triu(x, diagonal=0, name=None):
    return torch.triu(x, diagonal)


def meshgrid(*args, **kwargs):
    if len(args) == 1 and isinstance(args[0], (list, tuple)):
        args = args[0]
    return list(torch.meshgrid(*args, indexing="ij"))


def assign(x, output=None):
    if not isinstance(x, torch.Tensor):
        x = to_tensor(x)
    if output is None:
        return x.clone()
    with torch.no_grad():
        output.copy_(x)
    return output


def clone(x, name=None):
    return x.clone()


def rand(shape, dtype=None, name=None):
    return torch.rand(list(shape), dtype=_dt(dtype, torch.float32), device=_dev())


def randn(shape, dtype=None, name=None):
    return torch.randn(list(shape), dtype=_dt(dtype, torch.float32), device=_dev())


def randint(low=0, high=None, shape=(1,), dtype=None, name=None):
    if high is None:
        low, high = 0, low
    return torch.randint(low, high, list(shape), dtype=_dt(dtype, torch.int64), device=_dev())


def randperm(n, dtype=None, name=None):
    return torch.randperm(n, dtype=_dt(dtype, torch.int64), device=_dev())


def normal(mean=0.0, std=1.0, shape=None, name=None):
    if isinstance(mean, torch.Tensor) or isinstance(std, torch.Tensor):
        return torch.normal(mean, std)
    return torch.randn(list(shape), device=_dev()) * std + mean


def uniform(shape, dtype=None, min=-1.0, max=1.0, seed=0, name=None):
    t = torch.rand(list(shape), dtype=_dt(dtype, torch.float32), device=_dev())
    return t * (max - min) + min
