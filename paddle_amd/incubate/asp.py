"""Automatic SParsity: 2:4 semi-structured pruning
(reference: python/paddle/incubate/asp/__init__.py, asp.py).

CDNA4 MFMA supports 4:2 structured sparsity on the A operand (the
headline "with sparsity" FLOPS numbers); this module produces the 2:4
masks.  `prune_model` applies magnitude-based 2-out-of-4 masks to every
Linear weight; `decorate` wraps an optimizer so masks are re-applied
after each step (masked weights stay zero through training).
"""
from __future__ import annotations

import torch

_masks: dict[int, torch.Tensor] = {}
_excluded: set[str] = set()


def set_excluded_layers(model, param_names=()):
    """Mark parameter names to skip during pruning."""
    _excluded.update(param_names)


def reset_excluded_layers(model=None):
    _excluded.clear()


def _mask_2to4(w: torch.Tensor) -> torch.Tensor:
    """2:4 mask along the input (last) dim: keep top-2 |w| per group of 4."""
    out_f, in_f = w.shape
    pad = (-in_f) % 4
    wp = torch.nn.functional.pad(w.abs().float(), (0, pad))
    g = wp.reshape(out_f, -1, 4)
    idx = g.topk(2, dim=-1).indices
    m = torch.zeros_like(g, dtype=torch.bool).scatter_(-1, idx, True)
    return m.reshape(out_f, -1)[:, :in_f]


def prune_model(model, n=2, m=4, mask_algo="mask_1d", with_mask=True):
    """Apply 2:4 magnitude masks to 2-D weights of the model in place."""
    pruned = {}
    for name, p in model.named_parameters():
        if p.dim() != 2 or min(p.shape) < 4 or name in _excluded:
            continue
        if "bias" in name or "norm" in name or "embedding" in name.lower():
            continue
        mask = _mask_2to4(p.detach())
        with torch.no_grad():
            p.mul_(mask)
        _masks[id(p)] = mask
        pruned[name] = mask
    return pruned


def decorate(optimizer):
    """Wrap optimizer.step so pruned weights stay exactly zero."""
    inner_step = optimizer.step

    def step(*a, **kw):
        out = inner_step(*a, **kw)
        with torch.no_grad():
            for group_params in _iter_params(optimizer):
                mask = _masks.get(id(group_params))
                if mask is not None:
                    group_params.mul_(mask.to(group_params.device))
        return out

    optimizer.step = step
    return optimizer


def _iter_params(optimizer):
    if hasattr(optimizer, "_params"):  # paddle_amd.optimizer.Optimizer
        yield from optimizer._params
        return
    torch_opt = getattr(optimizer, "_opt", optimizer)
    if hasattr(torch_opt, "param_groups"):
        for g in torch_opt.param_groups:
            yield from g["params"]


def calculate_density(t: torch.Tensor) -> float:
    return float((t != 0).float().mean())


def check_sparsity(t: torch.Tensor, n=2, m=4) -> bool:
    """True if every group of `m` along the last dim has ≤ n nonzeros."""
    in_f = t.shape[-1]
    pad = (-in_f) % m
    tp = torch.nn.functional.pad(t.float(), (0, pad))
    g = tp.reshape(*tp.shape[:-1], -1, m)
    return bool(((g != 0).sum(-1) <= n).all())
