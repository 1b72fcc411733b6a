"""Shape/layout manipulation ops (reference: python/paddle/tensor/manipulation.py).

Paddle's `axis` keyword maps to torch's `dim`; reshape/transpose follow
paddle semantics (transpose takes a full perm list).
"""
from __future__ import annotations

import torch


def _resolve_shape(x, shape):
    # paddle reshape semantics (manipulation.py reshape docs): 0 copies
    # the corresponding INPUT dim, -1 infers; torch has no 0 rule
    out = []
    for i, s in enumerate(shape):
        s = int(s.item()) if isinstance(s, torch.Tensor) else int(s)
        out.append(x.shape[i] if s == 0 else s)
    return out


def reshape(x, shape, name=None):
    """paddle.reshape: 0 copies the input dim, -1 infers.  (The
    Tensor.reshape METHOD keeps torch semantics -- patching it globally
    would break torch-internal empty-tensor reshapes.)"""
    return torch.reshape(x, _resolve_shape(x, shape))


def reshape_(x, shape, name=None):
    shape = _resolve_shape(x, shape)
    return x.reshape_(*shape) if hasattr(x, "reshape_") else x.view(shape)


def view(x, shape_or_dtype, name=None):
    if isinstance(shape_or_dtype, (list, tuple)):
        return x.view(list(shape_or_dtype))
    from .. import framework
    return x.view(framework.convert_dtype(shape_or_dtype))


def transpose(x, perm, name=None):
    return x.permute(list(perm))


def concat(x, axis=0, name=None):
    if isinstance(axis, torch.Tensor):
        axis = int(axis.item())
    return torch.cat(list(x), dim=axis)


def stack(x, axis=0, name=None):
    return torch.stack(list(x), dim=axis)


def unstack(x, axis=0, num=None):
    return list(torch.unbind(x, dim=axis))


def unbind(x, axis=0):
    return list(torch.unbind(x, dim=axis))


def split(x, num_or_sections, axis=0, name=None):
    if isinstance(axis, torch.Tensor):
        axis = int(axis.item())
    dim_size = x.shape[axis]
    if isinstance(num_or_sections, int):
        chunk = dim_size // num_or_sections
        return list(torch.split(x, chunk, dim=axis))
    sections = [s if s != -1 else dim_size - sum(v for v in num_or_sections if v != -1)
                for s in num_or_sections]
    return list(torch.split(x, sections, dim=axis))


def chunk(x, chunks, axis=0, name=None):
    return list(torch.chunk(x, chunks, dim=axis))


def squeeze(x, axis=None, name=None):
    if axis is None:
        return torch.squeeze(x)
    if isinstance(axis, (list, tuple)):
        for a in sorted([a % x.dim() for a in axis], reverse=True):
            x = torch.squeeze(x, a)
        return x
    return torch.squeeze(x, axis)


def squeeze_(x, axis=None, name=None):
    return squeeze(x, axis)


def unsqueeze(x, axis, name=None):
    if isinstance(axis, (list, tuple)):
        for a in axis:
            x = torch.unsqueeze(x, a)
        return x
    return torch.unsqueeze(x, axis)


def unsqueeze_(x, axis, name=None):
    return x.unsqueeze_(axis)


def flatten(x, start_axis=0, stop_axis=-1, name=None):
    return torch.flatten(x, start_axis, stop_axis)


def flip(x, axis, name=None):
    if isinstance(axis, int):
        axis = [axis]
    return torch.flip(x, axis)


def roll(x, shifts, axis=None, name=None):
    return torch.roll(x, shifts, dims=axis if axis is not None else ())


def tile(x, repeat_times, name=None):
    return x.repeat(list(repeat_times))


def expand(x, shape, name=None):
    return x.expand(list(shape))


def expand_as(x, y, name=None):
    return x.expand_as(y)


def broadcast_to(x, shape, name=None):
    return torch.broadcast_to(x, list(shape))


def cast(x, dtype):
    from .. import framework
    return x.to(framework.convert_dtype(dtype))


def gather(x, index, axis=0, name=None):
    # paddle.gather == torch.index_select
    if index.dim() > 1:
        index = index.flatten()
    return torch.index_select(x, axis, index)


def gather_nd(x, index, name=None):
    idx = index.long()
    out_shape = list(idx.shape[:-1]) + list(x.shape[idx.shape[-1]:])
    flat_idx = idx.reshape(-1, idx.shape[-1])
    res = x[tuple(flat_idx[:, i] for i in range(flat_idx.shape[1]))]
    return res.reshape(out_shape)


def scatter(x, index, updates, overwrite=True, name=None):
    out = x.clone()
    if overwrite:
        out[index.long()] = updates
    else:
        out[index.long()] = 0
        out.index_add_(0, index.long(), updates)
    return out


def index_select(x, index, axis=0, name=None):
    return torch.index_select(x, axis, index.long())


def masked_select(x, mask, name=None):
    return torch.masked_select(x, mask)


def take_along_axis(arr, indices, axis, broadcast=True):
    idx = indices.long()
    if broadcast:
        shape = list(arr.shape)
        shape[axis] = idx.shape[axis]
        idx = idx.expand(shape) if idx.shape != tuple(shape) else idx
    return torch.gather(arr, axis, idx)


def put_along_axis(arr, indices, values, axis, reduce="assign", include_self=True, broadcast=True):
    idx = indices.long()
    if not isinstance(values, torch.Tensor):
        values = torch.full_like(idx, values, dtype=arr.dtype)
    values = values.to(arr.dtype)
    if values.shape != idx.shape:
        values = values.expand_as(idx)
    out = arr.clone()
    if reduce == "assign":
        return out.scatter_(axis, idx, values)
    if reduce in ("add", "mul", "amin", "amax", "mean"):
        red = {"add": "sum", "mul": "prod", "amin": "amin", "amax": "amax", "mean": "mean"}[reduce]
        return out.scatter_reduce_(axis, idx, values, reduce=red, include_self=include_self)
    raise ValueError(reduce)


def where(condition, x=None, y=None, name=None):
    if x is None and y is None:
        return torch.nonzero(condition, as_tuple=False)
    return torch.where(condition, x, y)


def numel(x, name=None):
    return torch.tensor(x.numel())


def shape(x):
    return torch.tensor(list(x.shape), dtype=torch.int64)


def slice(x, axes, starts, ends):
    out = x
    for ax, s, e in zip(axes, starts, ends):
        if isinstance(s, torch.Tensor):
            s = int(s.item())
        if isinstance(e, torch.Tensor):
            e = int(e.item())
        out = out.narrow(ax, s if s >= 0 else out.shape[ax] + s,
                         min(e, out.shape[ax]) - (s if s >= 0 else out.shape[ax] + s))
    return out


def unique(x, return_index=False, return_inverse=False, return_counts=False, axis=None, dtype="int64", name=None):
    res = torch.unique(x, sorted=True, return_inverse=return_inverse,
                       return_counts=return_counts, dim=axis)
    if not (return_index or return_inverse or return_counts):
        return res if isinstance(res, torch.Tensor) else res[0]
    outs = list(res) if isinstance(res, tuple) else [res]
    if return_index:
        vals = outs[0]
        flat = x.flatten() if axis is None else x
        # first-occurrence index
        idx = torch.stack([(flat == v).nonzero()[0, 0] for v in vals]) if axis is None else None
        outs.insert(1, idx)
    return tuple(outs)


def repeat_interleave(x, repeats, axis=None, name=None):
    return torch.repeat_interleave(x, repeats, dim=axis)
