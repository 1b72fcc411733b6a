"""paddle.sparse parity subset (reference: python/paddle/sparse/ --
COO/CSR tensors + elementwise/matmul)."""
from __future__ import annotations

import torch


def sparse_coo_tensor(indices, values, shape=None, dtype=None, place=None,
                      stop_gradient=True):
    idx = indices if isinstance(indices, torch.Tensor) else torch.as_tensor(indices)
    vals = values if isinstance(values, torch.Tensor) else torch.as_tensor(values)
    t = torch.sparse_coo_tensor(idx.long(), vals, size=shape)
    return t.coalesce()


def sparse_csr_tensor(crows, cols, values, shape=None, dtype=None, place=None,
                      stop_gradient=True):
    return torch.sparse_csr_tensor(
        torch.as_tensor(crows).long(), torch.as_tensor(cols).long(),
        torch.as_tensor(values), size=shape)


def is_sparse_coo(x):
    return x.layout == torch.sparse_coo


def is_sparse_csr(x):
    return x.layout == torch.sparse_csr


def matmul(x, y, name=None):
    return torch.sparse.mm(x, y) if x.layout != torch.strided else torch.matmul(x, y)


def masked_matmul(x, y, mask, name=None):
    return torch.sparse.sampled_addmm(
        torch.zeros_like(mask) if mask.layout == torch.sparse_csr else mask, x, y)


def add(x, y, name=None):
    return x + y


def multiply(x, y, name=None):
    return x * y


def to_dense(x):
    return x.to_dense()


def to_sparse_coo(x, sparse_dim=None):
    return x.to_sparse(sparse_dim) if sparse_dim else x.to_sparse()


def to_sparse_csr(x):
    return x.to_sparse_csr()


class nn:
    """paddle.sparse.nn namespace placeholder (conv3d etc. are later work)."""

    class ReLU(torch.nn.Module):
        def forward(self, x):
            if x.layout == torch.sparse_coo:
                return torch.sparse_coo_tensor(x.indices(), torch.relu(x.values()),
                                               x.shape)
            return torch.relu(x)


# -- elementwise/unary over sparse values (reference: sparse/unary.py) -------
import torch as _t


def _unary(fn):
    def g(x, *a, **kw):
        kw.pop("name", None)
        if x.is_sparse:
            xc = x.coalesce()
            return _t.sparse_coo_tensor(xc.indices(), fn(xc.values(), *a, **kw),
                                        xc.shape)
        if x.layout == _t.sparse_csr:
            return _t.sparse_csr_tensor(x.crow_indices(), x.col_indices(),
                                        fn(x.values(), *a, **kw), x.shape)
        return fn(x, *a, **kw)
    g.__name__ = fn.__name__
    return g


sin = _unary(_t.sin)
tan = _unary(_t.tan)
asin = _unary(_t.asin)
atan = _unary(_t.atan)
sinh = _unary(_t.sinh)
tanh = _unary(_t.tanh)
asinh = _unary(_t.asinh)
atanh = _unary(_t.atanh)
sqrt = _unary(_t.sqrt)
square = _unary(_t.square)
log1p = _unary(_t.log1p)
abs = _unary(_t.abs)
pow = _unary(_t.pow)
neg = _unary(_t.neg)
deg2rad = _unary(_t.deg2rad)
rad2deg = _unary(_t.rad2deg)
expm1 = _unary(_t.expm1)
isnan = _unary(_t.isnan)


def cast(x, index_dtype=None, value_dtype=None, name=None):
    from .. import framework as _fw
    xc = x.coalesce() if x.is_sparse else x
    vals = xc.values()
    if value_dtype is not None:
        vals = vals.to(_fw.convert_dtype(value_dtype))
    idx = xc.indices()
    if index_dtype is not None:
        idx = idx.to(_fw.convert_dtype(index_dtype))
    return _t.sparse_coo_tensor(idx, vals, xc.shape)


def mv(x, vec, name=None):
    return _t.mv(x, vec)


def addmm(input, x, y, beta=1.0, alpha=1.0, name=None):
    return _t.sparse.addmm(input, x, y, beta=beta, alpha=alpha)


def subtract(x, y, name=None):
    return (x - y).coalesce() if x.is_sparse else x - y


def divide(x, y, name=None):
    if x.is_sparse:
        xc = x.coalesce()
        yv = y.coalesce().values() if isinstance(y, _t.Tensor) and y.is_sparse else y
        return _t.sparse_coo_tensor(xc.indices(), xc.values() / yv, xc.shape)
    return x / y


def transpose(x, perm, name=None):
    assert len(perm) == 2, "sparse transpose: 2-D only in this build"
    return x.t().coalesce() if x.is_sparse else x.permute(perm)


def sum(x, axis=None, dtype=None, keepdim=False, name=None):
    if axis is None:
        return _t.sparse.sum(x) if x.is_sparse else x.sum()
    return _t.sparse.sum(x, dim=axis)


def coalesce(x, name=None):
    return x.coalesce()


def is_same_shape(x, y):
    return list(x.shape) == list(y.shape)


def reshape(x, shape, name=None):
    return x.reshape(shape)


def slice(x, axes, starts, ends, name=None):
    d = x.to_dense()
    idx = [__builtins__['slice'](None)] * d.dim() if isinstance(__builtins__, dict) else [None]
    import builtins
    idx = [builtins.slice(None)] * d.dim()
    for ax, st, en in zip(axes, starts, ends):
        idx[ax] = builtins.slice(st, en)
    return d[tuple(idx)].to_sparse()


def mask_as(x, mask, name=None):
    mc = mask.coalesce()
    dense = x.to_dense() if x.is_sparse else x
    vals = dense[tuple(mc.indices())]
    return _t.sparse_coo_tensor(mc.indices(), vals, dense.shape)


def pca_lowrank(x, q=None, center=True, niter=2, name=None):
    return _t.pca_lowrank(x, q=q, center=center, niter=niter)
