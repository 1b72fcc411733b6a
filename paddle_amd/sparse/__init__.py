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


def _values_op(x, fn):
    if getattr(x, "is_sparse", False):
        xc = x.coalesce()
        return torch.sparse_coo_tensor(xc.indices(), fn(xc.values()), x.shape)
    return fn(x)


class nn:
    """paddle.sparse.nn (reference sparse/nn/__init__.py: activations over
    sparse values; Conv/SubmConv/MaxPool3D run via a dense round-trip --
    correct semantics, sized for the moderate sparse workloads the CPU
    suite exercises; a gather-scatter sparse conv kernel is future work).
    Layouts: paddle sparse conv is channels-last ([N, D, H, W, C])."""

    class ReLU(torch.nn.Module):
        def forward(self, x):
            return _values_op(x, torch.relu)

    class ReLU6(torch.nn.Module):
        def forward(self, x):
            return _values_op(x, torch.nn.functional.relu6)

    class LeakyReLU(torch.nn.Module):
        def __init__(self, negative_slope=0.01):
            super().__init__()
            self.ns = negative_slope

        def forward(self, x):
            return _values_op(x, lambda v: torch.nn.functional.leaky_relu(v, self.ns))

    class Softmax(torch.nn.Module):
        """CSR per-row softmax over stored values (reference semantics)."""

        def __init__(self, axis=-1):
            super().__init__()

        def forward(self, x):
            if getattr(x, "is_sparse", False):
                return torch.sparse.softmax(x.coalesce(), dim=-1)
            if x.layout == torch.sparse_csr:
                coo = x.to_sparse_coo(2) if hasattr(x, "to_sparse_coo") else x.to_sparse()
                return torch.sparse.softmax(coo.coalesce(), dim=-1).to_sparse_csr()
            return torch.softmax(x, dim=-1)

    class BatchNorm(torch.nn.Module):
        """BatchNorm over the stored values' channel dim."""

        def __init__(self, num_features, momentum=0.9, epsilon=1e-5,
                     data_format="NDHWC", **kw):
            super().__init__()
            self.bn = torch.nn.BatchNorm1d(num_features, eps=epsilon,
                                           momentum=1 - momentum)

        def forward(self, x):
            if getattr(x, "is_sparse", False):
                xc = x.coalesce()
                return torch.sparse_coo_tensor(xc.indices(),
                                               self.bn(xc.values()), x.shape)
            return self.bn(x)

    SyncBatchNorm = BatchNorm

    class _SparseConvNd(torch.nn.Module):
        def __init__(self, in_channels, out_channels, kernel_size, stride=1,
                     padding=0, dilation=1, groups=1, subm=False, nd=3,
                     bias_attr=None, **kw):
            super().__init__()
            conv_cls = torch.nn.Conv3d if nd == 3 else torch.nn.Conv2d
            self.conv = conv_cls(in_channels, out_channels, kernel_size,
                                 stride=stride, padding=padding,
                                 dilation=dilation, groups=groups,
                                 bias=bias_attr is not False)
            self.subm = subm
            self.nd = nd

        def forward(self, x):
            # x: sparse COO [N, spatial..., C] (paddle channels-last)
            dense = x.to_dense() if getattr(x, "is_sparse", False) else x
            perm = (0, self.nd + 1) + tuple(range(1, self.nd + 1))
            out = self.conv(dense.permute(*perm))
            inv = (0,) + tuple(range(2, self.nd + 2)) + (1,)
            out = out.permute(*inv).contiguous()
            if self.subm and getattr(x, "is_sparse", False):
                # submanifold: output nonzeros only at input active sites
                mask = (x.to_dense().abs().sum(-1, keepdim=True) > 0)
                out = out * mask
            return out.to_sparse(self.nd + 1) if getattr(x, "is_sparse", False) else out

    class Conv3D(_SparseConvNd):
        def __init__(self, *a, **kw):
            super().__init__(*a, nd=3, subm=False, **kw)

    class SubmConv3D(_SparseConvNd):
        def __init__(self, *a, **kw):
            super().__init__(*a, nd=3, subm=True, **kw)

    class Conv2D(_SparseConvNd):
        def __init__(self, *a, **kw):
            super().__init__(*a, nd=2, subm=False, **kw)

    class SubmConv2D(_SparseConvNd):
        def __init__(self, *a, **kw):
            super().__init__(*a, nd=2, subm=True, **kw)

    class MaxPool3D(torch.nn.Module):
        def __init__(self, kernel_size, stride=None, padding=0, **kw):
            super().__init__()
            self.pool = torch.nn.MaxPool3d(kernel_size, stride, padding)

        def forward(self, x):
            dense = x.to_dense() if getattr(x, "is_sparse", False) else x
            out = self.pool(dense.permute(0, 4, 1, 2, 3))
            out = out.permute(0, 2, 3, 4, 1).contiguous()
            return out.to_sparse(4) if getattr(x, "is_sparse", False) else out


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
