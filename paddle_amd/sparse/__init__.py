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
