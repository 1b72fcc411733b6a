"""Linear algebra ops (reference: python/paddle/tensor/linalg.py).

matmul dispatches to torch.matmul, which on this stack is hipBLASLt /
rocBLAS -- plain library GEMMs stay on the vendor library; fused
GEMM-epilogue variants live in paddle_amd.incubate.nn.functional and use
our MFMA HIP kernels.
"""
from __future__ import annotations

import torch


def matmul(x, y, transpose_x=False, transpose_y=False, name=None):
    if transpose_x:
        x = x.transpose(-1, -2)
    if transpose_y:
        y = y.transpose(-1, -2)
    return torch.matmul(x, y)


def mm(input, mat2, name=None):
    return torch.matmul(input, mat2)


def bmm(x, y, name=None):
    return torch.bmm(x, y)


def mv(x, vec, name=None):
    return torch.mv(x, vec)


def dot(x, y, name=None):
    if x.dim() == 2:
        return (x * y).sum(-1)
    return torch.dot(x, y)


def outer(x, y, name=None):
    return torch.outer(x.flatten(), y.flatten())


def t(input, name=None):
    return input.t() if input.dim() >= 2 else input


def norm(x, p=None, axis=None, keepdim=False, name=None):
    if p is None:
        p = "fro" if axis is None else 2
    if axis is None:
        return torch.linalg.norm(x.flatten(), ord=2 if p == "fro" else p)
    return torch.linalg.vector_norm(x, ord=p, dim=axis, keepdim=keepdim)


def dist(x, y, p=2, name=None):
    return torch.dist(x, y, p)


def cross(x, y, axis=9, name=None):
    if axis == 9:
        # paddle default: first axis with dim 3
        for i, s in enumerate(x.shape):
            if s == 3:
                axis = i
                break
    return torch.cross(x, y, dim=axis)


def tensordot(x, y, axes=2, name=None):
    return torch.tensordot(x, y, dims=axes)


def einsum(equation, *operands):
    return torch.einsum(equation, *operands)


def transpose(x, perm, name=None):
    return x.permute(list(perm))
