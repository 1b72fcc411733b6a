"""paddle.linalg parity (reference: python/paddle/tensor/linalg.py +
python/paddle/linalg.py exports) -- decompositions/solvers via torch.linalg
(rocSOLVER/hipBLAS on GPU)."""
from __future__ import annotations

import torch

from .tensor.linalg import (  # noqa: F401
    cross,
    dist,
    dot,
    matmul,
    norm,
    t,
)


def cholesky(x, upper=False, name=None):
    c = torch.linalg.cholesky(x)
    return c.transpose(-1, -2).conj() if upper else c


def cholesky_solve(x, y, upper=False, name=None):
    return torch.cholesky_solve(x, y, upper=upper)


def cond(x, p=None, name=None):
    return torch.linalg.cond(x, p=p)


def corrcoef(x, rowvar=True, name=None):
    return torch.corrcoef(x if rowvar else x.t())


def cov(x, rowvar=True, ddof=True, fweights=None, aweights=None, name=None):
    return torch.cov(x if rowvar else x.t(), correction=1 if ddof else 0,
                     fweights=fweights, aweights=aweights)


def det(x, name=None):
    return torch.linalg.det(x)


def eig(x, name=None):
    return torch.linalg.eig(x)


def eigh(x, UPLO="L", name=None):
    return torch.linalg.eigh(x, UPLO=UPLO)


def eigvals(x, name=None):
    return torch.linalg.eigvals(x)


def eigvalsh(x, UPLO="L", name=None):
    return torch.linalg.eigvalsh(x, UPLO=UPLO)


def inv(x, name=None):
    return torch.linalg.inv(x)


def lstsq(x, y, rcond=None, driver=None, name=None):
    r = torch.linalg.lstsq(x, y, rcond=rcond, driver=driver)
    return r.solution, r.residuals, r.rank, r.singular_values


def lu(x, pivot=True, get_infos=False, name=None):
    lu_t, piv, info = torch.linalg.lu_factor_ex(x)
    if get_infos:
        return lu_t, piv, info
    return lu_t, piv


def matrix_power(x, n, name=None):
    return torch.linalg.matrix_power(x, n)


def matrix_rank(x, tol=None, hermitian=False, name=None):
    return torch.linalg.matrix_rank(x, tol=tol, hermitian=hermitian)


def multi_dot(x, name=None):
    return torch.linalg.multi_dot(x)


def pinv(x, rcond=1e-15, hermitian=False, name=None):
    return torch.linalg.pinv(x, rtol=rcond, hermitian=hermitian)


def qr(x, mode="reduced", name=None):
    return torch.linalg.qr(x, mode=mode)


def slogdet(x, name=None):
    s, l = torch.linalg.slogdet(x)
    return torch.stack([s, l])


def solve(x, y, name=None):
    return torch.linalg.solve(x, y)


def svd(x, full_matrices=False, name=None):
    u, s, vh = torch.linalg.svd(x, full_matrices=full_matrices)
    return u, s, vh


def svdvals(x, name=None):
    return torch.linalg.svdvals(x)


def triangular_solve(x, y, upper=True, transpose=False, unitriangular=False, name=None):
    return torch.linalg.solve_triangular(
        x.transpose(-1, -2) if transpose else x, y, upper=upper,
        unitriangular=unitriangular)


def vector_norm(x, p=2, axis=None, keepdim=False, name=None):
    return torch.linalg.vector_norm(x, ord=p, dim=axis, keepdim=keepdim)


def householder_product(x, tau, name=None):
    return torch.linalg.householder_product(x, tau)


def lu_unpack(x, y, unpack_ludata=True, unpack_pivots=True, name=None):
    return torch.lu_unpack(x, y, unpack_data=unpack_ludata,
                           unpack_pivots=unpack_pivots)


def pca_lowrank(x, q=None, center=True, niter=2, name=None):
    return torch.pca_lowrank(x, q=q, center=center, niter=niter)


def svd_lowrank(x, q=6, niter=2, M=None, name=None):
    return torch.svd_lowrank(x, q=q, niter=niter, M=M)
