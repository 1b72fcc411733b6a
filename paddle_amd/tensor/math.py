"""Math ops (reference: python/paddle/tensor/math.py, logic.py, stat.py).

Thin dispatch to torch with paddle keyword conventions (axis→dim,
keepdim semantics identical).  Reductions over all axes return 0-d
tensors like paddle.
"""
from __future__ import annotations

import torch


def _axis(axis):
    if isinstance(axis, torch.Tensor):
        return int(axis.item()) if axis.numel() == 1 else [int(v) for v in axis]
    if isinstance(axis, (list, tuple)):
        return list(axis)
    return axis


# -- elementwise binary ------------------------------------------------------
def add(x, y, name=None):
    return torch.add(x, y)


def subtract(x, y, name=None):
    return torch.subtract(x, y)


def multiply(x, y, name=None):
    return torch.multiply(x, y)


def divide(x, y, name=None):
    return torch.divide(x, y)


def floor_divide(x, y, name=None):
    return torch.div(x, y, rounding_mode="trunc") if x.is_floating_point() else torch.floor_divide(x, y)


def mod(x, y, name=None):
    return torch.remainder(x, y)


remainder = mod


def pow(x, y, name=None):
    return torch.pow(x, y)


def maximum(x, y, name=None):
    return torch.maximum(x, y)


def minimum(x, y, name=None):
    return torch.minimum(x, y)


def scale(x, scale=1.0, bias=0.0, bias_after_scale=True, act=None, name=None):
    out = x * scale + bias if bias_after_scale else (x + bias) * scale
    return out


# -- elementwise unary -------------------------------------------------------
def abs(x, name=None):
    return torch.abs(x)


def ceil(x, name=None):
    return torch.ceil(x)


def floor(x, name=None):
    return torch.floor(x)


def round(x, name=None):
    return torch.round(x)


def trunc(x, name=None):
    return torch.trunc(x)


def exp(x, name=None):
    return torch.exp(x)


def log(x, name=None):
    return torch.log(x)


def log2(x, name=None):
    return torch.log2(x)


def log10(x, name=None):
    return torch.log10(x)


def sqrt(x, name=None):
    return torch.sqrt(x)


def rsqrt(x, name=None):
    return torch.rsqrt(x)


def square(x, name=None):
    return torch.square(x)


def reciprocal(x, name=None):
    return torch.reciprocal(x)


def sign(x, name=None):
    return torch.sign(x)


def sin(x, name=None):
    return torch.sin(x)


def cos(x, name=None):
    return torch.cos(x)


def tan(x, name=None):
    return torch.tan(x)


def sinh(x, name=None):
    return torch.sinh(x)


def cosh(x, name=None):
    return torch.cosh(x)


def tanh(x, name=None):
    return torch.tanh(x)


def erf(x, name=None):
    return torch.erf(x)


def clip(x, min=None, max=None, name=None):
    return torch.clamp(x, min=min, max=max)


# -- reductions --------------------------------------------------------------
def _reduce(fn, x, axis, keepdim, **kw):
    axis = _axis(axis)
    if axis is None:
        out = fn(x, **kw)
        if keepdim:
            out = out.reshape([1] * x.dim())
        return out
    return fn(x, dim=axis, keepdim=keepdim, **kw)


def sum(x, axis=None, dtype=None, keepdim=False, name=None):
    from .. import framework
    dt = framework.convert_dtype(dtype) if dtype else None
    axis = _axis(axis)
    if axis is None:
        out = torch.sum(x, dtype=dt)
        return out.reshape([1] * x.dim()) if keepdim else out
    return torch.sum(x, dim=axis, keepdim=keepdim, dtype=dt)


def mean(x, axis=None, keepdim=False, name=None):
    return _reduce(torch.mean, x, axis, keepdim)


def max(x, axis=None, keepdim=False, name=None):
    axis = _axis(axis)
    if axis is None:
        out = torch.max(x)
        return out.reshape([1] * x.dim()) if keepdim else out
    return torch.max(x, dim=axis, keepdim=keepdim).values


def min(x, axis=None, keepdim=False, name=None):
    axis = _axis(axis)
    if axis is None:
        out = torch.min(x)
        return out.reshape([1] * x.dim()) if keepdim else out
    return torch.min(x, dim=axis, keepdim=keepdim).values


def amax(x, axis=None, keepdim=False, name=None):
    return torch.amax(x, dim=_axis(axis) if axis is not None else tuple(range(x.dim())), keepdim=keepdim)


def amin(x, axis=None, keepdim=False, name=None):
    return torch.amin(x, dim=_axis(axis) if axis is not None else tuple(range(x.dim())), keepdim=keepdim)


def prod(x, axis=None, keepdim=False, dtype=None, name=None):
    from .. import framework
    dt = framework.convert_dtype(dtype) if dtype else None
    axis = _axis(axis)
    if axis is None:
        return torch.prod(x, dtype=dt)
    if isinstance(axis, list):
        for a in sorted(axis, reverse=True):
            x = torch.prod(x, dim=a, keepdim=keepdim, dtype=dt)
        return x
    return torch.prod(x, dim=axis, keepdim=keepdim, dtype=dt)


def logsumexp(x, axis=None, keepdim=False, name=None):
    axis = _axis(axis)
    if axis is None:
        axis = list(range(x.dim()))
    return torch.logsumexp(x, dim=axis, keepdim=keepdim)


def cumsum(x, axis=None, dtype=None, name=None):
    from .. import framework
    dt = framework.convert_dtype(dtype) if dtype else None
    if axis is None:
        return torch.cumsum(x.flatten(), dim=0, dtype=dt)
    return torch.cumsum(x, dim=axis, dtype=dt)


def cumprod(x, dim=None, dtype=None, name=None):
    from .. import framework
    dt = framework.convert_dtype(dtype) if dtype else None
    return torch.cumprod(x, dim=dim, dtype=dt)


def add_n(inputs, name=None):
    if isinstance(inputs, torch.Tensor):
        return inputs
    out = inputs[0].clone()
    for t in inputs[1:]:
        out = out + t
    return out


# -- comparisons / logic -----------------------------------------------------
def equal(x, y, name=None):
    return torch.eq(x, y)


def not_equal(x, y, name=None):
    return torch.ne(x, y)


def greater_than(x, y, name=None):
    return torch.gt(x, y)


def greater_equal(x, y, name=None):
    return torch.ge(x, y)


def less_than(x, y, name=None):
    return torch.lt(x, y)


def less_equal(x, y, name=None):
    return torch.le(x, y)


def equal_all(x, y, name=None):
    return torch.tensor(torch.equal(x, y))


def allclose(x, y, rtol=1e-05, atol=1e-08, equal_nan=False, name=None):
    return torch.tensor(torch.allclose(x, y, rtol=rtol, atol=atol, equal_nan=equal_nan))


def logical_and(x, y, out=None, name=None):
    return torch.logical_and(x, y)


def logical_or(x, y, out=None, name=None):
    return torch.logical_or(x, y)


def logical_xor(x, y, out=None, name=None):
    return torch.logical_xor(x, y)


def logical_not(x, out=None, name=None):
    return torch.logical_not(x)


def all(x, axis=None, keepdim=False, name=None):
    axis = _axis(axis)
    if axis is None:
        return torch.all(x)
    return torch.all(x, dim=axis, keepdim=keepdim)


def any(x, axis=None, keepdim=False, name=None):
    axis = _axis(axis)
    if axis is None:
        return torch.any(x)
    return torch.any(x, dim=axis, keepdim=keepdim)


def isnan(x, name=None):
    return torch.isnan(x)


def isinf(x, name=None):
    return torch.isinf(x)


def isfinite(x, name=None):
    return torch.isfinite(x)


# -- search / sort -----------------------------------------------------------
def argmax(x, axis=None, keepdim=False, dtype="int64", name=None):
    return torch.argmax(x, dim=_axis(axis), keepdim=keepdim if axis is not None else False)


def argmin(x, axis=None, keepdim=False, dtype="int64", name=None):
    return torch.argmin(x, dim=_axis(axis), keepdim=keepdim if axis is not None else False)


def argsort(x, axis=-1, descending=False, stable=False, name=None):
    return torch.argsort(x, dim=axis, descending=descending, stable=stable)


def sort(x, axis=-1, descending=False, stable=False, name=None):
    return torch.sort(x, dim=axis, descending=descending, stable=stable).values


def topk(x, k, axis=None, largest=True, sorted=True, name=None):
    if isinstance(k, torch.Tensor):
        k = int(k.item())
    vals, idx = torch.topk(x, k, dim=_axis(axis) if axis is not None else -1,
                           largest=largest, sorted=sorted)
    return vals, idx


def bincount(x, weights=None, minlength=0, name=None):
    return torch.bincount(x, weights=weights, minlength=minlength)


# -- statistics long tail ----------------------------------------------------
def std(x, axis=None, unbiased=True, keepdim=False, name=None):
    axis = _axis(axis)
    if axis is None:
        return torch.std(x, correction=1 if unbiased else 0)
    return torch.std(x, dim=axis, correction=1 if unbiased else 0, keepdim=keepdim)


def var(x, axis=None, unbiased=True, keepdim=False, name=None):
    axis = _axis(axis)
    if axis is None:
        return torch.var(x, correction=1 if unbiased else 0)
    return torch.var(x, dim=axis, correction=1 if unbiased else 0, keepdim=keepdim)


def median(x, axis=None, keepdim=False, mode="avg", name=None):
    axis = _axis(axis)
    if axis is None:
        flat = x.flatten().sort().values
        n = flat.numel()
        if mode == "avg" and n % 2 == 0:
            return (flat[n // 2 - 1] + flat[n // 2]) / 2
        return flat[(n - 1) // 2] if mode == "min" else flat[n // 2]
    vals, idx = torch.median(x, dim=axis, keepdim=keepdim)
    return vals


def nanmedian(x, axis=None, keepdim=False, name=None):
    axis = _axis(axis)
    if axis is None:
        return torch.nanmedian(x)
    return torch.nanmedian(x, dim=axis, keepdim=keepdim).values


def nanmean(x, axis=None, keepdim=False, name=None):
    axis = _axis(axis)
    if axis is None:
        return torch.nanmean(x)
    return torch.nanmean(x, dim=axis, keepdim=keepdim)


def nansum(x, axis=None, dtype=None, keepdim=False, name=None):
    from .. import framework
    dt = framework.convert_dtype(dtype) if dtype else None
    axis = _axis(axis)
    if axis is None:
        return torch.nansum(x, dtype=dt)
    return torch.nansum(x, dim=axis, keepdim=keepdim, dtype=dt)


def kthvalue(x, k, axis=None, keepdim=False, name=None):
    ax = _axis(axis)
    if ax is None:
        ax = x.dim() - 1
    v, i = torch.kthvalue(x, k, dim=ax, keepdim=keepdim)
    return v, i


def mode(x, axis=-1, keepdim=False, name=None):
    v, i = torch.mode(x, dim=_axis(axis), keepdim=keepdim)
    return v, i


def quantile(x, q, axis=None, keepdim=False, interpolation="linear", name=None):
    qq = torch.as_tensor(q, dtype=torch.float64, device=x.device)
    return torch.quantile(x.double(), qq, dim=_axis(axis), keepdim=keepdim,
                          interpolation=interpolation).to(x.dtype)


def diff(x, n=1, axis=-1, prepend=None, append=None, name=None):
    return torch.diff(x, n=n, dim=axis, prepend=prepend, append=append)


def trapezoid(y, x=None, dx=None, axis=-1, name=None):
    if dx is not None:
        return torch.trapezoid(y, dx=dx, dim=axis)
    return torch.trapezoid(y, x=x, dim=axis)


def take(x, index, mode="raise", name=None):
    return torch.take(x, index.long())


def lerp(x, y, weight, name=None):
    return torch.lerp(x, y, weight)


def addmm(input, x, y, beta=1.0, alpha=1.0, name=None):
    return torch.addmm(input, x, y, beta=beta, alpha=alpha)


def inner(x, y, name=None):
    return torch.inner(x, y)


def kron(x, y, name=None):
    return torch.kron(x, y)


def gcd(x, y, name=None):
    return torch.gcd(x, y)


def lcm(x, y, name=None):
    return torch.lcm(x, y)


def heaviside(x, y, name=None):
    return torch.heaviside(x, y)


def frac(x, name=None):
    return torch.frac(x)


def deg2rad(x, name=None):
    return torch.deg2rad(x)


def rad2deg(x, name=None):
    return torch.rad2deg(x)


def angle(x, name=None):
    return torch.angle(x)


def conj(x, name=None):
    return torch.conj(x)


def real(x, name=None):
    return torch.real(x)


def imag(x, name=None):
    return torch.imag(x)


def log1p(x, name=None):
    return torch.log1p(x)


def expm1(x, name=None):
    return torch.expm1(x)


def atan(x, name=None):
    return torch.atan(x)


def atan2(x, y, name=None):
    return torch.atan2(x, y)


def asin(x, name=None):
    return torch.asin(x)


def acos(x, name=None):
    return torch.acos(x)


def asinh(x, name=None):
    return torch.asinh(x)


def acosh(x, name=None):
    return torch.acosh(x)


def atanh(x, name=None):
    return torch.atanh(x)


def erfinv(x, name=None):
    return torch.erfinv(x)


def digamma(x, name=None):
    return torch.digamma(x)


def lgamma(x, name=None):
    return torch.lgamma(x)


def logit(x, eps=None, name=None):
    return torch.logit(x, eps=eps)


def nextafter(x, y, name=None):
    return torch.nextafter(x, y)


def count_nonzero(x, axis=None, keepdim=False, name=None):
    return torch.count_nonzero(x, dim=_axis(axis))


def histogram(input, bins=100, min=0, max=0, name=None):
    return torch.histc(input.float(), bins=bins, min=min, max=max)
