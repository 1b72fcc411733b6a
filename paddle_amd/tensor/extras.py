"""Long-tail paddle.* API parity (reference: python/paddle/__init__.py
__all__ -- the thin-dispatch remainder: bitwise ops, stacks, special
functions, in-place variants, indexing scatter helpers).

Everything here is a direct mapping onto torch with paddle keyword
conventions; anything with real paddle-specific semantics (trace axes,
multiplex, shard_index, stanh...) is written out explicitly.
"""
from __future__ import annotations

import torch

# -- direct same-name torch dispatches ---------------------------------------
_DIRECT = [
    "block_diag", "diagflat", "diag_embed", "isclose", "cartesian_prod",
    "logaddexp", "logcumsumexp", "cummax", "cummin", "bucketize",
    "tensor_split", "hsplit", "dsplit", "vsplit", "isin", "isneginf",
    "isposinf", "isreal", "bitwise_and", "bitwise_or", "bitwise_xor",
    "bitwise_not", "rot90", "histogramdd", "complex", "cdist", "pdist",
    "nanquantile", "hstack", "vstack", "dstack", "column_stack",
    "logspace", "atleast_1d", "atleast_2d", "atleast_3d", "diagonal",
    "broadcast_tensors", "fmax", "fmin", "moveaxis", "renorm",
    "select_scatter", "nan_to_num", "index_add", "sgn", "frexp", "ldexp",
    "polar", "vander", "unflatten", "as_strided", "i0", "polygamma",
    "copysign", "hypot", "signbit", "unique_consecutive", "combinations",
    "diagonal_scatter", "slice_scatter", "neg", "randint_like",
    "masked_scatter", "tril_indices", "triu_indices",
]

_g = globals()
for _n in _DIRECT:
    _g[_n] = getattr(torch, _n)

row_stack = torch.vstack
as_complex = torch.view_as_complex
as_real = torch.view_as_real
i0e = torch.special.i0e
i1 = torch.special.i1
i1e = torch.special.i1e
sinc = torch.sinc
gammaln = torch.special.gammaln
gammainc = torch.special.gammainc
gammaincc = torch.special.gammaincc
multigammaln = torch.special.multigammaln
histogram_bin_edges = lambda input, bins=100, min=0.0, max=0.0, weight=None: \
    torch.histogram(input.float(), bins=bins,
                    range=None if (min == 0 and max == 0) else (float(min), float(max))).bin_edges
set_printoptions = torch.set_printoptions


# -- paddle-specific semantics ----------------------------------------------
def rank(x):
    return torch.tensor(x.dim())


def trace(x, offset=0, axis1=0, axis2=1, name=None):
    return torch.diagonal(x, offset=offset, dim1=axis1, dim2=axis2).sum(-1)


def increment(x, value=1.0, name=None):
    with torch.no_grad():
        x.add_(value)
    return x


def multiplex(inputs, index, name=None):
    stacked = torch.stack(list(inputs))          # [n, batch, ...]
    idx = index.reshape(-1).long()
    rows = torch.arange(idx.numel(), device=stacked.device)
    return stacked[idx, rows]


def shard_index(input, index_num, nshards, shard_id, ignore_value=-1):
    size = (index_num + nshards - 1) // nshards
    lo, hi = shard_id * size, (shard_id + 1) * size
    inside = (input >= lo) & (input < hi)
    return torch.where(inside, input - lo, torch.full_like(input, ignore_value))


def crop(x, shape=None, offsets=None, name=None):
    offsets = offsets or [0] * x.dim()
    shape = shape or list(x.shape)
    idx = tuple(slice(int(o), int(o) + int(s)) for o, s in zip(offsets, shape))
    return x[idx]


def stanh(x, scale_a=0.67, scale_b=1.7159, name=None):
    return scale_b * torch.tanh(scale_a * x)


def reverse(x, axis, name=None):
    axis = [axis] if isinstance(axis, int) else list(axis)
    return torch.flip(x, axis)


def strided_slice(x, axes, starts, ends, strides, name=None):
    idx = [slice(None)] * x.dim()
    for ax, st, en, sr in zip(axes, starts, ends, strides):
        idx[ax] = slice(st, en, sr)
    return x[tuple(idx)]


def standard_normal(shape, dtype=None, name=None):
    from .. import framework
    dt = framework.convert_dtype(dtype) if dtype else torch.float32
    return torch.randn(shape, dtype=dt)


def log_normal(mean=1.0, std=2.0, shape=None, dtype=None, name=None):
    out = standard_normal(shape or [1], dtype)
    return torch.exp(out * std + mean)


def binomial(count, prob, name=None):
    return torch.binomial(count.float(), prob.float())


def standard_gamma(alpha, name=None):
    return torch._standard_gamma(alpha)


def broadcast_shape(x_shape, y_shape):
    return list(torch.broadcast_shapes(tuple(x_shape), tuple(y_shape)))


def reduce_as(x, target, name=None):
    while x.dim() > target.dim():
        x = x.sum(0)
    for i, (a, b) in enumerate(zip(x.shape, target.shape)):
        if a != b:
            x = x.sum(i, keepdim=True)
    return x


def is_tensor(x):
    return isinstance(x, torch.Tensor)


def is_complex(x):
    return torch.is_complex(x)


def is_integer(x):
    return not x.is_floating_point() and not torch.is_complex(x)


def is_floating_point(x):
    return x.is_floating_point()


def is_empty(x, name=None):
    return torch.tensor(x.numel() == 0)


def tolist(x):
    return x.tolist()


def equal_all(x, y, name=None):
    return torch.tensor(torch.equal(x, y))


def cumulative_trapezoid(y, x=None, dx=None, axis=-1, name=None):
    if dx is not None:
        return torch.cumulative_trapezoid(y, dx=dx, dim=axis)
    return torch.cumulative_trapezoid(y, x=x, dim=axis)


def index_fill(x, index, axis, value, name=None):
    return x.index_fill(axis, index.long(), value)


def view_as(x, other, name=None):
    return x.view_as(other)


def unfold(x, axis, size, step, name=None):
    return x.unfold(axis, size, step)


def bitwise_left_shift(x, y, is_arithmetic=True, out=None, name=None):
    return torch.bitwise_left_shift(x, y)


def bitwise_right_shift(x, y, is_arithmetic=True, out=None, name=None):
    return torch.bitwise_right_shift(x, y)


def scatter_nd_add(x, index, updates, name=None):
    out = x.clone()
    k = index.shape[-1]
    flat_idx = index.reshape(-1, k).long()
    flat_upd = updates.reshape(flat_idx.shape[0],
                               *updates.shape[index.dim() - 1:])
    out.index_put_(tuple(flat_idx[:, d] for d in range(k)), flat_upd,
                   accumulate=True)
    return out


def scatter_nd(index, updates, shape, name=None):
    zeros = torch.zeros(shape, dtype=updates.dtype, device=updates.device)
    return scatter_nd_add(zeros, index, updates)


def check_shape(x, shape):
    return list(x.shape) == list(shape)


def disable_signal_handler():
    pass


class LazyGuard:
    def __enter__(self):
        return self

    def __exit__(self, *a):
        return False


# -- RNG state ---------------------------------------------------------------
def get_rng_state(device=None):
    return [torch.get_rng_state()]


def set_rng_state(state_list, device=None):
    torch.set_rng_state(state_list[0] if isinstance(state_list, list) else state_list)


def get_cuda_rng_state():
    return torch.cuda.get_rng_state_all() if torch.cuda.is_available() else []


def set_cuda_rng_state(states):
    if torch.cuda.is_available():
        torch.cuda.set_rng_state_all(states)


# -- dtype info ---------------------------------------------------------------
iinfo = torch.iinfo
finfo = torch.finfo


# -- in-place variants (paddle's foo_ family) --------------------------------
_INPLACE_METHODS = [
    "abs", "cos", "sin", "tan", "sinh", "tanh", "acos", "asin", "atan",
    "ceil", "floor", "round", "trunc", "frac", "exp", "expm1", "log",
    "log2", "log10", "log1p", "sqrt", "rsqrt", "reciprocal", "sigmoid",
    "erf", "erfinv", "neg", "square", "digamma", "lgamma", "logit", "i0",
    "sinc", "cumsum", "cumprod", "clip", "tril", "triu", "scatter",
    "bitwise_and", "bitwise_or", "bitwise_xor", "bitwise_not",
    "logical_and", "logical_or", "logical_not", "multiply", "divide",
    "pow", "remainder", "mod", "floor_divide", "gcd", "lcm", "hypot",
    "copysign", "ldexp", "nan_to_num", "polygamma", "renorm",
    "masked_fill", "masked_scatter", "equal", "less_than", "less_equal",
    "greater_than", "greater_equal", "bernoulli", "normal", "cauchy",
    "geometric", "log_normal", "where", "flatten", "transpose", "t",
    "gammainc", "gammaincc", "gammaln", "multigammaln",
    "bitwise_left_shift", "bitwise_right_shift", "fill_diagonal",
]

_TORCH_INPLACE = {
    "mod": "remainder_", "floor_divide": "floor_divide_",
    "less_than": "less_", "less_equal": "le_", "greater_than": "gt_",
    "greater_equal": "ge_", "equal": "eq_", "divide": "div_",
    "multiply": "mul_", "clip": "clamp_", "cumprod": None, "cumsum": None,
    "flatten": None, "transpose": None, "where": None, "t": "t_",
    "gammainc": None, "gammaincc": None, "gammaln": None,
    "multigammaln": "mvlgamma_", "logit": "logit_",
}


def _make_inplace(name):
    meth = _TORCH_INPLACE.get(name, name + "_")

    def fn(x, *args, **kwargs):
        kwargs.pop("name", None)
        if meth is not None and hasattr(x, meth):
            getattr(x, meth)(*args, **kwargs)
            return x
        # no real torch in-place -- compute and copy back
        from . import math as _m
        out = (getattr(_m, name, None) or getattr(torch, name))(x, *args, **kwargs)
        x.copy_(out)
        return x
    fn.__name__ = name + "_"
    return fn


for _n in _INPLACE_METHODS:
    _g[_n + "_"] = _make_inplace(_n)

# names whose plain form also needed defining here
logit_ = _make_inplace("logit")
cast_ = _make_inplace("astype")


def astype(x, dtype):  # helper for cast_
    from .. import framework
    return x.to(framework.convert_dtype(dtype))
