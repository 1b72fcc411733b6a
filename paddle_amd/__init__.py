"""paddle_amd -- an MI355X-native deep-learning framework with the
PaddlePaddle API surface.

Architecture: the tensor/autograd substrate is
PyTorch-ROCm; hot ops are hand-written gfx950 HIP kernels (csrc/)
exposed as torch custom ops; distributed training is RCCL over xGMI via
torch.distributed; the `paddle.*`-shaped Python API lives here.

This is a ground-up rebuild, not a port: no CUDA-compat layer, no
multi-backend dispatch, no codegen compiler.  Reference parity anchors
are cited per-module against /root/reference (PaddlePaddle/Paddle).
"""
from __future__ import annotations

import torch

from . import framework
from .framework import (  # noqa: F401
    CPUPlace,
    CUDAPlace,
    GPUPlace,
    bfloat16,
    bool_,
    complex128,
    complex64,
    float16,
    float32,
    float64,
    float8_e4m3fn,
    float8_e5m2,
    get_device,
    get_flags,
    int16,
    int32,
    int64,
    int8,
    is_compiled_with_cuda,
    is_compiled_with_rocm,
    seed,
    set_device,
    set_flags,
    uint8,
)
from . import tensor_patch as _tensor_patch

_tensor_patch.apply_patches()

Tensor = torch.Tensor

# ---------------------------------------------------------------------------
# creation ops (reference: python/paddle/tensor/creation.py)
# ---------------------------------------------------------------------------
from .tensor.creation import (  # noqa: F401
    arange,
    assign,
    clone,
    empty,
    empty_like,
    eye,
    full,
    full_like,
    linspace,
    meshgrid,
    ones,
    ones_like,
    rand,
    randint,
    randn,
    randperm,
    normal,
    uniform,
    to_tensor,
    tril,
    triu,
    zeros,
    zeros_like,
    diag,
)
from .tensor.manipulation import (  # noqa: F401
    broadcast_to,
    cast,
    chunk,
    concat,
    expand,
    expand_as,
    flatten,
    flip,
    gather,
    gather_nd,
    index_select,
    masked_select,
    numel,
    put_along_axis,
    repeat_interleave,
    reshape,
    reshape_,
    roll,
    scatter,
    shape,
    slice,
    split,
    squeeze,
    squeeze_,
    stack,
    take_along_axis,
    tile,
    transpose,
    unbind,
    unique,
    unsqueeze,
    unsqueeze_,
    unstack,
    view,
    where,
)
from .tensor.math import (  # noqa: F401
    abs,
    acos,
    acosh,
    addmm,
    angle,
    asin,
    asinh,
    atan,
    atan2,
    atanh,
    conj,
    count_nonzero,
    deg2rad,
    diff,
    digamma,
    erfinv,
    expm1,
    frac,
    gcd,
    heaviside,
    histogram,
    imag,
    inner,
    kron,
    kthvalue,
    lcm,
    lerp,
    lgamma,
    log1p,
    logit,
    median,
    mode,
    nanmean,
    nanmedian,
    nansum,
    nextafter,
    quantile,
    rad2deg,
    real,
    std,
    take,
    trapezoid,
    var,
    add,
    add_n,
    all,
    allclose,
    amax,
    amin,
    any,
    argmax,
    argmin,
    argsort,
    bincount,
    ceil,
    clip,
    cos,
    cosh,
    cumprod,
    cumsum,
    divide,
    equal,
    equal_all,
    erf,
    exp,
    floor,
    floor_divide,
    greater_equal,
    greater_than,
    isfinite,
    isinf,
    isnan,
    less_equal,
    less_than,
    log,
    log2,
    log10,
    logical_and,
    logical_not,
    logical_or,
    logical_xor,
    logsumexp,
    max,
    maximum,
    mean,
    min,
    minimum,
    mod,
    multiply,
    not_equal,
    pow,
    prod,
    reciprocal,
    remainder,
    round,
    rsqrt,
    scale,
    sign,
    sin,
    sinh,
    sort,
    sqrt,
    square,
    subtract,
    sum,
    tan,
    tanh,
    topk,
    trunc,
)
from .tensor.linalg import (  # noqa: F401
    bmm,
    cross,
    dist,
    dot,
    einsum,
    matmul,
    mm,
    mv,
    norm,
    outer,
    t,
    tensordot,
)
from .tensor.search import (  # noqa: F401
    index_sample,
    masked_fill,
    nonzero,
    searchsorted,
)
from .tensor.random import multinomial, bernoulli, poisson  # noqa: F401
from .tensor.einsum import einsum  # noqa: F401,F811

from . import nn  # noqa: F401
from . import optimizer  # noqa: F401
from . import amp  # noqa: F401
from . import io  # noqa: F401
from . import distributed  # noqa: F401
from . import autograd  # noqa: F401
from . import device  # noqa: F401
from . import metric  # noqa: F401
from . import vision  # noqa: F401
from . import static  # noqa: F401
from . import jit  # noqa: F401
from . import incubate  # noqa: F401
from . import models  # noqa: F401
from . import distribution  # noqa: F401
from . import profiler  # noqa: F401
from . import linalg  # noqa: F401
from . import fft  # noqa: F401
from . import base  # noqa: F401
from . import callbacks  # noqa: F401
from . import cost_model  # noqa: F401
from . import dataset  # noqa: F401
from . import hub  # noqa: F401
from . import utils  # noqa: F401
from . import signal  # noqa: F401
from . import sparse  # noqa: F401
from . import inference  # noqa: F401
from . import quantization  # noqa: F401
from . import text  # noqa: F401
from . import audio  # noqa: F401
from . import onnx  # noqa: F401
from .framework_io import (  # noqa: F401
    async_save,
    load,
    load_safetensors,
    save,
    save_safetensors,
)

# remaining top-level parity names
import torch as _torch  # noqa: E402
dtype = _torch.dtype
bool = _torch.bool  # noqa: A001 -- paddle exports `bool` as a dtype


def addmm_(input, x, y, beta=1.0, alpha=1.0, name=None):
    input.addmm_(x, y, beta=beta, alpha=alpha)
    return input


def floor_mod(x, y, name=None):
    return _torch.remainder(x, y)


def floor_mod_(x, y, name=None):
    x.remainder_(y)
    return x


def batch(reader, batch_size, drop_last=False):
    def _gen():
        buf = []
        for item in reader():
            buf.append(item)
            if len(buf) == batch_size:
                yield buf
                buf = []
        if buf and not drop_last:
            yield buf
    return _gen


def summary(net, input_size=None, dtypes=None, input=None):
    from .hapi_summary import summary as _s
    return _s(net, input_size, dtypes, input)


def flops(net, input_size, custom_ops=None, print_detail=False):
    from .hapi_summary import flops as _f
    return _f(net, input_size, custom_ops, print_detail)


def create_parameter(shape, dtype="float32", name=None, attr=None,
                     is_bias=False, default_initializer=None):
    from . import framework as _fw
    t = _torch.empty(shape, dtype=_fw.convert_dtype(dtype))
    if default_initializer is not None:
        default_initializer(t)
    elif is_bias or t.dim() < 2:
        _torch.nn.init.zeros_(t)
    else:
        _torch.nn.init.xavier_normal_(t)
    t.requires_grad_(True)
    return t


class CUDAPinnedPlace:
    def __repr__(self):
        return "Place(gpu_pinned)"


DataParallel = distributed.DataParallel



# tensor-method long tail at top level
def put_along_axis_(arr, indices, values, axis, reduce="assign"):
    import torch as _tt
    if reduce == "assign":
        arr.scatter_(axis, indices.long(),
                     values if isinstance(values, _tt.Tensor) else
                     _tt.full_like(indices, values, dtype=arr.dtype))
    else:
        arr.scatter_reduce_(axis, indices.long(), values,
                            {"add": "sum", "mul": "prod"}.get(reduce, reduce))
    return arr


def top_p_sampling(x, ps, threshold=None, seed=None, name=None):
    """Nucleus sampling over the last dim (reference:
    paddle/phi/kernels/gpu/top_p_sampling_kernel.cu)."""
    import torch as _tt
    probs = _tt.softmax(x.float(), dim=-1)
    sp, si = probs.sort(dim=-1, descending=True)
    cum = sp.cumsum(-1)
    keep = cum - sp < ps.unsqueeze(-1)
    sp = sp * keep
    sp = sp / sp.sum(-1, keepdim=True)
    choice = _tt.multinomial(sp.reshape(-1, sp.shape[-1]), 1)
    ids = si.reshape(-1, si.shape[-1]).gather(1, choice).reshape(*x.shape[:-1], 1)
    scores = probs.reshape(-1, probs.shape[-1]).gather(1, choice).reshape(*x.shape[:-1], 1)
    return scores, ids


def scale_(x, scale=1.0, bias=0.0, bias_after_scale=True, act=None, name=None):
    import torch as _tt
    with _tt.no_grad():
        if bias_after_scale:
            x.mul_(scale).add_(bias)
        else:
            x.add_(bias).mul_(scale)
    return x


def create_tensor(dtype, name=None, persistable=False):
    from . import framework as _fw
    return _torch.empty(0, dtype=_fw.convert_dtype(dtype))


from .linalg import (  # noqa: E402,F401
    cond, eigvals, eigvalsh, householder_product, lu_unpack, multi_dot,
    pca_lowrank, pinv, svd_lowrank,
)

# long-tail export parity: everything public in tensor/extras.py
from .tensor import extras as _extras  # noqa: E402
for _n in dir(_extras):
    if not _n.startswith("_") and _n not in globals():
        globals()[_n] = getattr(_extras, _n)
del _extras, _n

from .hapi import Model  # noqa: F401
from .param_attr import ParamAttr  # noqa: F401
from .autograd import grad, no_grad, enable_grad, set_grad_enabled, is_grad_enabled  # noqa: F401

disable_static = static.disable_static
enable_static = static.enable_static
in_dynamic_mode = lambda: not static._static_mode

__version__ = "0.1.0"

from . import version_mod as version  # noqa: E402  (paddle.version module)


def is_grad_enabled_():
    return torch.is_grad_enabled()


def get_default_dtype():
    return framework.dtype_name(torch.get_default_dtype())


def set_default_dtype(d):
    torch.set_default_dtype(framework.convert_dtype(d))


def grad_(outputs, inputs, grad_outputs=None, retain_graph=None, create_graph=False):
    return torch.autograd.grad(outputs, inputs, grad_outputs, retain_graph=retain_graph, create_graph=create_graph)


def synchronize(device=None):
    if torch.cuda.is_available():
        torch.cuda.synchronize(device)

# bind the tensor-method long tail now that every top-level name exists
from . import tensor_patch as _tp  # noqa: E402
_tp._patch_method_long_tail()
del _tp
