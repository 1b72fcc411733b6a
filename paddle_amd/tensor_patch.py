"""Paddle-style Tensor surface on torch.Tensor.

The reference implements a CPython Tensor type with paddle semantics
(paddle/fluid/pybind/eager.cc, eager_method.cc).  Our substrate is
torch.Tensor; this module grafts the paddle-visible API differences onto
the torch.Tensor class once at import:

  - ``stop_gradient`` property      (inverse of requires_grad)
  - ``astype`` / ``cast``           (dtype casts by paddle name)
  - ``numpy()``                     (works on GPU + bf16)
  - ``scale``, ``clip``, ``tile``, ``unsqueeze_``-style aliases
  - ``set_value``, ``clear_gradient``, ``_md5sum`` etc. used by fleet code

Monkey-patching the class is intentional: it gives every tensor the
paddle surface with zero per-op wrapper overhead (the reference keeps
the hot path in C for the same reason).
"""
from __future__ import annotations

import torch

from . import framework

_PATCHED = False


def _stop_gradient_get(self: torch.Tensor) -> bool:
    return not self.requires_grad


def _stop_gradient_set(self: torch.Tensor, value: bool):
    if self.is_leaf or not value:
        self.requires_grad_(not value)
    # non-leaf with stop_gradient=True: detach semantics handled by callers


def _astype(self: torch.Tensor, dtype):
    return self.to(framework.convert_dtype(dtype))


def _paddle_numpy(self: torch.Tensor, force=True):
    t = self.detach()
    if t.device.type != "cpu":
        t = t.cpu()
    if t.dtype == torch.bfloat16:
        import numpy as np
        # numpy has no bf16: paddle returns uint16 view; we return float32
        # for usability in tests (value-preserving).
        return t.float().numpy()
    return t.numpy()


def _scale(self, scale=1.0, bias=0.0, bias_after_scale=True, act=None, name=None):
    if bias_after_scale:
        out = self * scale + bias
    else:
        out = (self + bias) * scale
    return out


def _clear_gradient(self, set_to_zero=True):
    if self.grad is not None:
        if set_to_zero:
            self.grad.zero_()
        else:
            self.grad = None


def _set_value(self, value):
    with torch.no_grad():
        if not isinstance(value, torch.Tensor):
            value = torch.as_tensor(value, dtype=self.dtype, device=self.device)
        self.copy_(value.to(device=self.device, dtype=self.dtype))
    return self


def _item_compat(self, *args):
    if args:
        return self.flatten()[list(args) if len(args) > 1 else args[0]].item()
    return torch.Tensor.item(self)


def _get_tensor(self):
    return self


def _place(self):
    if self.device.type == "cuda":
        return framework.GPUPlace(self.device.index or 0)
    return framework.CPUPlace()


def apply_patches():
    global _PATCHED
    if _PATCHED:
        return
    _PATCHED = True
    T = torch.Tensor
    T.stop_gradient = property(_stop_gradient_get, _stop_gradient_set)
    T.astype = _astype
    T.cast = _astype
    T.scale = _scale
    T.clear_gradient = _clear_gradient
    T.set_value = _set_value
    T.get_tensor = _get_tensor
    # paddle's .numpy() must work for GPU/bf16 tensors
    _orig_numpy = T.numpy

    def numpy_(self, force=False):
        if self.device.type != "cpu" or self.dtype == torch.bfloat16 or self.requires_grad:
            return _paddle_numpy(self)
        return _orig_numpy(self)

    T.numpy = numpy_
    if not hasattr(T, "place"):
        T.place = property(_place)
    # name attribute used by fleet bookkeeping
    if not hasattr(T, "name"):
        _names = {}

        def _name_get(self):
            return _names.get(id(self), f"tensor_{id(self)}")

        def _name_set(self, v):
            _names[id(self)] = v

        T.name = property(_name_get, _name_set)


def _patch_method_long_tail():
    """Bind the reference's tensor_method_func long tail as torch.Tensor
    methods (reference: python/paddle/tensor/__init__.py) -- each method
    is the same paddle function with the tensor as first argument."""
    import torch
    from . import __init__ as _root  # noqa: F401  (late-bound lookups below)
    import paddle_amd as P

    names = [
        "add_n", "as_complex", "as_real", "atleast_1d", "atleast_2d",
        "atleast_3d", "block_diag", "broadcast_shape", "broadcast_tensors",
        "bucketize", "cast_", "cdist", "concat", "cond", "create_parameter",
        "create_tensor", "cumulative_trapezoid", "eigvals", "eigvalsh",
        "equal_", "equal_all", "flatten_", "floor_mod", "floor_mod_",
        "gammainc", "gammainc_", "gammaincc", "gammaincc_", "gammaln",
        "gammaln_", "gather_nd", "greater_than", "greater_than_",
        "histogram_bin_edges", "histogramdd", "householder_product", "i0e",
        "i1", "i1e", "increment", "index_sample", "is_empty", "is_integer",
        "is_tensor", "isin", "less_than", "less_than_", "lu_unpack", "mod",
        "mod_", "multi_dot", "multigammaln", "multigammaln_", "multiplex",
        "pca_lowrank", "pinv", "polar", "put_along_axis", "put_along_axis_",
        "rank", "reduce_as", "reshape_", "reverse", "scale_", "scatter_nd",
        "scatter_nd_add", "shard_index", "slice", "stack", "stanh",
        "strided_slice", "svd_lowrank", "take_along_axis", "tensordot",
        "top_p_sampling", "trapezoid", "unstack", "vander", "where_",
    ]
    T = torch.Tensor
    for n in names:
        fn = getattr(P, n, None)
        if fn is None or hasattr(T, n):
            continue
        setattr(T, n, fn)

# called from paddle_amd/__init__.py AFTER all top-level names exist
