"""paddle.nn.Layer on the torch.nn.Module substrate.

Reference parity: python/paddle/nn/layer/layers.py:354 (Layer).  The
hook semantics (forward pre/post hooks, buffers, state_dict naming,
parameter iteration order) are load-bearing: sharding/TP/recompute rely
on them.  We subclass torch.nn.Module so autograd,
parameter registration and hooks come from torch, then graft the
paddle-visible API on top.
"""
from __future__ import annotations

from typing import Iterator, Optional, Tuple

import torch

from .. import framework
from .initializer import _apply_initializer, Constant, XavierNormal


class Layer(torch.nn.Module):
    def __init__(self, name_scope=None, dtype="float32"):
        super().__init__()
        self._dtype = framework.convert_dtype(dtype) if dtype else torch.float32

    # -- paddle-style construction helpers ----------------------------------
    def create_parameter(self, shape, attr=None, dtype=None, is_bias=False,
                         default_initializer=None):
        dtype = framework.convert_dtype(dtype) if dtype is not None else self._dtype
        t = torch.empty(list(shape), dtype=dtype, device=framework.get_default_device())
        init = None
        if attr is not None and getattr(attr, "initializer", None) is not None:
            init = attr.initializer
        if init is None:
            # set_global_initializer overrides layer defaults but not an
            # explicit ParamAttr initializer (reference initializer.py
            # set_global_initializer semantics)
            from . import initializer as _I
            init = (_I._GLOBAL_BIAS_INIT if is_bias
                    else _I._GLOBAL_WEIGHT_INIT)
        if init is None:
            init = default_initializer
        if init is None:
            init = Constant(0.0) if is_bias else XavierNormal()
        _apply_initializer(init, t)
        p = torch.nn.Parameter(t)
        if attr is not None and getattr(attr, "learning_rate", None) is not None:
            p.optimize_attr = {"learning_rate": attr.learning_rate}
        if attr is not None and getattr(attr, "trainable", True) is False:
            p.requires_grad_(False)
        return p

    def add_parameter(self, name, parameter):
        self.register_parameter(name, parameter)
        return parameter

    def add_sublayer(self, name, sublayer):
        self.add_module(name, sublayer)
        return sublayer

    def create_tensor(self, name=None, persistable=False, dtype=None):
        return torch.empty(0, dtype=framework.convert_dtype(dtype) if dtype else self._dtype)

    # -- traversal (paddle names) -------------------------------------------
    def sublayers(self, include_self=False):
        out = []
        for m in self.modules():
            if m is self and not include_self:
                continue
            out.append(m)
        return out

    def named_sublayers(self, prefix="", include_self=False, layers_set=None):
        for name, m in self.named_modules(prefix=prefix):
            if m is self and not include_self:
                continue
            yield name, m

    def children(self):
        return super().children()

    def parameters(self, include_sublayers=True, recurse=None):
        r = include_sublayers if recurse is None else recurse
        return list(torch.nn.Module.parameters(self, recurse=r))

    def named_parameters(self, prefix="", include_sublayers=True, recurse=None,
                         remove_duplicate=True):
        r = include_sublayers if recurse is None else recurse
        return torch.nn.Module.named_parameters(self, prefix=prefix, recurse=r,
                                                remove_duplicate=remove_duplicate)

    def buffers(self, include_sublayers=True):
        return list(super().buffers(recurse=include_sublayers))

    # paddle register_buffer has `persistable` (inverse of torch persistent)
    def register_buffer(self, name, tensor, persistable=True):
        super().register_buffer(name, tensor, persistent=persistable)

    # -- state dict ----------------------------------------------------------
    def state_dict(self, *args, destination=None, prefix="", include_sublayers=True,
                   structured_name_prefix="", use_hook=True, keep_vars=False, **kw):
        # torch recurses via child.state_dict(destination=..., prefix=...);
        # paddle's kwarg is structured_name_prefix -- honor both.
        return super().state_dict(*args, destination=destination,
                                  prefix=prefix or structured_name_prefix,
                                  keep_vars=keep_vars)

    def set_state_dict(self, state_dict, use_structured_name=True):
        # tolerate numpy arrays (paddle checkpoints store numpy)
        import numpy as np
        cleaned = {}
        for k, v in state_dict.items():
            if isinstance(v, np.ndarray):
                v = torch.from_numpy(v)
            cleaned[k] = v
        missing, unexpected = self.load_state_dict(cleaned, strict=False)
        return missing, unexpected

    set_dict = set_state_dict
    load_dict = set_state_dict

    # -- hooks (paddle names) ------------------------------------------------
    def register_forward_pre_hook(self, hook):
        # paddle hook signature: hook(layer, input) -> maybe new input
        return super().register_forward_pre_hook(hook)

    def register_forward_post_hook(self, hook):
        return super().register_forward_hook(hook)

    # -- mode / movement -----------------------------------------------------
    def train(self, mode: bool = True):
        return super().train(mode)

    def eval(self):
        return super().eval()

    @property
    def training_(self):
        return self.training

    def to(self, device=None, dtype=None, blocking=None):
        dev = framework._place_from_any(device) if device is not None else None
        dt = framework.convert_dtype(dtype) if dtype is not None else None
        return super().to(device=dev, dtype=dt)

    def full_name(self):
        return self.__class__.__name__.lower()

    def clear_gradients(self, set_to_zero=True):
        for p in self.parameters():
            if p.grad is not None:
                if set_to_zero:
                    p.grad.zero_()
                else:
                    p.grad = None

    def astype(self, dtype):
        return self.to(dtype=dtype)

    # paddle allows calling layers on positional/keyword
    def forward(self, *inputs, **kwargs):  # pragma: no cover - abstract
        raise NotImplementedError


class Sequential(Layer):
    def __init__(self, *layers):
        super().__init__()
        if len(layers) == 1 and isinstance(layers[0], (list, tuple)) and \
                layers[0] and isinstance(layers[0][0], (list, tuple)):
            # name, layer pairs
            for name, l in layers[0]:
                self.add_sublayer(str(name), l)
        else:
            for i, l in enumerate(layers):
                if isinstance(l, tuple):
                    self.add_sublayer(str(l[0]), l[1])
                else:
                    self.add_sublayer(str(i), l)

    def forward(self, x):
        for l in self._modules.values():
            x = l(x)
        return x

    def __getitem__(self, idx):
        return list(self._modules.values())[idx]

    def __len__(self):
        return len(self._modules)


class LayerList(Layer):
    def __init__(self, sublayers=None):
        super().__init__()
        if sublayers is not None:
            for i, l in enumerate(sublayers):
                self.add_sublayer(str(i), l)

    def append(self, sublayer):
        self.add_sublayer(str(len(self._modules)), sublayer)
        return self

    def __getitem__(self, idx):
        if isinstance(idx, slice):
            return list(self._modules.values())[idx]
        if idx < 0:
            idx += len(self._modules)
        return self._modules[str(idx)]

    def __setitem__(self, idx, layer):
        self.add_sublayer(str(idx), layer)

    def __len__(self):
        return len(self._modules)

    def __iter__(self):
        return iter(self._modules.values())


class ParameterList(Layer):
    def __init__(self, parameters=None):
        super().__init__()
        if parameters is not None:
            for i, p in enumerate(parameters):
                self.register_parameter(str(i), p)

    def append(self, parameter):
        self.register_parameter(str(len(self._parameters)), parameter)
        return self

    def __getitem__(self, idx):
        return self._parameters[str(idx)]

    def __len__(self):
        return len(self._parameters)

    def __iter__(self):
        return iter(self._parameters.values())
