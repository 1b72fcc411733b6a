"""paddle.nn.initializer parity (python/paddle/nn/initializer/)."""
from __future__ import annotations

import math

import torch


class Initializer:
    def __call__(self, tensor):
        _apply_initializer(self, tensor)


class Constant(Initializer):
    def __init__(self, value=0.0):
        self.value = value


class Normal(Initializer):
    def __init__(self, mean=0.0, std=1.0):
        self.mean, self.std = mean, std


class TruncatedNormal(Initializer):
    def __init__(self, mean=0.0, std=1.0):
        self.mean, self.std = mean, std


class Uniform(Initializer):
    def __init__(self, low=-1.0, high=1.0):
        self.low, self.high = low, high


class XavierNormal(Initializer):
    def __init__(self, fan_in=None, fan_out=None, gain=1.0):
        self.fan_in, self.fan_out, self.gain = fan_in, fan_out, gain


class XavierUniform(Initializer):
    def __init__(self, fan_in=None, fan_out=None, gain=1.0):
        self.fan_in, self.fan_out, self.gain = fan_in, fan_out, gain


class KaimingNormal(Initializer):
    def __init__(self, fan_in=None, negative_slope=0.0, nonlinearity="relu"):
        self.fan_in = fan_in
        self.negative_slope = negative_slope
        self.nonlinearity = nonlinearity


class KaimingUniform(KaimingNormal):
    pass


class Assign(Initializer):
    def __init__(self, value):
        self.value = value


def _fans(t: torch.Tensor):
    if t.dim() < 2:
        return t.numel(), t.numel()
    # paddle convention: fan_in = shape[0]*receptive, fan_out = shape[1]*receptive
    receptive = 1
    for s in t.shape[2:]:
        receptive *= s
    return t.shape[0] * receptive, t.shape[1] * receptive


@torch.no_grad()
def _apply_initializer(init, t: torch.Tensor):
    if isinstance(init, Constant):
        t.fill_(init.value)
    elif isinstance(init, Normal):
        t.copy_(torch.randn_like(t, dtype=torch.float32).mul_(init.std).add_(init.mean).to(t.dtype))
    elif isinstance(init, TruncatedNormal):
        f = torch.empty(t.shape, dtype=torch.float32, device=t.device)
        torch.nn.init.trunc_normal_(f, mean=init.mean, std=init.std,
                                    a=init.mean - 2 * init.std, b=init.mean + 2 * init.std)
        t.copy_(f.to(t.dtype))
    elif isinstance(init, Uniform):
        t.copy_((torch.rand_like(t, dtype=torch.float32) * (init.high - init.low) + init.low).to(t.dtype))
    elif isinstance(init, (XavierNormal, XavierUniform)):
        fi, fo = _fans(t)
        fi = init.fan_in if init.fan_in is not None else fi
        fo = init.fan_out if init.fan_out is not None else fo
        if isinstance(init, XavierNormal):
            std = init.gain * math.sqrt(2.0 / (fi + fo))
            t.copy_(torch.randn_like(t, dtype=torch.float32).mul_(std).to(t.dtype))
        else:
            limit = init.gain * math.sqrt(6.0 / (fi + fo))
            t.copy_((torch.rand_like(t, dtype=torch.float32) * 2 * limit - limit).to(t.dtype))
    elif isinstance(init, (KaimingNormal, KaimingUniform)):
        fi, _ = _fans(t)
        fi = init.fan_in if init.fan_in is not None else fi
        gain = math.sqrt(2.0 / (1 + init.negative_slope ** 2))
        if isinstance(init, KaimingUniform):   # subclass: must test FIRST
            limit = gain * math.sqrt(3.0 / fi)
            t.copy_((torch.rand_like(t, dtype=torch.float32) * 2 * limit - limit).to(t.dtype))
        else:
            std = gain / math.sqrt(fi)
            t.copy_(torch.randn_like(t, dtype=torch.float32).mul_(std).to(t.dtype))
    elif isinstance(init, Assign):
        v = init.value
        if not isinstance(v, torch.Tensor):
            import numpy as np
            v = torch.as_tensor(np.asarray(v))
        t.copy_(v.to(dtype=t.dtype, device=t.device))
    elif callable(init):
        init(t)
    else:
        raise TypeError(f"unknown initializer {init!r}")


def set_global_initializer(weight_init=None, bias_init=None):
    # registry hook for API parity; per-layer attrs take precedence
    global _GLOBAL_WEIGHT_INIT, _GLOBAL_BIAS_INIT
    _GLOBAL_WEIGHT_INIT, _GLOBAL_BIAS_INIT = weight_init, bias_init


_GLOBAL_WEIGHT_INIT = None
_GLOBAL_BIAS_INIT = None


class Bilinear(Initializer):
    """Bilinear upsampling kernel init (reference: nn/initializer/Bilinear)."""

    def __call__(self, tensor):
        import math
        with torch.no_grad():
            h, w = tensor.shape[-2], tensor.shape[-1]
            f = math.ceil(w / 2)
            c = (2 * f - 1 - f % 2) / (2.0 * f)
            og_y = torch.arange(h, dtype=torch.float32).reshape(-1, 1)
            og_x = torch.arange(w, dtype=torch.float32).reshape(1, -1)
            filt = (1 - (og_x / f - c).abs()) * (1 - (og_y / f - c).abs())
            tensor.copy_(filt.expand_as(tensor))
        return tensor


class Orthogonal(Initializer):
    def __init__(self, gain=1.0, name=None):
        self.gain = gain

    def __call__(self, tensor):
        with torch.no_grad():
            torch.nn.init.orthogonal_(tensor, gain=self.gain)
        return tensor


class Dirac(Initializer):
    def __init__(self, groups=1, name=None):
        self.groups = groups

    def __call__(self, tensor):
        with torch.no_grad():
            torch.nn.init.dirac_(tensor, groups=self.groups)
        return tensor


def calculate_gain(nonlinearity, param=None):
    return torch.nn.init.calculate_gain(nonlinearity, param)
