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


# -- weight-only int8 (serving decode path) ----------------------------------
# parity: paddle/phi/kernels/weight_quantize_kernel.cu + funcs/weight_only_gemv.cu
def weight_quantize(w, algo="weight_only_int8"):
    """w [K, N] (paddle Linear layout) -> (qweight [K, N] int8 row-major,
    scale [N] fp32).  Per-output-channel absmax; w ~ qw * scale / 127.

    The [K, N] layout feeds the MFMA W-streaming decode kernel with
    coalesced rows (the round-1 [N, K] GEMV layout read 180-330 GB/s;
    the streamer reads at the memory roofline on HALF the bytes)."""
    import torch
    assert algo in ("weight_only_int8",)
    wf = w.detach().float()                          # [K, N]
    scale = wf.abs().amax(dim=0).clamp(min=1e-8)     # [N]
    q = torch.round(wf / scale.unsqueeze(0) * 127.0).clamp(-127, 127).to(torch.int8)
    return q.contiguous(), scale


def weight_only_linear(x, qweight, scale, bias=None, weight_dtype="int8"):
    """x [..., K] @ dequant(qweight) + bias.  Decode shapes (<= 32 rows)
    run the int8 MFMA W-streaming kernel; fallback dequantizes."""
    import torch
    from .. import _ext
    k, n = qweight.shape
    if x.is_cuda and _ext.use_native(x):
        rows = x.numel() // x.shape[-1]
        if rows <= 32 and n % 256 == 0 and k % 64 == 0 and x.dtype == torch.bfloat16:
            C = _ext.get_ext()
            b = bias.to(torch.bfloat16) if bias is not None else None
            out = C.decode_gemm_int8(x.reshape(-1, k), qweight,
                                     scale.float(), b)
            return out.reshape(*x.shape[:-1], n)
    w = (qweight.float() * scale.unsqueeze(0) / 127.0).to(x.dtype)  # [K,N]
    out = x @ w
    if bias is not None:
        out = out + bias
    return out
