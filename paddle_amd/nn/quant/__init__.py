"""paddle.nn.quant (reference: python/paddle/nn/quant/__init__.py --
weight_only linear + quant stubs)."""
from ...quantization import (  # noqa: F401
    weight_only_linear,
    weight_quantize,
)


def weight_dequantize(qweight, scale, algo="weight_only_int8", out_dtype=None):
    import torch
    # weight_quantize layout: qweight [K, N], per-output-channel scale [N]
    w = qweight.float() * scale.unsqueeze(0) / 127.0
    return w.to(out_dtype or torch.float16)


def llm_int8_linear(x, qweight, scale, bias=None, threshold=6.0):
    return weight_only_linear(x, qweight, scale, bias)
