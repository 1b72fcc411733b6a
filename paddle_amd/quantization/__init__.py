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


class BaseObserver(__import__("torch").nn.Module):
    """Activation-range observer (reference paddle/quantization/observer.py
    family): tracks running abs-max during calibration forwards."""

    def __init__(self, quant_bits=8):
        super().__init__()
        import torch
        self.quant_bits = quant_bits
        self.register_buffer("absmax", torch.zeros(()))

    def forward(self, x):
        import torch
        with torch.no_grad():
            m = x.detach().abs().amax().float()
            if m > self.absmax:
                self.absmax.fill_(m)
        return x

    def scale(self):
        qmax = 2 ** (self.quant_bits - 1) - 1
        return (self.absmax.clamp(min=1e-8) / qmax)


class BaseQuanter(__import__("torch").nn.Module):
    """Fake-quantizer (reference paddle/quantization/quanters/abs_max.py):
    quantize-dequantize with a straight-through estimator so QAT
    backprops through the rounding."""

    def __init__(self, quant_bits=8):
        super().__init__()
        import torch
        self.quant_bits = quant_bits
        self.register_buffer("absmax", torch.zeros(()))

    def forward(self, x):
        import torch
        qmax = 2 ** (self.quant_bits - 1) - 1
        with torch.no_grad():
            m = x.detach().abs().amax().float()
            if self.training and m > self.absmax:
                self.absmax.fill_(m)
        s = self.absmax.clamp(min=1e-8) / qmax
        q = (x / s).round().clamp(-qmax, qmax) * s
        # straight-through: forward quantized, backward identity
        return x + (q - x.to(q.dtype)).detach()


def quanter(name):
    """Reference paddle.quantization.quanter class decorator: registers a
    quanter factory under `name`."""
    def deco(cls):
        _QUANTER_REGISTRY[name] = cls
        return cls
    return deco


_QUANTER_REGISTRY = {"FakeQuanterWithAbsMax": BaseQuanter}


class QuantConfig:
    """reference paddle/quantization/config.py: which layers get which
    activation/weight quanters."""

    def __init__(self, activation=None, weight=None):
        self.activation = activation
        self.weight = weight
        self._types = None

    def add_type_config(self, layer_types, activation=None, weight=None):
        self._types = tuple(layer_types) if isinstance(layer_types, (list, tuple)) \
            else (layer_types,)
        if activation is not None:
            self.activation = activation
        if weight is not None:
            self.weight = weight

    def _match(self, layer):
        import torch
        if self._types is not None:
            return isinstance(layer, self._types)
        w = getattr(layer, "weight", None)
        return (type(layer).__name__ in ("Linear", "Conv2D", "Conv2d")
                and isinstance(w, torch.Tensor))


class _QuantedLayer(__import__("torch").nn.Module):
    """Wrapper inserted by PTQ/QAT: observer/fake-quant on the input and
    (for QAT) on the weight."""

    def __init__(self, layer, act_q, weight_q=None):
        super().__init__()
        self.layer = layer
        self.act_q = act_q
        self.weight_q = weight_q

    def forward(self, *args, **kwargs):
        args = (self.act_q(args[0]),) + args[1:]
        if self.weight_q is not None:
            w = self.layer.weight
            orig = w.data
            self.layer.weight.data = self.weight_q(orig)
            try:
                return self.layer(*args, **kwargs)
            finally:
                self.layer.weight.data = orig
        return self.layer(*args, **kwargs)


def _wrap_layers(model, config, make_act, make_w):
    n = 0
    for mod in list(model.modules()):
        for name, child in list(mod._modules.items()):
            if child is None or isinstance(child, _QuantedLayer):
                continue
            if config._match(child):
                mod._modules[name] = _QuantedLayer(child, make_act(),
                                                   make_w() if make_w else None)
                n += 1
    return n


class PTQ:
    """Post-training quantization (reference paddle/quantization/ptq.py):
    quantize() inserts observers, run calibration forwards, convert()
    bakes int8 weights + the observed activation fake-quant."""

    def __init__(self, config: QuantConfig = None):
        self.config = config or QuantConfig()

    def quantize(self, model, inplace=False):
        import copy
        m = model if inplace else copy.deepcopy(model)
        _wrap_layers(m, self.config, lambda: BaseObserver(), None)
        return m

    def convert(self, model, inplace=True):
        import torch
        qmax = 127
        for mod in model.modules():
            if isinstance(mod, _QuantedLayer):
                if isinstance(mod.act_q, BaseObserver):
                    fq = BaseQuanter()
                    fq.absmax.copy_(mod.act_q.absmax)
                    fq.eval()
                    mod.act_q = fq
                w = getattr(mod.layer, "weight", None)
                if isinstance(w, torch.Tensor):
                    with torch.no_grad():
                        s = w.abs().amax().clamp(min=1e-8) / qmax
                        w.copy_(((w / s).round().clamp(-qmax, qmax)) * s)
        return model


class QAT:
    """Quantization-aware training (reference paddle/quantization/qat.py):
    quantize() inserts straight-through fake-quanters on activations and
    weights so training sees int8 rounding."""

    def __init__(self, config: QuantConfig = None):
        self.config = config or QuantConfig()

    def quantize(self, model, inplace=False):
        import copy
        m = model if inplace else copy.deepcopy(model)
        act_cls = self.config.activation or BaseQuanter
        w_cls = self.config.weight or BaseQuanter
        act_f = act_cls if callable(act_cls) else BaseQuanter
        _wrap_layers(m, self.config, lambda: _mk(act_f), lambda: _mk(w_cls))
        return m

    def convert(self, model, inplace=True):
        for mod in model.modules():
            if isinstance(mod, _QuantedLayer):
                mod.eval()
        return model


def _mk(cls_or_inst):
    import copy
    if isinstance(cls_or_inst, type):
        return cls_or_inst()
    try:
        return copy.deepcopy(cls_or_inst)
    except Exception:
        return BaseQuanter()


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


class WeightOnlyLinear(__import__("torch").nn.Module):
    """Drop-in replacement for a (paddle-layout) nn.Linear holding int8
    weights; decode shapes run the MFMA W-streaming int8 kernel, larger
    shapes dequantize on the fly.  Reference: paddle/nn/quant
    weight-only path."""

    def __init__(self, qweight, scale, bias=None):
        super().__init__()
        self.register_buffer("qweight", qweight)
        self.register_buffer("scale", scale)
        self.register_buffer("wo_bias", bias)
        self.in_features = qweight.shape[0]
        self.out_features = qweight.shape[1]

    @classmethod
    def from_linear(cls, linear):
        qw, sc = weight_quantize(linear.weight.detach())
        dev = linear.weight.device
        bias = getattr(linear, "bias", None)
        return cls(qw.to(dev), sc.to(dev),
                   bias.detach().clone() if bias is not None else None)

    def forward(self, x):
        return weight_only_linear(x, self.qweight, self.scale, self.wo_bias)


def quantize_linears_(model, min_features=1024, skip=("lm_head",)):
    """Replace every Linear child of `model` (recursively) whose weight
    is at least [min_features, min_features] with a WeightOnlyLinear --
    int8 weights, ~2x less weight memory for serving.  Returns the
    number of layers converted."""
    import torch
    n = 0

    def walk(mod, prefix=""):
        nonlocal n
        for name, child in list(mod._modules.items()):
            if child is None:
                continue
            full = f"{prefix}{name}"
            if any(s in full for s in skip):
                continue
            w = getattr(child, "weight", None)
            if (type(child).__name__ == "Linear" and isinstance(w, torch.Tensor)
                    and w.dim() == 2 and min(w.shape) >= min_features):
                mod._modules[name] = WeightOnlyLinear.from_linear(child)
                n += 1
            else:
                walk(child, full + ".")

    walk(model)
    return n
