"""fp8 (OCP e4m3) compute path for gfx950 MFMA via hipBLASLt scaled-mm.

BASELINE config 5 support: GPT-MoE expert GEMMs in fp8.  Recipe
(standard delayed-scaling-free variant): per-tensor abs-max scaling at
call time, e4m3 storage, bf16 accum output; backward keeps dgrad/wgrad
in bf16 (fp8 wgrad is a round-2 item).

gfx950 note: OCP e4m3fn (max normal 448) -- NOT the MI300 fnuz variant
(guide §4).
"""
from __future__ import annotations

import torch

E4M3_MAX = 448.0


def _quant(x: torch.Tensor):
    amax = x.abs().amax().clamp(min=1e-12).float()
    scale = E4M3_MAX / amax
    q = (x.float() * scale).clamp(-E4M3_MAX, E4M3_MAX).to(torch.float8_e4m3fn)
    return q, (1.0 / scale).reshape(1)


def _quant_cached_t(w):
    """Per-expert fp8 weights, pre-transposed to the NT kernel's Bt [N,K]
    layout and cached per parameter version (one quantize per step)."""
    c = getattr(w, "_pa_q8t", None)
    if c is not None and c[0] == w._version:
        return c[1], c[2]
    q, s = _quant(w.transpose(1, 2).contiguous())   # [E, out, in] e4m3
    try:
        w._pa_q8t = (w._version, q, s)
    except (AttributeError, RuntimeError):
        pass
    return q, s


class _Fp8GroupedFFN(torch.autograd.Function):
    """Stacked-expert fp8 FFN on the own MX MFMA kernel (gemm_fp8.hip):
    activations quantized once per call, weights quantized+transposed once
    per STEP (version-cached); backward stays bf16 batched GEMMs (fp32 bmm
    has no MFMA path on CDNA4 -- the round-1 fp32 backward was the 3x
    slowdown)."""

    @staticmethod
    def forward(ctx, x, w1, b1, w2, b2):
        from ..ops import functional as hot
        if not x.is_cuda:
            h = (x.float() @ w1.float() + b1.float().unsqueeze(1))
            g = torch.nn.functional.gelu(h)
            out = (g @ w2.float() + b2.float().unsqueeze(1)).to(x.dtype)
            ctx.save_for_backward(x, h.to(x.dtype), w1, b1, w2, b2)
            return out
        from .. import _ext
        C = _ext.get_ext()
        qw1t, sw1 = _quant_cached_t(w1)
        qw2t, sw2 = _quant_cached_t(w2)
        # single-pass fused |x| amax (x.abs().amax() materializes |x|),
        # fused quantize, ONE grouped-GEMM launch with the per-expert bias
        # in the GEMM epilogue
        ax = C.amax_abs(x.contiguous()).clamp(min=1e-12)
        qx = C.quant_fp8(x.contiguous(), float(E4M3_MAX / ax))
        s1 = float(ax / E4M3_MAX) * float(sw1)
        h = C.gemm_fp8_nt_batched(qx, qw1t, s1, b1.contiguous())
        g = hot.bias_gelu(h, None)
        ag = C.amax_abs(g).clamp(min=1e-12)
        qg = C.quant_fp8(g.contiguous(), float(E4M3_MAX / ag))
        s2 = float(ag / E4M3_MAX) * float(sw2)
        out = C.gemm_fp8_nt_batched(qg, qw2t, s2, b2.contiguous())
        ctx.save_for_backward(x, h, w1, b1, w2, b2)
        return out

    @staticmethod
    def backward(ctx, dy):
        from ..ops import functional as hot
        x, h, w1, b1, w2, b2 = ctx.saved_tensors
        dt = x.dtype
        dyc = dy.to(dt).contiguous()
        g = hot.bias_gelu(h, None)                       # recompute gelu(h)
        from .. import _ext
        native = _ext.use_native(h) and w1.dtype == torch.bfloat16
        # NOTE: strided-batched bf16 GEMM with a TRANSPOSED B operand
        # memory-faults in this ROCm/hipBLASLt build (tools/moe_prof.py
        # bisect); use the own batched NT kernel (stored weights ARE the
        # NT B operand) or materialize transposes on the fallback path
        if native:
            C = _ext.get_ext()
            dg = C.gemm_bf16_nt_batched(dyc, w2)
        else:
            dg = torch.bmm(dyc, w2.transpose(1, 2).contiguous().to(dt))
        dw2 = torch.bmm(g.transpose(1, 2), dyc)
        db2 = dyc.sum(1, dtype=torch.float32).to(b2.dtype)
        if _ext.use_native(h):
            dz = _ext.get_ext().bias_gelu_bwd(dg.contiguous(), h, None)
        else:
            hf = h.float()
            cdf = 0.5 * (1 + torch.erf(hf * 0.7071067811865476))
            pdf = 0.3989422804014327 * torch.exp(-0.5 * hf * hf)
            dz = (dg.float() * (cdf + hf * pdf)).to(dt)
        if native:
            dx = C.gemm_bf16_nt_batched(dz, w1)
        else:
            dx = torch.bmm(dz, w1.transpose(1, 2).contiguous().to(dt))
        dw1 = torch.bmm(x.transpose(1, 2), dz)
        db1 = dz.sum(1, dtype=torch.float32).to(b1.dtype)
        return dx, dw1.to(w1.dtype), db1, dw2.to(w2.dtype), db2


def _own_fp8_ok(x):
    from .. import _ext
    return x.is_cuda and _ext.has_ext()


def fp8_gemm_nt(qx, qwt, scale_ab, bias=None):
    """C = scale_ab * qx @ qwt^T on the own MX MFMA kernel
    (csrc/kernels/gemm_fp8.hip; ~2x the bf16 MFMA rate)."""
    from .. import _ext
    C = _ext.get_ext()
    return C.gemm_fp8_nt(qx, qwt, float(scale_ab), bias)


class _Fp8Matmul(torch.autograd.Function):
    """out = x @ w  with x,w quantized to e4m3 for the MFMA fp8 path."""

    @staticmethod
    def forward(ctx, x, w):
        ctx.save_for_backward(x, w)
        x2 = x.reshape(-1, x.shape[-1])
        # dispatch note (measured, profiles/gemm_fp8_r02.txt): own MX kernel
        # 1.6-2.0 PF > bf16 hipBLASLt 1.2-1.3 PF; hipBLASLt fp8 via
        # _scaled_mm 2.4-3.3 PF stays the default (autotune-between-impls)
        if x.is_cuda and not hasattr(torch, "_scaled_mm") and _own_fp8_ok(x) \
                and x2.shape[1] % 128 == 0:
            qx, sx = _quant(x2)
            qw, sw = _quant(w)
            out = fp8_gemm_nt(qx.contiguous(), qw.t().contiguous(),
                              float(sx) * float(sw))
        elif x.is_cuda:
            qx, sx = _quant(x2)
            qw, sw = _quant(w)
            # _scaled_mm requires column-major B: pass w^T's transpose view
            wt = qw.t().contiguous().t()
            out = torch._scaled_mm(qx, wt, scale_a=sx.to(x.device),
                                   scale_b=sw.to(x.device),
                                   out_dtype=torch.bfloat16)
        else:
            out = (x2.float() @ w.float()).to(torch.bfloat16)
        return out.reshape(*x.shape[:-1], w.shape[-1])

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        dy2 = dy.reshape(-1, dy.shape[-1])
        x2 = x.reshape(-1, x.shape[-1])
        dx = (dy2 @ w.t().to(dy.dtype)).reshape(x.shape)
        dw = x2.t().to(dy.dtype) @ dy2
        return dx, dw.to(w.dtype)


def fp8_matmul(x, w):
    return _Fp8Matmul.apply(x, w)


class Fp8Linear(torch.nn.Module):
    """Drop-in Linear with fp8 forward GEMM (paddle layout [in, out])."""

    def __init__(self, in_features, out_features, has_bias=True, dtype=torch.bfloat16):
        super().__init__()
        self.weight = torch.nn.Parameter(
            torch.randn(in_features, out_features, dtype=dtype) * (in_features ** -0.5))
        self.bias = torch.nn.Parameter(torch.zeros(out_features, dtype=dtype)) \
            if has_bias else None

    def forward(self, x):
        out = fp8_matmul(x, self.weight)
        if self.bias is not None:
            out = out + self.bias
        return out


def convert_experts_to_fp8(moe_model):
    """Swap MoE expert compute to the fp8 GEMM path in-place -- handles
    both the grouped (stacked-weight bmm) and per-expert layouts."""
    from ..models.moe import ExpertMLP, GroupedExperts
    for mod in moe_model.modules():
        if isinstance(mod, GroupedExperts):
            mod.fp8 = True

            def gfwd(self, x):
                return _Fp8GroupedFFN.apply(x, self.w1, self.b1, self.w2,
                                            self.b2)

            mod.forward = gfwd.__get__(mod)
        if isinstance(mod, ExpertMLP):
            mod.fp8 = True

            def fwd(self, x):
                h = fp8_matmul(x, self.fc1.weight) + self.fc1.bias
                from ..ops import functional as hot
                h = torch.nn.functional.gelu(h.float()).to(x.dtype) if not x.is_cuda \
                    else hot.bias_gelu(h, None)
                return fp8_matmul(h, self.fc2.weight) + self.fc2.bias

            mod.forward = fwd.__get__(mod)
    return moe_model


def convert_linears_to_fp8(model, min_features=256, skip=("lm_head",)):
    """Swap paddle Linear forwards to fp8 GEMMs in place (full-model fp8
    training recipe; small layers and skipped names stay bf16 --
    reference: the fp8 quant variants of fused_bias_act / cublaslt fp8
    path in paddle/phi/kernels/fusion/).  Returns the converted count."""
    n = 0
    for name, mod in model.named_modules():
        if type(mod).__name__ != "Linear" or not hasattr(mod, "weight"):
            continue
        if mod.weight.dim() != 2:
            continue
        if any(s in name for s in skip):
            continue
        in_f, out_f = mod.weight.shape
        if min(in_f, out_f) < min_features:
            continue

        def fwd(self, x):
            out = fp8_matmul(x, self.weight)
            if getattr(self, "bias", None) is not None:
                out = out + self.bias
            return out

        mod.forward = fwd.__get__(mod)
        n += 1
    return n
