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


class _Fp8Matmul(torch.autograd.Function):
    """out = x @ w  with x,w quantized to e4m3 for the MFMA fp8 path."""

    @staticmethod
    def forward(ctx, x, w):
        ctx.save_for_backward(x, w)
        x2 = x.reshape(-1, x.shape[-1])
        if x.is_cuda:
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
                # per-expert fp8 GEMMs over the stacked weights
                outs = []
                for e in range(self.num_local):
                    h = fp8_matmul(x[e], self.w1[e]) + self.b1[e]
                    from ..ops import functional as hot
                    h = (torch.nn.functional.gelu(h.float()).to(x.dtype)
                         if not x.is_cuda else hot.bias_gelu(h, None))
                    outs.append(fp8_matmul(h, self.w2[e]) + self.b2[e])
                return torch.stack(outs)

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
