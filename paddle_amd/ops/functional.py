"""Autograd-integrated wrappers over the gfx950 HIP kernels.

Each op is a torch.autograd.Function whose forward dispatches to
paddle_amd._C on GPU (mandatory -- _ext.use_native raises if the
extension is missing) and to an fp32 torch reference on CPU.  The CPU
path doubles as the numerics oracle in tests/.

Reference op-signature parity anchors are cited per-op.
"""
from __future__ import annotations

import math
from typing import Optional

import torch

from .. import _ext

# jit.save tracing mode: emit plain substrate ops (no autograd.Function
# wrappers -- TorchScript cannot export Python function calls).  Forward
# inference semantics only; toggled by paddle_amd.jit._substrate_only.
_TRACE_SUBSTRATE = False


def _tracing():
    return _TRACE_SUBSTRATE


# ---------------------------------------------------------------------------
# layer_norm (paddle/phi/kernels/gpu/layer_norm_kernel.cu parity)
# ---------------------------------------------------------------------------
class _LayerNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b, eps):
        if _ext.use_native(x):
            C = _ext.get_ext()
            y, mean, rstd = C.layer_norm_fwd(x.contiguous(), w.contiguous(),
                                             b.contiguous() if b is not None else None, eps)
        else:
            xf = x.float()
            mean = xf.mean(-1).reshape(-1)
            var = xf.var(-1, unbiased=False).reshape(-1)
            rstd = torch.rsqrt(var + eps)
            d = x.shape[-1]
            xhat = (xf - mean.view(*x.shape[:-1], 1)) * rstd.view(*x.shape[:-1], 1)
            y = xhat * w.float() + (b.float() if b is not None else 0.0)
            y = y.to(x.dtype)
        ctx.save_for_backward(x, w, mean, rstd)
        ctx.has_bias = b is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, mean, rstd = ctx.saved_tensors
        if _ext.use_native(x):
            C = _ext.get_ext()
            dx, dw, db = C.layer_norm_bwd(dy.contiguous(), x, w, mean, rstd, ctx.has_bias)
        else:
            d = x.shape[-1]
            xf = x.float()
            dyf = dy.float()
            mu = mean.view(*x.shape[:-1], 1)
            rs = rstd.view(*x.shape[:-1], 1)
            xhat = (xf - mu) * rs
            g = dyf * w.float()
            s1 = (g * xhat).mean(-1, keepdim=True)
            s2 = g.mean(-1, keepdim=True)
            dx = (rs * (g - s2 - xhat * s1)).to(x.dtype)
            dw = (dyf * xhat).reshape(-1, d).sum(0).to(w.dtype)
            db = dyf.reshape(-1, d).sum(0).to(w.dtype)
        return dx, dw, (db if ctx.has_bias else None), None


def layer_norm(x, weight, bias=None, epsilon=1e-5):
    if _tracing():
        return torch.nn.functional.layer_norm(x, (x.shape[-1],), weight, bias,
                                              epsilon)
    return _LayerNorm.apply(x, weight, bias, epsilon)


# ---------------------------------------------------------------------------
# rms_norm (+ optional fused residual add)
# paddle parity: incubate/nn/functional/fused_rms_norm.py, rms_norm_kernel.cu
# ---------------------------------------------------------------------------
class _RMSNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, eps):
        if _ext.use_native(x):
            C = _ext.get_ext()
            y, rstd = C.rms_norm_fwd(x.contiguous(), None, w.contiguous(), eps)
        else:
            xf = x.float()
            rstd = torch.rsqrt(xf.square().mean(-1) + eps).reshape(-1)
            y = (xf * rstd.view(*x.shape[:-1], 1) * w.float()).to(x.dtype)
        ctx.save_for_backward(x, w, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, rstd = ctx.saved_tensors
        if _ext.use_native(x):
            C = _ext.get_ext()
            dx, dw = C.rms_norm_bwd(dy.contiguous(), x, w, rstd)
        else:
            d = x.shape[-1]
            xf, dyf = x.float(), dy.float()
            rs = rstd.view(*x.shape[:-1], 1)
            xhat = xf * rs
            g = dyf * w.float()
            s1 = (g * xhat).mean(-1, keepdim=True)
            dx = (rs * (g - xhat * s1)).to(x.dtype)
            dw = (dyf * xhat).reshape(-1, d).sum(0).to(w.dtype)
        return dx, dw, None


def rms_norm(x, weight, epsilon=1e-6):
    if _tracing():
        xf = x.float()
        y = xf * torch.rsqrt(xf.square().mean(-1, keepdim=True) + epsilon)
        return (y * weight.float()).to(x.dtype)
    return _RMSNorm.apply(x, weight, epsilon)


class _FusedRMSNormResidual(torch.autograd.Function):
    """y, xr = rms_norm(x + residual) -- returns normed out and the new
    residual stream (paddle fused_rms_norm with residual)."""

    @staticmethod
    def forward(ctx, x, residual, w, eps):
        if _ext.use_native(x):
            C = _ext.get_ext()
            y, rstd, xr = C.rms_norm_fwd(x.contiguous(), residual.contiguous(), w.contiguous(), eps)
        else:
            xr = x + residual
            xf = xr.float()
            rstd = torch.rsqrt(xf.square().mean(-1) + eps).reshape(-1)
            y = (xf * rstd.view(*x.shape[:-1], 1) * w.float()).to(x.dtype)
        ctx.save_for_backward(xr, w, rstd)
        return y, xr

    @staticmethod
    def backward(ctx, dy, dxr):
        xr, w, rstd = ctx.saved_tensors
        if _ext.use_native(xr):
            C = _ext.get_ext()
            dx, dw = C.rms_norm_bwd(dy.contiguous(), xr, w, rstd)
        else:
            d = xr.shape[-1]
            xf, dyf = xr.float(), dy.float()
            rs = rstd.view(*xr.shape[:-1], 1)
            xhat = xf * rs
            g = dyf * w.float()
            s1 = (g * xhat).mean(-1, keepdim=True)
            dx = (rs * (g - xhat * s1)).to(xr.dtype)
            dw = (dyf * xhat).reshape(-1, d).sum(0).to(w.dtype)
        if dxr is not None:
            dx = dx + dxr
        return dx, dx, dw, None


def fused_rms_norm(x, norm_weight, residual=None, epsilon=1e-6):
    if _tracing() and residual is not None:
        xr = x + residual
        return rms_norm(xr, norm_weight, epsilon), xr
    if residual is None:
        return rms_norm(x, norm_weight, epsilon)
    return _FusedRMSNormResidual.apply(x, residual, norm_weight, epsilon)


# ---------------------------------------------------------------------------
# softmax cross entropy (cross_entropy_kernel.cu parity; hard labels)
# ---------------------------------------------------------------------------
class _SoftmaxCE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, labels, ignore_index):
        shp = logits.shape
        v = shp[-1]
        l2 = logits.reshape(-1, v)
        lab = labels.reshape(-1)
        if _ext.use_native(logits):
            C = _ext.get_ext()
            loss, lse = C.softmax_ce_fwd(l2.contiguous(), lab.contiguous(), ignore_index)
        else:
            lf = l2.float()
            lse = torch.logsumexp(lf, -1)
            safe = lab.clamp(min=0)
            picked = lf.gather(1, safe.unsqueeze(1)).squeeze(1)
            loss = torch.where(lab == ignore_index, torch.zeros_like(lse), lse - picked)
        ctx.save_for_backward(l2, lab, lse)
        ctx.ignore_index = ignore_index
        ctx.in_shape = shp
        return loss.reshape(shp[:-1])

    @staticmethod
    def backward(ctx, dloss):
        l2, lab, lse = ctx.saved_tensors
        dl = dloss.reshape(-1).float().contiguous()
        if _ext.use_native(l2):
            C = _ext.get_ext()
            dlogits = C.softmax_ce_bwd(dl, l2, lab, lse, ctx.ignore_index)
        else:
            p = torch.exp(l2.float() - lse.unsqueeze(1))
            onehot = torch.zeros_like(p)
            safe = lab.clamp(min=0)
            onehot.scatter_(1, safe.unsqueeze(1), 1.0)
            g = torch.where((lab == ctx.ignore_index).unsqueeze(1),
                            torch.zeros_like(dl).unsqueeze(1), dl.unsqueeze(1))
            dlogits = (g * (p - onehot)).to(l2.dtype)
        return dlogits.reshape(ctx.in_shape), None, None


def softmax_cross_entropy(logits, labels, ignore_index=-100, reduction="none"):
    if _tracing():
        v = logits.shape[-1]
        loss = torch.nn.functional.cross_entropy(
            logits.reshape(-1, v).float(), labels.reshape(-1),
            ignore_index=ignore_index, reduction="none").reshape(labels.shape)
    else:
        loss = _SoftmaxCE.apply(logits, labels, ignore_index)
    if reduction == "mean":
        n_valid = (labels != ignore_index).sum().clamp(min=1)
        return loss.sum() / n_valid.to(loss.dtype)
    if reduction == "sum":
        return loss.sum()
    return loss


# ---------------------------------------------------------------------------
# bias + gelu (fused_bias_act parity; erf gelu)
# ---------------------------------------------------------------------------
class _BiasGelu(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, bias):
        ctx.save_for_backward(x, bias if bias is not None else torch.empty(0))
        ctx.has_bias = bias is not None
        if _ext.use_native(x):
            C = _ext.get_ext()
            return C.bias_gelu_fwd(x.contiguous(), bias.contiguous() if bias is not None else None)
        v = x.float() + (bias.float() if bias is not None else 0.0)
        return torch.nn.functional.gelu(v).to(x.dtype)

    @staticmethod
    def backward(ctx, dy):
        x, bias = ctx.saved_tensors
        b = bias if ctx.has_bias else None
        # d/dx gelu(x+b) == d/db gelu(x+b), so db = colsum(dx)
        if _ext.use_native(x):
            C = _ext.get_ext()
            dx = C.bias_gelu_bwd(dy.contiguous(), x, b.contiguous() if b is not None else None)
            db = C.colsum(dx).to(dy.dtype) if ctx.has_bias else None
        else:
            v = x.float() + (b.float() if b is not None else 0.0)
            cdf = 0.5 * (1.0 + torch.erf(v * 0.7071067811865476))
            pdf = 0.3989422804014327 * torch.exp(-0.5 * v * v)
            dx = (dy.float() * (cdf + v * pdf)).to(x.dtype)
            db = dx.float().reshape(-1, dx.shape[-1]).sum(0).to(dy.dtype) if ctx.has_bias else None
        return dx, db


def bias_gelu(x, bias=None):
    if _tracing():
        return torch.nn.functional.gelu(x + bias if bias is not None else x)
    return _BiasGelu.apply(x, bias)


# ---------------------------------------------------------------------------
# swiglu: silu(x[..., :d]) * x[..., d:]
# paddle parity: incubate/nn/functional/swiglu (fused_bias_act swiglu)
# ---------------------------------------------------------------------------
class _SwiGLU(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ctx.save_for_backward(x)
        if _ext.use_native(x):
            C = _ext.get_ext()
            return C.swiglu_fwd(x.contiguous())
        d = x.shape[-1] // 2
        g, u = x[..., :d].float(), x[..., d:].float()
        return (torch.nn.functional.silu(g) * u).to(x.dtype)

    @staticmethod
    def backward(ctx, dy):
        (x,) = ctx.saved_tensors
        if _ext.use_native(x):
            C = _ext.get_ext()
            return C.swiglu_bwd(dy.contiguous(), x)
        d = x.shape[-1] // 2
        g, u = x[..., :d].float(), x[..., d:].float()
        sig = torch.sigmoid(g)
        silu = g * sig
        dg = dy.float() * u * (sig * (1 + g * (1 - sig)))
        du = dy.float() * silu
        return torch.cat([dg, du], dim=-1).to(x.dtype)


def swiglu(x, y=None):
    if _tracing():
        if y is None:
            g, u = x.chunk(2, -1)
        else:
            g, u = x, y
        return torch.nn.functional.silu(g) * u
    if y is not None:
        x = torch.cat([x, y], dim=-1)
    return _SwiGLU.apply(x)


# ---------------------------------------------------------------------------
# rotary embedding (fused_rope_kernel.cu RotateHalf parity; neox style)
# ---------------------------------------------------------------------------
_rope_cache = {}


def build_rope_cache(seq_len, head_dim, base=10000.0, device=None, dtype=torch.float32):
    key = (seq_len, head_dim, base, str(device))
    if key not in _rope_cache:
        half = head_dim // 2
        inv = 1.0 / (base ** (torch.arange(0, half, dtype=torch.float32, device=device) / half))
        t = torch.arange(seq_len, dtype=torch.float32, device=device)
        freqs = torch.outer(t, inv)  # [S, half]
        _rope_cache[key] = (freqs.cos().contiguous(), freqs.sin().contiguous())
    return _rope_cache[key]


class _Rope(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, cos_t, sin_t, pos_offset):
        ctx.save_for_backward(cos_t, sin_t)
        ctx.pos_offset = pos_offset
        if _ext.use_native(x):
            C = _ext.get_ext()
            return C.rope_fwd(x.contiguous(), cos_t, sin_t, pos_offset, False)
        return _rope_ref(x, cos_t, sin_t, pos_offset, conj=False)

    @staticmethod
    def backward(ctx, dy):
        cos_t, sin_t = ctx.saved_tensors
        if _ext.use_native(dy):
            C = _ext.get_ext()
            return C.rope_fwd(dy.contiguous(), cos_t, sin_t, ctx.pos_offset, True), None, None, None
        return _rope_ref(dy, cos_t, sin_t, ctx.pos_offset, conj=True), None, None, None


def _rope_ref(x, cos_t, sin_t, pos_offset, conj):
    # x: [B, S, H, D]
    b, s, h, d = x.shape
    half = d // 2
    c = cos_t[pos_offset:pos_offset + s].view(1, s, 1, half).float()
    sn = sin_t[pos_offset:pos_offset + s].view(1, s, 1, half).float()
    if conj:
        sn = -sn
    x1, x2 = x[..., :half].float(), x[..., half:].float()
    y1 = x1 * c - x2 * sn
    y2 = x2 * c + x1 * sn
    return torch.cat([y1, y2], dim=-1).to(x.dtype)


def fused_rotary_position_embedding(q, k=None, v=None, sin=None, cos=None,
                                    position_ids=None, use_neox_rotary_style=True,
                                    base=10000.0, pos_offset=0):
    if _tracing():
        b, s, h, d = q.shape
        if cos is None or sin is None:
            cos_t, sin_t = build_rope_cache(s + pos_offset, d, base, q.device)
        else:
            cos_t = cos.float().reshape(-1, d)[..., : d // 2]
            sin_t = sin.float().reshape(-1, d)[..., : d // 2]
        outs = [_rope_ref(q.float(), cos_t, sin_t, pos_offset, False).to(q.dtype)]
        if k is not None:
            outs.append(_rope_ref(k.float(), cos_t, sin_t, pos_offset, False).to(k.dtype))
        if v is not None:
            outs.append(v)
        return tuple(outs) if len(outs) > 1 else outs[0]
    """paddle.incubate.nn.functional.fused_rotary_position_embedding parity
    (SURVEY.md A.7).  q/k/v: [B, S, H, D]."""
    bq, s, hq, d = q.shape
    if cos is None or sin is None:
        cos_t, sin_t = build_rope_cache(s + pos_offset, d, base, q.device)
    else:
        cos_t, sin_t = cos.float().reshape(-1, d)[..., : d // 2].contiguous(), \
                       sin.float().reshape(-1, d)[..., : d // 2].contiguous()
    outs = [_Rope.apply(q, cos_t, sin_t, pos_offset)]
    if k is not None:
        outs.append(_Rope.apply(k, cos_t, sin_t, pos_offset))
    if v is not None:
        outs.append(v)
    return tuple(outs) if len(outs) > 1 else outs[0]


# ---------------------------------------------------------------------------
# flash attention (flash_attn_kernel.cu API parity)
# ---------------------------------------------------------------------------
def _fa_native_ok(q):
    return (_ext.use_native(q) and q.dtype == torch.bfloat16
            and q.shape[-1] in (64, 128) and q.stride(-1) == 1)


def _fa_seed_offset(fixed_seed_offset=None):
    """Reference dropout RNG contract (funcs/dropout_impl.cu.h:129): a
    (seed, offset) pair; fixed_seed_offset pins it (recompute replays the
    torch CPU generator state, so the default path is replay-deterministic
    under fleet.recompute as well)."""
    if fixed_seed_offset is not None:
        return int(fixed_seed_offset[0]), int(fixed_seed_offset[1])
    seed = int(torch.initial_seed()) & 0x7FFFFFFFFFFFFFFF
    offset = int(torch.randint(0, 2 ** 31, (1,)).item())
    return seed, offset


class _FlashAttn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, scale, causal, mask=None, dropout=0.0,
                seed=0, offset=0):
        # q,k,v: [B, H, S, D] bf16 (views allowed; D contiguous)
        if _fa_native_ok(q):
            C = _ext.get_ext()
            mk = mask.to(torch.bfloat16) if mask is not None else None
            o, lse = C.flash_attn_fwd(q, k, v, None, scale, causal, mk,
                                      dropout, seed, offset)
            ctx.native = True
            ctx.mask = mk
        else:
            dm = None
            if dropout > 0.0:
                g = torch.Generator(device="cpu").manual_seed(
                    (seed * 1000003 + offset) & 0x7FFFFFFFFFFFFFFF)
                dm = (torch.rand(q.shape[0], q.shape[1], q.shape[2], k.shape[2],
                                 generator=g) >= dropout).to(q.device)
            o, lse = _sdpa_ref(q, k, v, scale, causal, mask, dm, dropout)
            ctx.native = False
            ctx.mask = mask
            ctx.drop_mask = dm
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.scale, ctx.causal = scale, causal
        ctx.dropout, ctx.seed, ctx.offset = dropout, seed, offset
        ctx.mark_non_differentiable(lse)
        return o, lse

    @staticmethod
    def backward(ctx, do, _dlse):
        q, k, v, o, lse = ctx.saved_tensors
        if ctx.native:
            C = _ext.get_ext()
            if do.stride(-1) != 1:
                do = do.contiguous()
            kw = dict(scale=ctx.scale, causal=ctx.causal, mask=ctx.mask,
                      pdrop=ctx.dropout, seed=ctx.seed, offset=ctx.offset)
            if k.shape[1] != q.shape[1]:  # GQA: expand for bwd kernel
                rep = q.shape[1] // k.shape[1]
                ke = k.repeat_interleave(rep, dim=1)
                ve = v.repeat_interleave(rep, dim=1)
                dq, dke, dve = C.flash_attn_bwd(do, q, ke.contiguous(), ve.contiguous(),
                                                o, lse, **kw)
                hkv = k.shape[1]
                dk = dke.view(k.shape[0], hkv, rep, k.shape[2], k.shape[3]).sum(2)
                dv = dve.view(v.shape[0], hkv, rep, v.shape[2], v.shape[3]).sum(2)
            else:
                dq, dk, dv = C.flash_attn_bwd(do, q, k, v, o, lse, **kw)
        else:
            dq, dk, dv = _sdpa_ref_bwd(do, q, k, v, lse, ctx.scale, ctx.causal,
                                       ctx.mask, getattr(ctx, "drop_mask", None),
                                       ctx.dropout)
        return dq, dk, dv, None, None, None, None, None, None


class _QKVFlashAttn(torch.autograd.Function):
    """Zero-copy causal attention on a packed [B, S, 3, H, D] qkv tensor
    (the GPT hot path): q/k/v are strided views, the output is written
    directly in [B, S, H*D] layout, and backward writes dqkv in place --
    no transpose/cat kernels at all."""

    @staticmethod
    def forward(ctx, qkv, scale, causal, dropout=0.0, seed=0, offset=0):
        b, s, three, h, d = qkv.shape
        q = qkv[:, :, 0].permute(0, 2, 1, 3)  # [b,h,s,d] strided view
        k = qkv[:, :, 1].permute(0, 2, 1, 3)
        v = qkv[:, :, 2].permute(0, 2, 1, 3)
        if _fa_native_ok(q):
            C = _ext.get_ext()
            out = torch.empty(b, s, h, d, dtype=qkv.dtype, device=qkv.device)
            o_view = out.permute(0, 2, 1, 3)
            _, lse = C.flash_attn_fwd(q, k, v, o_view, scale, causal, None,
                                      dropout, seed, offset)
            ctx.native = True
            ctx.save_for_backward(qkv, out, lse)
            ctx.drop_mask = None
        else:
            dm = None
            if dropout > 0.0:
                g = torch.Generator(device="cpu").manual_seed(
                    (seed * 1000003 + offset) & 0x7FFFFFFFFFFFFFFF)
                dm = (torch.rand(b, h, s, s, generator=g) >= dropout).to(qkv.device)
            o, lse = _sdpa_ref(q.contiguous(), k.contiguous(), v.contiguous(),
                               scale, causal, None, dm, dropout)
            out = o.permute(0, 2, 1, 3).reshape(b, s, h, d)
            ctx.native = False
            ctx.save_for_backward(qkv, out, lse)
            ctx.drop_mask = dm
        ctx.scale, ctx.causal = scale, causal
        ctx.dropout, ctx.seed, ctx.offset = dropout, seed, offset
        return out.reshape(b, s, h * d)

    @staticmethod
    def backward(ctx, do):
        qkv, out, lse = ctx.saved_tensors
        b, s, three, h, d = qkv.shape
        q = qkv[:, :, 0].permute(0, 2, 1, 3)
        k = qkv[:, :, 1].permute(0, 2, 1, 3)
        v = qkv[:, :, 2].permute(0, 2, 1, 3)
        o_view = out.permute(0, 2, 1, 3)
        do_view = do.reshape(b, s, h, d).permute(0, 2, 1, 3)
        if ctx.native:
            C = _ext.get_ext()
            if do_view.stride(-1) != 1:
                do_view = do.contiguous().reshape(b, s, h, d).permute(0, 2, 1, 3)
            dqkv = torch.empty_like(qkv)
            dq = dqkv[:, :, 0].permute(0, 2, 1, 3)
            dk = dqkv[:, :, 1].permute(0, 2, 1, 3)
            dv = dqkv[:, :, 2].permute(0, 2, 1, 3)
            C.flash_attn_bwd(do_view, q, k, v, o_view, lse, dq, dk, dv,
                             ctx.scale, ctx.causal, None, ctx.dropout,
                             ctx.seed, ctx.offset)
        else:
            dq, dk, dv = _sdpa_ref_bwd(do_view.contiguous(), q.contiguous(),
                                       k.contiguous(), v.contiguous(), lse,
                                       ctx.scale, ctx.causal, None,
                                       ctx.drop_mask, ctx.dropout)
            # each [b,h,s,d]; stack -> [3,b,h,s,d]; permute -> [b,s,3,h,d]
            dqkv = torch.stack([dq, dk, dv], dim=0).permute(1, 3, 0, 2, 4).contiguous()
        return dqkv, None, None, None, None, None


def qkv_flash_attention(qkv, scale=None, causal=True, dropout=0.0,
                        training=True):
    """qkv: [B, S, 3, H, D] packed (straight out of the fused QKV GEMM)."""
    if scale is None:
        scale = 1.0 / math.sqrt(qkv.shape[-1])
    if _tracing():
        b, s, three, h, d = qkv.shape
        q = qkv[:, :, 0].permute(0, 2, 1, 3)
        k = qkv[:, :, 1].permute(0, 2, 1, 3)
        v = qkv[:, :, 2].permute(0, 2, 1, 3)
        o = torch.nn.functional.scaled_dot_product_attention(
            q, k, v, is_causal=causal, scale=scale)
        return o.permute(0, 2, 1, 3).reshape(b, s, h * d)
    p = dropout if training else 0.0
    seed, offset = _fa_seed_offset() if p > 0 else (0, 0)
    return _QKVFlashAttn.apply(qkv, scale, causal, p, seed, offset)


def _sdpa_ref(q, k, v, scale, causal, attn_mask=None, drop_mask=None,
              dropout=0.0):
    # fp32 reference; returns (o, lse) with lse = logsumexp of scaled scores.
    # attn_mask: additive [B|1, H|1, Sq, Skv]; drop_mask: uint8 keep mask
    # (same shape as scores) with 1/(1-p) rescale -- matches the HIP
    # kernel's semantics (normalizer from undropped P).
    qf, kf, vf = q.float(), k.float(), v.float()
    if k.shape[1] != q.shape[1]:
        rep = q.shape[1] // k.shape[1]
        kf = kf.repeat_interleave(rep, dim=1)
        vf = vf.repeat_interleave(rep, dim=1)
    s = torch.matmul(qf, kf.transpose(-1, -2)) * scale
    if attn_mask is not None:
        s = s + attn_mask.float()
    if causal:
        sq, skv = s.shape[-2], s.shape[-1]
        mask = torch.ones(sq, skv, dtype=torch.bool, device=s.device).tril()
        s = s.masked_fill(~mask, float("-inf"))
    lse = torch.logsumexp(s, -1)
    p = torch.exp(s - lse.unsqueeze(-1))
    if drop_mask is not None:
        p = p * drop_mask.float() / (1.0 - dropout)
    o = torch.matmul(p, vf)
    return o.to(q.dtype), lse


def _sdpa_ref_bwd(do, q, k, v, lse, scale, causal, attn_mask=None,
                  drop_mask=None, dropout=0.0):
    qf, kf, vf, dof = q.float(), k.float(), v.float(), do.float()
    rep = 1
    if k.shape[1] != q.shape[1]:
        rep = q.shape[1] // k.shape[1]
        kf = kf.repeat_interleave(rep, dim=1)
        vf = vf.repeat_interleave(rep, dim=1)
    s = torch.matmul(qf, kf.transpose(-1, -2)) * scale
    if attn_mask is not None:
        s = s + attn_mask.float()
    if causal:
        sq, skv = s.shape[-2], s.shape[-1]
        mask = torch.ones(sq, skv, dtype=torch.bool, device=s.device).tril()
        s = s.masked_fill(~mask, float("-inf"))
    p = torch.exp(s - lse.unsqueeze(-1).float())
    if drop_mask is not None:
        keep = drop_mask.float() / (1.0 - dropout)
        pd = p * keep
        dv = torch.matmul(pd.transpose(-1, -2), dof)
        dp_raw = torch.matmul(dof, vf.transpose(-1, -2))
        dp = dp_raw * keep
        # delta = rowsum(dO*O) = rowsum(dPd * Pd)
        delta = (dp_raw * pd).sum(-1, keepdim=True)
        ds = p * (dp - delta) * scale
        dq = torch.matmul(ds, kf)
        dk = torch.matmul(ds.transpose(-1, -2), qf)
        if rep > 1:
            b, h, skv_, d = dk.shape
            dk = dk.view(b, h // rep, rep, skv_, d).sum(2)
            dv = dv.view(b, h // rep, rep, skv_, d).sum(2)
        return dq.to(q.dtype), dk.to(k.dtype), dv.to(v.dtype)
    dv = torch.matmul(p.transpose(-1, -2), dof)
    dp = torch.matmul(dof, vf.transpose(-1, -2))
    delta = (dp * p).sum(-1, keepdim=True)
    ds = p * (dp - delta) * scale
    dq = torch.matmul(ds, kf)
    dk = torch.matmul(ds.transpose(-1, -2), qf)
    if rep > 1:
        b, h, skv, d = dk.shape
        dk = dk.view(b, h // rep, rep, skv, d).sum(2)
        dv = dv.view(b, h // rep, rep, skv, d).sum(2)
    return dq.to(q.dtype), dk.to(k.dtype), dv.to(v.dtype)


def flash_attention(q, k, v, dropout=0.0, causal=False, scale=None,
                    return_softmax_lse=False, layout="bshd", attn_mask=None,
                    training=True, fixed_seed_offset=None, rng_name="",
                    name=None):
    """paddle.nn.functional.flash_attention parity
    (python/paddle/nn/functional/flash_attention.py:195; mask + Philox
    dropout per flash_attn_kernel.cu:41 / dropout_impl.cu.h:129).

    layout "bshd": q [batch, seq, heads, head_dim] (paddle convention);
    internally computed as [b, h, s, d].
    """
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    p = dropout if training else 0.0
    if _tracing():
        qt, kt, vt = (t.transpose(1, 2) for t in (q, k, v)) \
            if layout == "bshd" else (q, k, v)
        o = torch.nn.functional.scaled_dot_product_attention(
            qt, kt, vt, attn_mask=attn_mask, is_causal=causal and attn_mask is None,
            scale=scale, enable_gqa=kt.shape[1] != qt.shape[1])
        o = o.transpose(1, 2) if layout == "bshd" else o
        return o, None
    seed, offset = _fa_seed_offset(fixed_seed_offset) if p > 0 else (0, 0)
    if layout == "bshd":
        # strided views -- the kernel is stride-aware, no copies
        qt = q.transpose(1, 2)
        kt = k.transpose(1, 2)
        vt = v.transpose(1, 2)
    else:
        qt, kt, vt = q, k, v
    o, lse = _FlashAttn.apply(qt, kt, vt, scale, causal, attn_mask, p,
                              seed, offset)
    if layout == "bshd":
        o = o.transpose(1, 2)
    if return_softmax_lse:
        return o, lse  # [B, H, Sq] fp32 logsumexp of scaled scores
    return o, None


def scaled_dot_product_attention(query, key, value, attn_mask=None,
                                 dropout_p=0.0, is_causal=False, training=True):
    """paddle.nn.functional.scaled_dot_product_attention parity (bshd);
    boolean masks become additive -inf masks (kernel takes additive bf16)."""
    if attn_mask is not None and attn_mask.dtype == torch.bool:
        attn_mask = torch.zeros_like(attn_mask, dtype=query.dtype).masked_fill(
            ~attn_mask, float("-inf"))
    if attn_mask is not None and attn_mask.dim() == 3:
        attn_mask = attn_mask.unsqueeze(1)
    if attn_mask is not None and attn_mask.dim() == 2:
        attn_mask = attn_mask.unsqueeze(0).unsqueeze(0)
    out, _ = flash_attention(query, key, value, dropout=dropout_p,
                             causal=is_causal, attn_mask=attn_mask,
                             training=training)
    return out


# ---------------------------------------------------------------------------
# dropout + residual add (fused_dropout_add_kernel.cu parity)
# ---------------------------------------------------------------------------
class _DropoutAdd(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, residual, p, training):
        ctx.p = p if training else 0.0
        if _ext.use_native(x):
            C = _ext.get_ext()
            seed = int(torch.cuda.default_generators[x.device.index].initial_seed()) & 0x7FFFFFFF \
                if x.is_cuda else 0
            import random
            offset = random.getrandbits(31)
            outs = C.dropout_add_fwd(x.contiguous(),
                                     residual.contiguous() if residual is not None else None,
                                     ctx.p, seed, offset)
            if ctx.p > 0:
                y, mask = outs
                ctx.save_for_backward(mask)
            else:
                y = outs[0]
                ctx.save_for_backward()
        else:
            if ctx.p > 0:
                mask = (torch.rand_like(x, dtype=torch.float32) >= ctx.p)
                y = x * mask.to(x.dtype) / (1 - ctx.p)
                ctx.save_for_backward(mask)
            else:
                y = x.clone()
                ctx.save_for_backward()
            if residual is not None:
                y = y + residual
        ctx.has_res = residual is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        p = ctx.p
        dres = dy if ctx.has_res else None
        if p == 0:
            return dy, dres, None, None
        (mask,) = ctx.saved_tensors
        if _ext.use_native(dy):
            C = _ext.get_ext()
            dx = C.dropout_add_bwd(dy.contiguous(), mask, p)
        else:
            dx = dy * mask.to(dy.dtype) / (1 - p)
        return dx, dres, None, None


def dropout_add(x, residual=None, p=0.0, training=True):
    if _tracing():
        return x + residual if residual is not None else x
    return _DropoutAdd.apply(x, residual, p, training)


# ---------------------------------------------------------------------------
# embedding (embedding_grad_kernel.cu parity; atomic scatter-add grad)
# ---------------------------------------------------------------------------
class _Embedding(torch.autograd.Function):
    @staticmethod
    def forward(ctx, ids, table, padding_idx):
        ctx.save_for_backward(ids)
        ctx.vocab = table.shape[0]
        ctx.padding_idx = padding_idx
        ctx.table_dtype = table.dtype
        if _ext.use_native(table):
            C = _ext.get_ext()
            return C.embedding_fwd(table.contiguous(), ids.contiguous().long(),
                                   padding_idx if padding_idx is not None else -1)
        return torch.nn.functional.embedding(ids.long(), table, padding_idx=padding_idx)

    @staticmethod
    def backward(ctx, dout):
        (ids,) = ctx.saved_tensors
        if _ext.use_native(dout):
            C = _ext.get_ext()
            dtable = C.embedding_bwd(dout.contiguous(), ids.contiguous().long(), ctx.vocab,
                                     ctx.padding_idx if ctx.padding_idx is not None else -1)
            dtable = dtable.to(ctx.table_dtype)
        else:
            d = dout.shape[-1]
            dtable = torch.zeros(ctx.vocab, d, dtype=torch.float32, device=dout.device)
            flat_ids = ids.reshape(-1).long()
            ok = torch.ones_like(flat_ids, dtype=torch.bool)
            if ctx.padding_idx is not None:
                ok = flat_ids != ctx.padding_idx
            dtable.index_add_(0, flat_ids[ok], dout.reshape(-1, d)[ok].float())
            dtable = dtable.to(ctx.table_dtype)
        return None, dtable, None


def embedding(ids, table, padding_idx=None):
    if _tracing():
        return torch.nn.functional.embedding(ids, table, padding_idx)
    return _Embedding.apply(ids, table, padding_idx)


# ---------------------------------------------------------------------------
# optimizer primitives
# ---------------------------------------------------------------------------
def fused_adamw_step(master, param_out, grad, m, v, lr, beta1, beta2, eps,
                     weight_decay, step, grad_scale=1.0):
    """In-place fused AdamW on a flat fp32 master shard.  param_out may be
    a bf16 view of the model weights (written by the kernel) or None.
    grad may be bf16 (read direct from the flat reduce buffer) or fp32;
    grad_scale folds grad-clip and the 1/world reduce divide into the
    kernel so no separate mul/cast pass touches HBM.
    adamw_ parity: paddle/phi/kernels/gpu/adamw_kernel.cu."""
    if master.is_cuda and _ext.use_native(master):
        C = _ext.get_ext()
        C.adamw(master, param_out, grad.contiguous(), m, v, lr, beta1, beta2,
                eps, weight_decay, beta1 ** step, beta2 ** step, grad_scale)
        return
    # torch reference (CPU tests)
    g = grad.float()
    if grad_scale != 1.0:
        g = g * grad_scale
    master.mul_(1 - lr * weight_decay)
    m.mul_(beta1).add_(g, alpha=1 - beta1)
    v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
    mhat = m / (1 - beta1 ** step)
    vhat = v / (1 - beta2 ** step)
    master.addcdiv_(mhat, vhat.sqrt().add_(eps), value=-lr)
    if param_out is not None:
        param_out.copy_(master.to(param_out.dtype))


def l2_norm_squared(x):
    if x.is_cuda and _ext.use_native(x):
        C = _ext.get_ext()
        return C.l2norm_sq(x.contiguous())
    return x.float().square().sum().reshape(1)


# ---------------------------------------------------------------------------
# paged-KV decode attention (serving; block_multihead_attention parity)
# ---------------------------------------------------------------------------
def paged_decode_attention(q, k_cache, v_cache, block_table, seq_lens, scale=None):
    """One decode step. q: [B, H, D]; k/v_cache: [nblocks, block_size, HKV, D];
    block_table: int32 [B, max_blocks]; seq_lens: int32 [B]."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if _ext.use_native(q) and q.dtype == torch.bfloat16 and q.shape[-1] in (64, 128):
        C = _ext.get_ext()
        return C.decode_attention(q.contiguous(), k_cache.contiguous(),
                                  v_cache.contiguous(), block_table.int().contiguous(),
                                  seq_lens.int().contiguous(), scale)
    # reference: gather each sequence's KV then plain attention
    B, H, D = q.shape
    HKV = k_cache.shape[2]
    bs = k_cache.shape[1]
    rep = H // HKV
    out = torch.empty_like(q)
    for b in range(B):
        S = int(seq_lens[b])
        nb = (S + bs - 1) // bs
        blocks = block_table[b, :nb].long()
        k = k_cache[blocks].reshape(-1, HKV, D)[:S]  # [S, HKV, D]
        v = v_cache[blocks].reshape(-1, HKV, D)[:S]
        kf = k.float().repeat_interleave(rep, dim=1)   # [S, H, D]
        vf = v.float().repeat_interleave(rep, dim=1)
        s = torch.einsum("hd,shd->hs", q[b].float(), kf) * scale
        p = torch.softmax(s, dim=-1)
        out[b] = torch.einsum("hs,shd->hd", p, vf).to(q.dtype)
    return out


# ---------------------------------------------------------------------------
# fused FFN: fc1(+bias+GELU in the hipBLASLt epilogue) -> fc2.
# Backward folds dGELU + fc1 bias-grad into fc2's dgrad GEMM (DGELU_BGRAD),
# so the [tokens, 4h] activation is never touched by a separate elementwise
# pass.  GELU is the tanh approximation (Tensile epilogue).
# Parity: paddle incubate FusedFeedForward / fused_gemm_epilogue
# (paddle/phi/kernels/fusion/gpu/fused_gemm_epilogue_kernel.cu).
# ---------------------------------------------------------------------------
class _FusedFFN(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w1, b1, w2, b2):
        C = _ext.get_ext()
        xs = x.shape
        x2 = x.reshape(-1, xs[-1]).contiguous()
        g, z = C.fc1_gelu_fwd(x2, w1, b1)
        y = torch.addmm(b2, g, w2)
        ctx.save_for_backward(x2, w1, w2, z, g)
        ctx.xshape = xs
        return y.reshape(*xs[:-1], w2.shape[1])

    @staticmethod
    def backward(ctx, dy):
        C = _ext.get_ext()
        x2, w1, w2, z, g = ctx.saved_tensors
        dy2 = dy.reshape(-1, dy.shape[-1]).contiguous()
        dz1, db1 = C.fc2_dgrad_dgelu(dy2, w2, z)   # dGELU + bias-grad fused
        dw2 = g.t() @ dy2
        db2 = dy2.sum(0)
        dx = dz1 @ w1.t()
        dw1 = x2.t() @ dz1
        return dx.reshape(ctx.xshape), dw1, db1, dw2, db2


def fused_ffn(x, w1, b1, w2, b2):
    """y = gelu(x @ w1 + b1) @ w2 + b2 with epilogue-fused bias/GELU.
    Paddle Linear layout: w1 [K, 4K], w2 [4K, K]."""
    return _FusedFFN.apply(x, w1, b1, w2, b2)


# ---------------------------------------------------------------------------
# own-MFMA fused linear / FFN (gemm.hip 8-phase kernel + epilogues).
# fwd runs NT with a per-step cached W^T; dgrad is NT directly because
# paddle's W[in,out] IS the NT B-operand; wgrad picks own-TN vs hipBLASLt
# from the autotune table.  Reference: fused_gemm_epilogue_kernel.cu +
# matmul_kernel_impl.h:914 dispatch pattern.
# ---------------------------------------------------------------------------
from . import gemm_dispatch as _gd  # noqa: E402


def _wgrad(x2, dy2):
    if _gd.use_own("tn", x2.shape[1], dy2.shape[1], x2.shape[0]):
        return _gd.gemm_tn(x2, dy2)
    return torch.matmul(x2.t(), dy2)


def _colsum(dy2, dtype):
    C = _ext.get_ext()
    return C.colsum(dy2).to(dtype)


class _FusedLinearOwn(torch.autograd.Function):
    """y = x @ W (+bias) on the own NT kernel (W paddle-layout [K, N])."""

    @staticmethod
    def forward(ctx, x, w, bias):
        xs = x.shape
        x2 = x.reshape(-1, xs[-1]).contiguous()
        wt = _gd.weight_t(w)
        if bias is not None:
            y2 = _gd.gemm_nt(x2, wt, epilogue=1, bias=bias.contiguous())
        else:
            y2 = _gd.gemm_nt(x2, wt)
        ctx.save_for_backward(x2, w)
        ctx.xshape, ctx.has_bias = xs, bias is not None
        return y2.reshape(*xs[:-1], w.shape[1])

    @staticmethod
    def backward(ctx, dy):
        x2, w = ctx.saved_tensors
        dy2 = dy.reshape(-1, dy.shape[-1]).contiguous()
        dx = _gd.gemm_nt(dy2, w) if _gd.use_own("nt", dy2.shape[0], w.shape[0], w.shape[1]) \
            else torch.matmul(dy2, w.t())
        dw = _wgrad(x2, dy2)
        db = _colsum(dy2, w.dtype) if ctx.has_bias else None
        return dx.reshape(ctx.xshape), dw, db


class _FusedFFNOwn(torch.autograd.Function):
    """FFN pair on the own NT kernel: fc1 carries bias+GELU in the GEMM
    epilogue (pre-activation saved as aux), fc2's dgrad carries dGELU in
    its epilogue -- the [tokens, 4h] tensor never sees a separate
    elementwise pass in either direction."""

    @staticmethod
    def forward(ctx, x, w1, b1, w2, b2):
        xs = x.shape
        x2 = x.reshape(-1, xs[-1]).contiguous()
        g, z = _gd.gemm_nt(x2, _gd.weight_t(w1), epilogue=2, bias=b1.contiguous())
        if b2 is not None:
            y = _gd.gemm_nt(g, _gd.weight_t(w2), epilogue=1, bias=b2.contiguous())
        else:
            y = _gd.gemm_nt(g, _gd.weight_t(w2))
        ctx.save_for_backward(x2, w1, w2, z, g)
        ctx.xshape, ctx.has_b2 = xs, b2 is not None
        return y.reshape(*xs[:-1], w2.shape[1])

    @staticmethod
    def backward(ctx, dy):
        x2, w1, w2, z, g = ctx.saved_tensors
        dy2 = dy.reshape(-1, dy.shape[-1]).contiguous()
        # dz = (dy @ W2^T) * gelu'(z): DGELU epilogue on fc2's dgrad
        dz = _gd.gemm_nt(dy2, w2, epilogue=3, aux=z)
        dw2 = _wgrad(g, dy2)
        db2 = _colsum(dy2, w2.dtype) if ctx.has_b2 else None
        dx = _gd.gemm_nt(dz, w1) if _gd.use_own("nt", dz.shape[0], w1.shape[0], w1.shape[1]) \
            else torch.matmul(dz, w1.t())
        dw1 = _wgrad(x2, dz)
        db1 = _colsum(dz, w1.dtype)
        return dx.reshape(ctx.xshape), dw1, db1, dw2, db2


def _own_linear_ok(x, w, layout="nt"):
    if not (x.is_cuda and x.dtype == torch.bfloat16 and w.dtype == torch.bfloat16
            and _ext.use_native(x)):
        return False
    m = x.numel() // x.shape[-1]
    return _gd.use_own(layout, m, w.shape[1], w.shape[0])


def fused_linear_own(x, w, bias=None):
    return _FusedLinearOwn.apply(x, w, bias)


def fused_ffn_own(x, w1, b1, w2, b2):
    return _FusedFFNOwn.apply(x, w1, b1, w2, b2)


_lt_ffn_ok = None


def _fused_ffn_available(x):
    global _lt_ffn_ok
    if not (x.is_cuda and x.dtype == torch.bfloat16 and _ext.use_native(x)):
        return False
    if _lt_ffn_ok is None:
        C = _ext.get_ext()
        if not hasattr(C, "lt_epilogue_probe"):
            _lt_ffn_ok = False
        else:
            # GELU_AUX_BIAS=164 fwd, DGELU_BGRAD=208 bwd -- both must have
            # Tensile kernels in this hipBLASLt build
            _lt_ffn_ok = (C.lt_epilogue_probe(1024, 1024, 1024, 164) > 0 and
                          C.lt_epilogue_probe(1024, 1024, 1024, 208) > 0)
    return _lt_ffn_ok


def fused_linear_param_grad_add(x, dy, dweight=None, dbias=None,
                                multi_precision=False, has_bias=True):
    """dW += X^T dY (and db += colsum(dY)) in single accumulating GEMM /
    reduction launches -- no intermediate dW allocation.
    Parity: paddle/phi/kernels/fusion/gpu/fused_linear_param_grad_add_kernel.cu
    (used by the sequence-parallel backward overlap)."""
    x2 = x.reshape(-1, x.shape[-1])
    dy2 = dy.reshape(-1, dy.shape[-1])
    if dweight is None:
        dweight = torch.zeros(x2.shape[1], dy2.shape[1], dtype=x.dtype,
                              device=x.device)
    dweight.addmm_(x2.t(), dy2)   # beta=1 accumulate inside the GEMM epilogue
    if has_bias:
        if dbias is None:
            dbias = torch.zeros(dy2.shape[1], dtype=dy.dtype, device=dy.device)
        if dy.is_cuda and _ext.use_native(dy):
            C = _ext.get_ext()
            dbias.add_(C.colsum(dy2.contiguous()).to(dbias.dtype))
        else:
            dbias.add_(dy2.sum(0).to(dbias.dtype))
        return dweight, dbias
    return dweight, None


class _FlashAttnVarlen(torch.autograd.Function):
    """Single-launch ragged attention over packed [total, H, D] tensors
    (reference flash_attn_unpadded / flash_attn_kernel.cu:41 varlen via
    cu_seqlens; ours maps each 128-row block to its sequence in-kernel)."""

    @staticmethod
    def forward(ctx, q, k, v, cu_q, cu_k, scale, causal, dropout, seed, offset):
        C = _ext.get_ext()
        o, lse = C.flash_attn_varlen_fwd(q, k, v, cu_q, cu_k, scale, causal,
                                         dropout, seed, offset)
        ctx.save_for_backward(q, k, v, o, lse, cu_q, cu_k)
        ctx.scale, ctx.causal = scale, causal
        ctx.dropout, ctx.seed, ctx.offset = dropout, seed, offset
        ctx.mark_non_differentiable(lse)
        return o, lse

    @staticmethod
    def backward(ctx, do, _dlse):
        q, k, v, o, lse, cu_q, cu_k = ctx.saved_tensors
        C = _ext.get_ext()
        dq, dk, dv = C.flash_attn_varlen_bwd(do, q, k, v, o, lse, cu_q, cu_k,
                                             ctx.scale, ctx.causal, ctx.dropout,
                                             ctx.seed, ctx.offset)
        return dq, dk, dv, None, None, None, None, None, None, None


def flash_attn_varlen_func(q, k, v, cu_seqlens_q, cu_seqlens_k, max_seqlen_q,
                           max_seqlen_k, scale=None, causal=False,
                           dropout=0.0, training=True, return_softmax_lse=False):
    """Varlen (ragged) flash attention over packed [total_tokens, H, D]
    inputs with cu_seqlens boundaries (reference: flash_attn_unpadded,
    python/paddle/nn/functional/flash_attention.py:593).

    Native path: ONE kernel launch for the whole mixed-length batch
    (per-block sequence bounds resolved in-kernel); CPU fallback groups
    by shape."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    p = dropout if training else 0.0
    if (_ext.use_native(q) and q.dtype == torch.bfloat16
            and q.shape[-1] in (64, 128) and k.shape[1] == q.shape[1]):
        cu_q = cu_seqlens_q.to(torch.int32)
        cu_k = cu_seqlens_k.to(torch.int32)
        seed, offset = _fa_seed_offset() if p > 0 else (0, 0)
        out, lse = _FlashAttnVarlen.apply(q, k, v, cu_q, cu_k, scale, causal,
                                          p, seed, offset)
        return (out, lse) if return_softmax_lse else out
    import collections
    nq = cu_seqlens_q.tolist()
    nk = cu_seqlens_k.tolist()
    out = torch.empty_like(q)
    groups = collections.defaultdict(list)
    for i in range(len(nq) - 1):
        groups[(nq[i + 1] - nq[i], nk[i + 1] - nk[i])].append(i)
    for (lq, lk), idxs in groups.items():
        qg = torch.stack([q[nq[i]:nq[i] + lq] for i in idxs]).transpose(1, 2)
        kg = torch.stack([k[nk[i]:nk[i] + lk] for i in idxs]).transpose(1, 2)
        vg = torch.stack([v[nk[i]:nk[i] + lk] for i in idxs]).transpose(1, 2)
        og, _ = flash_attention(qg.transpose(1, 2), kg.transpose(1, 2),
                                vg.transpose(1, 2), causal=causal, scale=scale)
        for j, i in enumerate(idxs):
            out[nq[i]:nq[i] + lq] = og[j]
    if return_softmax_lse:
        return out, None
    return out
