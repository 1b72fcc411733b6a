"""paddle.incubate.nn.functional parity: fused functional ops.

Reference signatures: SURVEY.md A.7 (fused_rms_norm, fused_rotary_
position_embedding, fused_dropout_add, swiglu, fused_bias_act...).
"""
from __future__ import annotations

import torch

from ....ops import functional as hot
from ....ops.functional import (  # noqa: F401
    fused_rotary_position_embedding,
    swiglu,
)


def fused_rms_norm(x, norm_weight, norm_bias=None, epsilon=1e-6, begin_norm_axis=-1,
                   bias=None, residual=None, quant_scale=-1, **kw):
    out = hot.fused_rms_norm(x, norm_weight, residual=residual, epsilon=epsilon)
    if residual is not None:
        return out  # (y, residual_out)
    return out, None


def fused_layer_norm(x, norm_weight, norm_bias, epsilon=1e-5, begin_norm_axis=-1,
                     bias=None, residual=None, **kw):
    if residual is not None:
        x = x + residual
        res_out = x
        y = hot.layer_norm(x, norm_weight, norm_bias, epsilon)
        return y, res_out
    return hot.layer_norm(x, norm_weight, norm_bias, epsilon), None


def fused_dropout_add(x, y, p=0.5, training=True, mode="upscale_in_train", name=None):
    return hot.dropout_add(x, y, p, training)


def fused_bias_act(x, bias=None, act_method="gelu", **kw):
    if act_method == "gelu":
        return hot.bias_gelu(x, bias)
    if act_method == "swiglu":
        if bias is not None:
            x = x + bias
        return hot.swiglu(x)
    raise ValueError(act_method)


def fused_linear(x, weight, bias=None, transpose_weight=False, name=None):
    w = weight.t() if transpose_weight else weight
    out = torch.matmul(x, w)
    if bias is not None:
        out = out + bias
    return out


def fused_linear_activation(x, y, bias, trans_x=False, trans_y=False, activation="gelu"):
    if trans_x:
        x = x.transpose(-1, -2)
    w = y.t() if trans_y else y
    out = torch.matmul(x, w)
    if activation == "gelu":
        return hot.bias_gelu(out, bias)
    out = out + bias
    if activation == "relu":
        out = torch.relu(out)
    return out


# fused functional forms (reference: incubate/nn/functional/__init__.py)
import torch as _t


def fused_matmul_bias(x, y, bias=None, transpose_x=False, transpose_y=False,
                      name=None):
    """GEMM + bias in one hipBLASLt call (torch.addmm epilogue path)."""
    a = x.t() if transpose_x else x
    b = y.t() if transpose_y else y
    if bias is not None and a.dim() == 2:
        return _t.addmm(bias, a, b)
    out = a @ b
    return out + bias if bias is not None else out


def fused_multi_head_attention(x, qkv_weight, linear_weight, pre_layer_norm=False,
                               pre_ln_scale=None, pre_ln_bias=None, ln_scale=None,
                               ln_bias=None, pre_ln_epsilon=1e-5, qkv_bias=None,
                               linear_bias=None, cache_kv=None, attn_mask=None,
                               dropout_rate=0.0, attn_dropout_rate=0.0,
                               ln_epsilon=1e-5, training=True, mode='upscale_in_train',
                               ring_id=-1, add_residual=True, num_heads=None,
                               transpose_qkv_wb=False, name=None):
    """Functional form of FusedMultiHeadAttention (reference:
    fused_attention_kernel.cu flow: [preLN ->] qkv -> FMHA -> proj ->
    [dropout ->] residual [-> LN])."""
    from ....ops import functional as hot
    residual = x
    h = x
    if pre_layer_norm:
        h = hot.layer_norm(h, pre_ln_scale, pre_ln_bias, pre_ln_epsilon)
    b, s, d = h.shape
    if transpose_qkv_wb:
        qkv = h @ qkv_weight            # [d, 3d]
        if qkv_bias is not None:
            qkv = qkv + qkv_bias
        nh = num_heads
        qkv = qkv.reshape(b, s, 3, nh, d // nh)
    else:
        nh = qkv_weight.shape[1]
        qkv = _t.einsum("bsd,thnd->bsthn" if qkv_weight.dim() == 4 else "bsd,dt->bst",
                        h, qkv_weight)
        if qkv.dim() == 5:
            pass
        if qkv_bias is not None:
            qkv = qkv + qkv_bias
        qkv = qkv.reshape(b, s, 3, nh, d // nh)
    att = hot.qkv_flash_attention(qkv.contiguous(), causal=False)
    att = att.reshape(b, s, d)
    out = att @ linear_weight
    if linear_bias is not None:
        out = out + linear_bias
    if dropout_rate and training:
        out = _t.nn.functional.dropout(out, dropout_rate)
    if add_residual:
        out = out + residual
    if not pre_layer_norm and ln_scale is not None:
        from ....ops import functional as hot2
        out = hot2.layer_norm(out, ln_scale, ln_bias, ln_epsilon)
    return out


def fused_feedforward(x, linear1_weight, linear2_weight, linear1_bias=None,
                      linear2_bias=None, ln1_scale=None, ln1_bias=None,
                      ln2_scale=None, ln2_bias=None, dropout1_rate=0.5,
                      dropout2_rate=0.5, activation="relu", ln1_epsilon=1e-5,
                      ln2_epsilon=1e-5, pre_layer_norm=False, training=True,
                      mode='upscale_in_train', ring_id=-1, add_residual=True,
                      name=None):
    """Functional FusedFeedForward (reference: fused_feedforward_kernel.cu:
    [preLN ->] GEMM1 -> act(+dropout) -> GEMM2 (+dropout) -> residual
    [-> LN])."""
    from ....ops import functional as hot
    residual = x
    h = x
    if pre_layer_norm:
        h = hot.layer_norm(h, ln1_scale, ln1_bias, ln1_epsilon)
    h = h @ linear1_weight
    if activation == "gelu":
        h = hot.bias_gelu(h, linear1_bias)
    else:
        if linear1_bias is not None:
            h = h + linear1_bias
        h = _t.relu(h)
    if dropout1_rate and training:
        h = _t.nn.functional.dropout(h, dropout1_rate)
    h = h @ linear2_weight
    if linear2_bias is not None:
        h = h + linear2_bias
    if dropout2_rate and training:
        h = _t.nn.functional.dropout(h, dropout2_rate)
    out = h + residual if add_residual else h
    if not pre_layer_norm and ln2_scale is not None:
        out = hot.layer_norm(out, ln2_scale, ln2_bias, ln2_epsilon)
    return out


def fused_bias_dropout_residual_layer_norm(x, residual, bias=None,
                                           ln_scale=None, ln_bias=None,
                                           dropout_rate=0.5, ln_epsilon=1e-5,
                                           training=True,
                                           mode='upscale_in_train', name=None):
    from ....ops import functional as hot
    h = x + bias if bias is not None else x
    h = hot.dropout_add(h, residual, dropout_rate, training)
    return hot.layer_norm(h, ln_scale, ln_bias, ln_epsilon)


def fused_multi_transformer(*args, **kwargs):
    raise NotImplementedError(
        "functional fused_multi_transformer: use the "
        "incubate.nn.FusedMultiTransformer layer (paged decode path)")


def fused_moe(x, gate_weight, expert_weights1, expert_weights2, **kwargs):
    raise NotImplementedError(
        "fused_moe (cutlass inference kernel): use models.moe.MoELayer "
        "with grouped experts")


def variable_length_memory_efficient_attention(query, key, value, seq_lens=None,
                                               kv_seq_lens=None, mask=None,
                                               scale=None, causal=False,
                                               pre_cache_length=0):
    from ....ops.functional import flash_attn_varlen_func
    raise NotImplementedError(
        "use paddle.ops.functional.flash_attn_varlen_func (cu_seqlens form)")
