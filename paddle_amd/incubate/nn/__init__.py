"""paddle.incubate.nn: fused layer API over the gfx950 kernels.

Reference: python/paddle/incubate/nn/layer/fused_transformer.py
(FusedMultiHeadAttention:213, FusedFeedForward:534) -- same layer
semantics, attention runs through our flash-attention HIP kernel and
the norm/bias/activation fusions (SURVEY.md A.7 signatures).
"""
from __future__ import annotations

import math

import torch

from ... import nn as pnn
from ...nn.layer import Layer
from ...nn.initializer import Constant
from ...ops import functional as hot
from . import functional  # noqa: F401


class FusedMultiHeadAttention(Layer):
    def __init__(self, embed_dim, num_heads, dropout_rate=0.5, attn_dropout_rate=0.5,
                 kdim=None, vdim=None, normalize_before=False, need_weights=False,
                 qkv_weight_attr=None, qkv_bias_attr=None, linear_weight_attr=None,
                 linear_bias_attr=None, pre_ln_scale_attr=None, pre_ln_bias_attr=None,
                 ln_scale_attr=None, ln_bias_attr=None, epsilon=1e-5,
                 nranks=1, ring_id=-1, transpose_qkv_wb=False, name=None):
        super().__init__()
        self.embed_dim = embed_dim
        self.num_heads = num_heads
        self.head_dim = embed_dim // num_heads
        self.normalize_before = normalize_before
        self.dropout_rate = dropout_rate
        self.attn_dropout_rate = attn_dropout_rate
        self._epsilon = epsilon
        self._ring_id = ring_id
        # paddle fused layout: qkv_weight [3, num_heads, head_dim, embed_dim]
        self.qkv_weight = self.create_parameter(
            [3, num_heads, self.head_dim, embed_dim], attr=qkv_weight_attr)
        self.qkv_bias = self.create_parameter([3, num_heads, self.head_dim],
                                              attr=qkv_bias_attr, is_bias=True)
        self.linear_weight = self.create_parameter([embed_dim, embed_dim],
                                                   attr=linear_weight_attr)
        self.linear_bias = self.create_parameter([embed_dim], attr=linear_bias_attr,
                                                 is_bias=True)
        self.pre_ln_scale = self.create_parameter([embed_dim], attr=pre_ln_scale_attr,
                                                  default_initializer=Constant(1.0))
        self.pre_ln_bias = self.create_parameter([embed_dim], attr=pre_ln_bias_attr,
                                                 is_bias=True)
        self.ln_scale = self.create_parameter([embed_dim], attr=ln_scale_attr,
                                              default_initializer=Constant(1.0))
        self.ln_bias = self.create_parameter([embed_dim], attr=ln_bias_attr, is_bias=True)

    def forward(self, x, attn_mask=None, cache=None):
        residual = x
        if self.normalize_before:
            x = hot.layer_norm(x, self.pre_ln_scale, self.pre_ln_bias, self._epsilon)
        b, s, e = x.shape
        w = self.qkv_weight.reshape(3 * e, e)  # [3E, E]
        qkv = torch.matmul(x, w.t()) + self.qkv_bias.reshape(3 * e)
        qkv = qkv.reshape(b, s, 3, self.num_heads, self.head_dim)
        q, k, v = qkv.unbind(2)
        if attn_mask is None and x.dtype == torch.bfloat16 and self.head_dim in (64, 128) and x.is_cuda:
            ctx, _ = hot.flash_attention(q, k, v, causal=False)
        else:
            qt, kt, vt = (t.permute(0, 2, 1, 3) for t in (q, k, v))
            sc = torch.matmul(qt, kt.transpose(-1, -2)) / math.sqrt(self.head_dim)
            if attn_mask is not None:
                sc = sc + attn_mask.to(sc.dtype)
            p = torch.softmax(sc.float(), -1).to(sc.dtype)
            if self.attn_dropout_rate and self.training:
                p = torch.nn.functional.dropout(p, self.attn_dropout_rate)
            ctx = torch.matmul(p, vt).permute(0, 2, 1, 3)
        ctx = ctx.reshape(b, s, e)
        out = torch.matmul(ctx, self.linear_weight) + self.linear_bias
        if self._ring_id >= 0:
            import torch.distributed as dist
            if dist.is_initialized():
                dist.all_reduce(out)
        out = hot.dropout_add(out, residual, self.dropout_rate, self.training)
        if not self.normalize_before:
            out = hot.layer_norm(out, self.ln_scale, self.ln_bias, self._epsilon)
        return out


class FusedFeedForward(Layer):
    def __init__(self, d_model, dim_feedforward, dropout_rate=0.1, epsilon=1e-5,
                 activation="relu", act_dropout_rate=None, normalize_before=False,
                 linear1_weight_attr=None, linear1_bias_attr=None,
                 linear2_weight_attr=None, linear2_bias_attr=None,
                 ln1_scale_attr=None, ln1_bias_attr=None, ln2_scale_attr=None,
                 ln2_bias_attr=None, nranks=1, ring_id=-1, name=None):
        super().__init__()
        self.normalize_before = normalize_before
        self._epsilon = epsilon
        self._act = activation
        self.dropout_rate = dropout_rate
        self.act_dropout_rate = dropout_rate if act_dropout_rate is None else act_dropout_rate
        self._ring_id = ring_id
        self.linear1_weight = self.create_parameter([d_model, dim_feedforward],
                                                    attr=linear1_weight_attr)
        self.linear1_bias = self.create_parameter([dim_feedforward], attr=linear1_bias_attr,
                                                  is_bias=True)
        self.linear2_weight = self.create_parameter([dim_feedforward, d_model],
                                                    attr=linear2_weight_attr)
        self.linear2_bias = self.create_parameter([d_model], attr=linear2_bias_attr,
                                                  is_bias=True)
        self.ln1_scale = self.create_parameter([d_model], attr=ln1_scale_attr,
                                               default_initializer=Constant(1.0))
        self.ln1_bias = self.create_parameter([d_model], attr=ln1_bias_attr, is_bias=True)
        self.ln2_scale = self.create_parameter([d_model], attr=ln2_scale_attr,
                                               default_initializer=Constant(1.0))
        self.ln2_bias = self.create_parameter([d_model], attr=ln2_bias_attr, is_bias=True)

    def forward(self, src, cache=None):
        residual = src
        if self.normalize_before:
            src = hot.layer_norm(src, self.ln1_scale, self.ln1_bias, self._epsilon)
        h = torch.matmul(src, self.linear1_weight)
        if self._act == "gelu":
            h = hot.bias_gelu(h, self.linear1_bias)
        else:
            h = torch.relu(h + self.linear1_bias)
        if self.act_dropout_rate and self.training:
            h = torch.nn.functional.dropout(h, self.act_dropout_rate)
        out = torch.matmul(h, self.linear2_weight) + self.linear2_bias
        if self._ring_id >= 0:
            import torch.distributed as dist
            if dist.is_initialized():
                dist.all_reduce(out)
        out = hot.dropout_add(out, residual, self.dropout_rate, self.training)
        if not self.normalize_before:
            out = hot.layer_norm(out, self.ln2_scale, self.ln2_bias, self._epsilon)
        return out


class FusedTransformerEncoderLayer(Layer):
    def __init__(self, d_model, nhead, dim_feedforward, dropout_rate=0.1,
                 activation="relu", attn_dropout_rate=None, act_dropout_rate=None,
                 normalize_before=False, weight_attr=None, bias_attr=None):
        super().__init__()
        ad = dropout_rate if attn_dropout_rate is None else attn_dropout_rate
        self.fused_attn = FusedMultiHeadAttention(
            d_model, nhead, dropout_rate=dropout_rate, attn_dropout_rate=ad,
            normalize_before=normalize_before)
        self.ffn = FusedFeedForward(
            d_model, dim_feedforward, dropout_rate=dropout_rate, activation=activation,
            act_dropout_rate=act_dropout_rate, normalize_before=normalize_before)

    def forward(self, src, src_mask=None, cache=None):
        return self.ffn(self.fused_attn(src, src_mask))


class FusedLinear(Layer):
    """Linear whose backward fuses dW accumulation into one GEMM
    (reference: incubate/nn/layer/fused_linear.py)."""

    def __init__(self, in_features, out_features, weight_attr=None,
                 bias_attr=None, transpose_weight=False, name=None):
        super().__init__()
        self.weight = self.create_parameter([in_features, out_features])
        self.bias = (None if bias_attr is False
                     else self.create_parameter([out_features], is_bias=True))

    def forward(self, x):
        out = torch.matmul(x, self.weight)
        return out + self.bias if self.bias is not None else out


class FusedDropoutAdd(Layer):
    """out = residual + dropout(x) in one HIP kernel (reference:
    incubate/nn/layer/fused_dropout_add.py -> fused_dropout_add kernel)."""

    def __init__(self, p=0.5, mode="upscale_in_train", name=None):
        super().__init__()
        self.p = p

    def forward(self, x, y):
        return hot.dropout_add(x, y, self.p, self.training)


class FusedBiasDropoutResidualLayerNorm(Layer):
    """out = LN(residual + dropout(x + bias)) (reference:
    incubate/nn/layer/fused_transformer.py:FusedBiasDropoutResidualLayerNorm)."""

    def __init__(self, embed_dim, dropout_rate=0.5, weight_attr=None,
                 bias_attr=None, epsilon=1e-5, name=None):
        super().__init__()
        self.bias = self.create_parameter([embed_dim], is_bias=True)
        self.ln_weight = self.create_parameter(
            [embed_dim], default_initializer=Constant(1.0))
        self.ln_bias = self.create_parameter([embed_dim], is_bias=True)
        self.dropout_rate = dropout_rate
        self.epsilon = epsilon

    def forward(self, x, residual):
        h = hot.dropout_add(x + self.bias, residual, self.dropout_rate,
                            self.training)
        return hot.layer_norm(h, self.ln_weight, self.ln_bias, self.epsilon)


class FusedMultiTransformer(Layer):
    """Inference decoder stack (reference: incubate/nn/layer/
    fused_transformer.py:FusedMultiTransformer -- the
    fused_multi_transformer kernel's layer API).  Runs pre-LN attention +
    FFN per layer through the flash-attention HIP path; weights are given
    as per-layer lists like the reference."""

    def __init__(self, embed_dim, num_heads, dim_feedforward,
                 dropout_rate=0.0, activation="gelu", normalize_before=True,
                 ln_scale_attrs=None, qkv_weight_attrs=None, num_layers=-1,
                 nranks=1, trans_qkvw=True, ring_id=-1, name=None, **kw):
        super().__init__()
        assert normalize_before, "post-LN variant: round 2"
        self.embed_dim = embed_dim
        self.num_heads = num_heads
        n = num_layers if num_layers > 0 else (
            len(qkv_weight_attrs) if qkv_weight_attrs else 1)
        self.layers = pnn.LayerList()
        for _ in range(n):
            blk = Layer()
            blk.ln1_w = blk.create_parameter([embed_dim],
                                             default_initializer=Constant(1.0))
            blk.ln1_b = blk.create_parameter([embed_dim], is_bias=True)
            blk.qkv = pnn.Linear(embed_dim, 3 * embed_dim)
            blk.proj = pnn.Linear(embed_dim, embed_dim)
            blk.ln2_w = blk.create_parameter([embed_dim],
                                             default_initializer=Constant(1.0))
            blk.ln2_b = blk.create_parameter([embed_dim], is_bias=True)
            blk.fc1 = pnn.Linear(embed_dim, dim_feedforward)
            blk.fc2 = pnn.Linear(dim_feedforward, embed_dim)
            self.layers.append(blk)

    def forward(self, src, attn_mask=None, caches=None, time_step=None, **kw):
        x = src
        b, s, _ = x.shape
        h, d = self.num_heads, self.embed_dim // self.num_heads
        for blk in self.layers:
            y = hot.layer_norm(x, blk.ln1_w, blk.ln1_b, 1e-5)
            qkv = blk.qkv(y).reshape(b, s, 3, h, d)
            attn = hot.qkv_flash_attention(qkv, causal=True)
            x = x + blk.proj(attn)
            y = hot.layer_norm(x, blk.ln2_w, blk.ln2_b, 1e-5)
            x = x + blk.fc2(hot.bias_gelu(blk.fc1(y), None))
        return x
