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
