"""paddle.nn.functional parity (python/paddle/nn/functional/).

Hot ops dispatch to the gfx950 HIP kernels via paddle_amd.ops; the rest
map to torch.nn.functional with paddle conventions (weight layouts:
Linear weight is [in, out]).
"""
from __future__ import annotations

import math

import torch
import torch.nn.functional as TF

from ... import framework
from ...ops import functional as hot
from ...ops.functional import (  # noqa: F401
    flash_attention,
    scaled_dot_product_attention,
    fused_rotary_position_embedding,
    swiglu,
)

# re-export flash_attention submodule-style too
sdp_kernel = None


def linear(x, weight, bias=None, name=None):
    # paddle weight layout: [in_features, out_features].  addmm fuses the
    # bias into the hipBLASLt epilogue (saves one HBM pass per call).
    if bias is not None and x.dim() >= 2:
        x2 = x.reshape(-1, x.shape[-1])
        out = torch.addmm(bias, x2, weight)
        return out.reshape(*x.shape[:-1], weight.shape[-1])
    out = torch.matmul(x, weight)
    if bias is not None:
        out = out + bias
    return out


def embedding(x, weight, padding_idx=None, sparse=False, name=None):
    return hot.embedding(x, weight, padding_idx)


def layer_norm(x, normalized_shape, weight=None, bias=None, epsilon=1e-5, name=None):
    if weight is None:
        return TF.layer_norm(x, normalized_shape if isinstance(normalized_shape, (list, tuple))
                             else (normalized_shape,), eps=epsilon)
    return hot.layer_norm(x, weight, bias, epsilon)


def rms_norm(x, weight, epsilon=1e-6):
    return hot.rms_norm(x, weight, epsilon)


def relu(x, name=None):
    return TF.relu(x)


def relu_(x):
    return TF.relu_(x)


def relu6(x, name=None):
    return TF.relu6(x)


def gelu(x, approximate=False, name=None):
    if x.is_cuda and not approximate:
        return hot.bias_gelu(x, None)
    return TF.gelu(x, approximate="tanh" if approximate else "none")


def silu(x, name=None):
    return TF.silu(x)


def sigmoid(x, name=None):
    return torch.sigmoid(x)


def tanh(x, name=None):
    return torch.tanh(x)


def softmax(x, axis=-1, dtype=None, name=None):
    dt = framework.convert_dtype(dtype) if dtype else None
    return TF.softmax(x, dim=axis, dtype=dt)


def log_softmax(x, axis=-1, dtype=None, name=None):
    dt = framework.convert_dtype(dtype) if dtype else None
    return TF.log_softmax(x, dim=axis, dtype=dt)


def leaky_relu(x, negative_slope=0.01, name=None):
    return TF.leaky_relu(x, negative_slope)


def elu(x, alpha=1.0, name=None):
    return TF.elu(x, alpha)


def hardswish(x, name=None):
    return TF.hardswish(x)


def hardsigmoid(x, slope=1 / 6, offset=0.5, name=None):
    return (x * slope + offset).clamp(0, 1)


def mish(x, name=None):
    return TF.mish(x)


def swish(x, name=None):
    return TF.silu(x)


def softplus(x, beta=1, threshold=20, name=None):
    return TF.softplus(x, beta, threshold)


def dropout(x, p=0.5, axis=None, training=True, mode="upscale_in_train", name=None):
    if not training or p == 0:
        return x
    if axis is not None:
        # paddle axis-dropout: mask broadcast along the other axes
        shape = [1] * x.dim()
        axes = [axis] if isinstance(axis, int) else list(axis)
        for a in axes:
            shape[a] = x.shape[a]
        mask = (torch.rand(shape, device=x.device) >= p).to(x.dtype)
        if mode == "upscale_in_train":
            return x * mask / (1 - p)
        return x * mask
    if mode == "upscale_in_train":
        return TF.dropout(x, p, training)
    return x * (torch.rand_like(x, dtype=torch.float32) >= p).to(x.dtype)


def dropout2d(x, p=0.5, training=True, data_format="NCHW", name=None):
    return TF.dropout2d(x, p, training)


def cross_entropy(input, label, weight=None, ignore_index=-100, reduction="mean",
                  soft_label=False, axis=-1, use_softmax=True, label_smoothing=0.0,
                  name=None):
    if soft_label or weight is not None or label_smoothing > 0 or not use_softmax:
        lf = input.float()
        return TF.cross_entropy(lf.reshape(-1, lf.shape[-1]),
                                label.reshape(-1) if not soft_label else label.reshape(-1, lf.shape[-1]),
                                weight=weight, ignore_index=ignore_index,
                                reduction=reduction, label_smoothing=label_smoothing)
    loss = hot.softmax_cross_entropy(input, label, ignore_index, reduction="none")
    if reduction == "mean":
        n_valid = (label != ignore_index).sum().clamp(min=1)
        return loss.sum() / n_valid.to(loss.dtype)
    if reduction == "sum":
        return loss.sum()
    return loss.unsqueeze(-1)  # paddle keeps trailing dim for reduction='none'


def softmax_with_cross_entropy(logits, label, soft_label=False, ignore_index=-100,
                               return_softmax=False, axis=-1):
    loss = hot.softmax_cross_entropy(logits, label.squeeze(-1) if label.dim() == logits.dim() else label,
                                     ignore_index, reduction="none").unsqueeze(-1)
    if return_softmax:
        return loss, TF.softmax(logits.float(), dim=axis).to(logits.dtype)
    return loss


def mse_loss(input, label, reduction="mean", name=None):
    return TF.mse_loss(input, label, reduction=reduction)


def l1_loss(input, label, reduction="mean", name=None):
    return TF.l1_loss(input, label, reduction=reduction)


def nll_loss(input, label, weight=None, ignore_index=-100, reduction="mean", name=None):
    return TF.nll_loss(input, label, weight, ignore_index=ignore_index, reduction=reduction)


def binary_cross_entropy(input, label, weight=None, reduction="mean", name=None):
    return TF.binary_cross_entropy(input, label, weight, reduction=reduction)


def binary_cross_entropy_with_logits(logit, label, weight=None, reduction="mean",
                                     pos_weight=None, name=None):
    return TF.binary_cross_entropy_with_logits(logit, label, weight, reduction=reduction,
                                               pos_weight=pos_weight)


def smooth_l1_loss(input, label, reduction="mean", delta=1.0, name=None):
    return TF.smooth_l1_loss(input, label, reduction=reduction, beta=delta)


def kl_div(input, label, reduction="mean", name=None):
    return TF.kl_div(input, label, reduction=reduction)


# -- conv / pool ------------------------------------------------------------
def conv2d(x, weight, bias=None, stride=1, padding=0, dilation=1, groups=1,
           data_format="NCHW", name=None):
    return TF.conv2d(x, weight, bias, stride, padding, dilation, groups)


def conv1d(x, weight, bias=None, stride=1, padding=0, dilation=1, groups=1,
           data_format="NCL", name=None):
    return TF.conv1d(x, weight, bias, stride, padding, dilation, groups)


def conv2d_transpose(x, weight, bias=None, stride=1, padding=0, output_padding=0,
                     groups=1, dilation=1, data_format="NCHW", output_size=None, name=None):
    return TF.conv_transpose2d(x, weight, bias, stride, padding, output_padding, groups, dilation)


def max_pool2d(x, kernel_size, stride=None, padding=0, return_mask=False,
               ceil_mode=False, data_format="NCHW", name=None):
    out = TF.max_pool2d(x, kernel_size, stride, padding, ceil_mode=ceil_mode,
                        return_indices=return_mask)
    return out


def avg_pool2d(x, kernel_size, stride=None, padding=0, ceil_mode=False,
               exclusive=True, divisor_override=None, data_format="NCHW", name=None):
    return TF.avg_pool2d(x, kernel_size, stride, padding, ceil_mode=ceil_mode,
                         count_include_pad=not exclusive, divisor_override=divisor_override)


def adaptive_avg_pool2d(x, output_size, data_format="NCHW", name=None):
    return TF.adaptive_avg_pool2d(x, output_size)


def batch_norm(x, running_mean, running_var, weight, bias, training=False,
               momentum=0.9, epsilon=1e-5, data_format="NCHW", use_global_stats=None,
               name=None):
    return TF.batch_norm(x, running_mean, running_var, weight, bias,
                         training=training, momentum=1 - momentum, eps=epsilon)


def interpolate(x, size=None, scale_factor=None, mode="nearest", align_corners=False,
                align_mode=0, data_format="NCHW", name=None):
    ac = align_corners if mode in ("linear", "bilinear", "bicubic", "trilinear") else None
    return TF.interpolate(x, size=size, scale_factor=scale_factor, mode=mode,
                          align_corners=ac)


def pad(x, pad, mode="constant", value=0.0, data_format="NCHW", name=None):
    return TF.pad(x, list(pad), mode=mode, value=value)


def unfold(x, kernel_sizes, strides=1, paddings=0, dilations=1, name=None):
    return TF.unfold(x, kernel_sizes, dilations, paddings, strides)


def one_hot(x, num_classes, name=None):
    return TF.one_hot(x.long(), num_classes).float()


def normalize(x, p=2, axis=1, epsilon=1e-12, name=None):
    return TF.normalize(x, p=p, dim=axis, eps=epsilon)


def glu(x, axis=-1, name=None):
    return TF.glu(x, dim=axis)


def label_smooth(label, prior_dist=None, epsilon=0.1, name=None):
    n = label.shape[-1]
    if prior_dist is not None:
        return (1 - epsilon) * label + epsilon * prior_dist
    return (1 - epsilon) * label + epsilon / n
