"""Common layers (reference: python/paddle/nn/layer/common.py, norm.py).

Paddle layout conventions preserved: Linear.weight is [in, out]
(checkpoint-compatible with .pdparams); Embedding.weight [vocab, dim].
"""
from __future__ import annotations

import numbers

import torch

from .. import framework
from ..ops import functional as hot
from . import functional as F
from .initializer import Constant, Normal, XavierNormal, _apply_initializer
from .layer import Layer


class Linear(Layer):
    def __init__(self, in_features, out_features, weight_attr=None, bias_attr=None,
                 name=None):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.weight = self.create_parameter([in_features, out_features], attr=weight_attr,
                                            default_initializer=_attr_init(weight_attr))
        if bias_attr is False:
            self.bias = None
        else:
            self.bias = self.create_parameter([out_features], attr=bias_attr, is_bias=True,
                                              default_initializer=_attr_init(bias_attr, bias=True))

    def forward(self, x):
        return F.linear(x, self.weight, self.bias)

    def extra_repr(self):
        return f"in={self.in_features}, out={self.out_features}"


def _attr_init(attr, bias=False):
    if attr is None or attr is True:
        return None
    init = getattr(attr, "initializer", None)
    return init


class Embedding(Layer):
    def __init__(self, num_embeddings, embedding_dim, padding_idx=None, sparse=False,
                 weight_attr=None, name=None):
        super().__init__()
        self._padding_idx = padding_idx
        self.num_embeddings = num_embeddings
        self.embedding_dim = embedding_dim
        self.weight = self.create_parameter([num_embeddings, embedding_dim], attr=weight_attr,
                                            default_initializer=_attr_init(weight_attr) or XavierNormal())
        if padding_idx is not None:
            with torch.no_grad():
                self.weight[padding_idx].fill_(0)

    def forward(self, x):
        return F.embedding(x, self.weight, self._padding_idx)


class LayerNorm(Layer):
    def __init__(self, normalized_shape, epsilon=1e-5, weight_attr=None, bias_attr=None,
                 name=None):
        super().__init__()
        if isinstance(normalized_shape, numbers.Integral):
            normalized_shape = [normalized_shape]
        self._normalized_shape = list(normalized_shape)
        self._epsilon = epsilon
        n = 1
        for s in normalized_shape:
            n *= s
        if weight_attr is False:
            self.weight = None
        else:
            self.weight = self.create_parameter(self._normalized_shape, attr=weight_attr,
                                                default_initializer=_attr_init(weight_attr) or Constant(1.0))
        if bias_attr is False:
            self.bias = None
        else:
            self.bias = self.create_parameter(self._normalized_shape, attr=bias_attr, is_bias=True)

    def forward(self, x):
        if self.weight is not None and len(self._normalized_shape) == 1:
            return hot.layer_norm(x, self.weight, self.bias, self._epsilon)
        return torch.nn.functional.layer_norm(x, self._normalized_shape, self.weight,
                                              self.bias, self._epsilon)


class RMSNorm(Layer):
    def __init__(self, hidden_size, epsilon=1e-6, weight_attr=None, name=None):
        super().__init__()
        self._epsilon = epsilon
        self.weight = self.create_parameter([hidden_size], attr=weight_attr,
                                            default_initializer=Constant(1.0))

    def forward(self, x):
        return hot.rms_norm(x, self.weight, self._epsilon)


class Dropout(Layer):
    def __init__(self, p=0.5, axis=None, mode="upscale_in_train", name=None):
        super().__init__()
        self.p = p
        self.axis = axis
        self.mode = mode

    def forward(self, x):
        return F.dropout(x, self.p, axis=self.axis, training=self.training, mode=self.mode)


class Dropout2D(Layer):
    def __init__(self, p=0.5, data_format="NCHW", name=None):
        super().__init__()
        self.p = p

    def forward(self, x):
        return F.dropout2d(x, self.p, self.training)


class Flatten(Layer):
    def __init__(self, start_axis=1, stop_axis=-1):
        super().__init__()
        self.start_axis, self.stop_axis = start_axis, stop_axis

    def forward(self, x):
        return torch.flatten(x, self.start_axis, self.stop_axis)


class Identity(Layer):
    def __init__(self, *args, **kwargs):
        super().__init__()

    def forward(self, x):
        return x


class Upsample(Layer):
    def __init__(self, size=None, scale_factor=None, mode="nearest", align_corners=False,
                 align_mode=0, data_format="NCHW", name=None):
        super().__init__()
        self.size, self.scale_factor = size, scale_factor
        self.mode, self.align_corners = mode, align_corners

    def forward(self, x):
        return F.interpolate(x, self.size, self.scale_factor, self.mode, self.align_corners)


class Pad2D(Layer):
    def __init__(self, padding, mode="constant", value=0.0, data_format="NCHW", name=None):
        super().__init__()
        self.padding, self.mode, self.value = padding, mode, value

    def forward(self, x):
        return F.pad(x, self.padding, self.mode, self.value)
