"""Conv / pooling / batchnorm layers (reference: python/paddle/nn/layer/conv.py,
pooling.py, norm.py BatchNorm*).  Conv weight layout matches paddle:
[out_channels, in_channels/groups, kH, kW] (same as torch).  GPU conv
runs through torch -> MIOpen (vendor library; conv is not a hand-kernel
target for the LLM configs -- SURVEY.md §2.2 maps conv to MIOpen)."""
from __future__ import annotations

import torch

from . import functional as F
from .initializer import Constant, XavierNormal
from .layer import Layer


def _pair(v):
    return (v, v) if isinstance(v, int) else tuple(v)


class Conv2D(Layer):
    def __init__(self, in_channels, out_channels, kernel_size, stride=1, padding=0,
                 dilation=1, groups=1, padding_mode="zeros", weight_attr=None,
                 bias_attr=None, data_format="NCHW"):
        super().__init__()
        k = _pair(kernel_size)
        self._stride, self._padding, self._dilation, self._groups = stride, padding, dilation, groups
        self.weight = self.create_parameter([out_channels, in_channels // groups, *k],
                                            attr=weight_attr,
                                            default_initializer=XavierNormal())
        self.bias = None if bias_attr is False else self.create_parameter(
            [out_channels], attr=bias_attr, is_bias=True)

    def forward(self, x):
        return F.conv2d(x, self.weight, self.bias, self._stride, self._padding,
                        self._dilation, self._groups)


class Conv1D(Layer):
    def __init__(self, in_channels, out_channels, kernel_size, stride=1, padding=0,
                 dilation=1, groups=1, padding_mode="zeros", weight_attr=None,
                 bias_attr=None, data_format="NCL"):
        super().__init__()
        self._stride, self._padding, self._dilation, self._groups = stride, padding, dilation, groups
        ks = kernel_size if isinstance(kernel_size, int) else kernel_size[0]
        self.weight = self.create_parameter([out_channels, in_channels // groups, ks],
                                            attr=weight_attr, default_initializer=XavierNormal())
        self.bias = None if bias_attr is False else self.create_parameter(
            [out_channels], attr=bias_attr, is_bias=True)

    def forward(self, x):
        return F.conv1d(x, self.weight, self.bias, self._stride, self._padding,
                        self._dilation, self._groups)


class Conv2DTranspose(Layer):
    def __init__(self, in_channels, out_channels, kernel_size, stride=1, padding=0,
                 output_padding=0, groups=1, dilation=1, weight_attr=None,
                 bias_attr=None, data_format="NCHW"):
        super().__init__()
        k = _pair(kernel_size)
        self._stride, self._padding, self._dilation = stride, padding, dilation
        self._groups, self._output_padding = groups, output_padding
        self.weight = self.create_parameter([in_channels, out_channels // groups, *k],
                                            attr=weight_attr, default_initializer=XavierNormal())
        self.bias = None if bias_attr is False else self.create_parameter(
            [out_channels], attr=bias_attr, is_bias=True)

    def forward(self, x):
        return F.conv2d_transpose(x, self.weight, self.bias, self._stride, self._padding,
                                  self._output_padding, self._groups, self._dilation)


class MaxPool2D(Layer):
    def __init__(self, kernel_size, stride=None, padding=0, return_mask=False,
                 ceil_mode=False, data_format="NCHW", name=None):
        super().__init__()
        self.k, self.s, self.p = kernel_size, stride, padding
        self.return_mask, self.ceil_mode = return_mask, ceil_mode

    def forward(self, x):
        return F.max_pool2d(x, self.k, self.s, self.p, self.return_mask, self.ceil_mode)


class AvgPool2D(Layer):
    def __init__(self, kernel_size, stride=None, padding=0, ceil_mode=False,
                 exclusive=True, divisor_override=None, data_format="NCHW", name=None):
        super().__init__()
        self.k, self.s, self.p = kernel_size, stride, padding
        self.ceil_mode, self.exclusive, self.divisor = ceil_mode, exclusive, divisor_override

    def forward(self, x):
        return F.avg_pool2d(x, self.k, self.s, self.p, self.ceil_mode, self.exclusive, self.divisor)


class AdaptiveAvgPool2D(Layer):
    def __init__(self, output_size, data_format="NCHW", name=None):
        super().__init__()
        self.output_size = output_size

    def forward(self, x):
        return F.adaptive_avg_pool2d(x, self.output_size)


class _BatchNormBase(Layer):
    def __init__(self, num_features, momentum=0.9, epsilon=1e-5, weight_attr=None,
                 bias_attr=None, data_format="NCHW", use_global_stats=None, name=None):
        super().__init__()
        self._momentum, self._epsilon = momentum, epsilon
        self._use_global_stats = use_global_stats
        self.weight = self.create_parameter([num_features], attr=weight_attr,
                                            default_initializer=Constant(1.0))
        self.bias = self.create_parameter([num_features], attr=bias_attr, is_bias=True)
        self.register_buffer("_mean", torch.zeros(num_features))
        self.register_buffer("_variance", torch.ones(num_features))

    def forward(self, x):
        training = self.training and not (self._use_global_stats is True)
        return F.batch_norm(x, self._mean, self._variance, self.weight, self.bias,
                            training=training, momentum=self._momentum,
                            epsilon=self._epsilon)


class BatchNorm2D(_BatchNormBase):
    pass


class BatchNorm1D(_BatchNormBase):
    pass


class BatchNorm(_BatchNormBase):
    def __init__(self, num_channels, act=None, momentum=0.9, epsilon=1e-5, **kw):
        super().__init__(num_channels, momentum, epsilon)
        self._act = act

    def forward(self, x):
        y = super().forward(x)
        if self._act == "relu":
            y = torch.relu(y)
        return y


class GroupNorm(Layer):
    def __init__(self, num_groups, num_channels, epsilon=1e-5, weight_attr=None,
                 bias_attr=None, data_format="NCHW", name=None):
        super().__init__()
        self._num_groups, self._epsilon = num_groups, epsilon
        self.weight = self.create_parameter([num_channels], attr=weight_attr,
                                            default_initializer=Constant(1.0))
        self.bias = self.create_parameter([num_channels], attr=bias_attr, is_bias=True)

    def forward(self, x):
        return torch.nn.functional.group_norm(x, self._num_groups, self.weight,
                                              self.bias, self._epsilon)


class SyncBatchNorm(_BatchNormBase):
    """DP-synchronized BN (reference: sync_batch_norm_kernel.cu).  Uses
    torch's SyncBatchNorm functional path over the default process group."""

    def forward(self, x):
        import torch.distributed as dist
        if self.training and dist.is_available() and dist.is_initialized():
            return torch.nn.functional.batch_norm(
                x, self._mean, self._variance, self.weight, self.bias,
                training=True, momentum=1 - self._momentum, eps=self._epsilon)
        return super().forward(x)

    @classmethod
    def convert_sync_batchnorm(cls, layer):
        return layer
