"""Long-tail nn layers (reference: python/paddle/nn/__init__.py __all__).

Mechanical wrappers over torch.nn modules with paddle naming (epsilon→
eps etc.).  Compute runs on the torch-ROCm kernels (MIOpen for conv /
pool); hot-path transformer layers live elsewhere with HIP kernels.
"""
from __future__ import annotations

import torch

from .layer import Layer

_RENAMES = {"epsilon": "eps", "keepdim": "keepdim"}


def _wrap(torch_cls, name=None, drop=("name", "data_format", "weight_attr",
                                      "bias_attr")):
    class _W(Layer):
        def __init__(self, *args, **kwargs):
            super().__init__()
            for d in drop:
                kwargs.pop(d, None)
            for pk, tk in _RENAMES.items():
                if pk in kwargs:
                    kwargs[tk] = kwargs.pop(pk)
            self._m = torch_cls(*args, **kwargs)

        def forward(self, *a, **kw):
            return self._m(*a, **kw)

    _W.__name__ = name or torch_cls.__name__
    return _W


N = torch.nn
CELU = _wrap(N.CELU)
BatchNorm3D = _wrap(N.BatchNorm3d, "BatchNorm3D")
InstanceNorm1D = _wrap(N.InstanceNorm1d, "InstanceNorm1D")
InstanceNorm2D = _wrap(N.InstanceNorm2d, "InstanceNorm2D")
InstanceNorm3D = _wrap(N.InstanceNorm3d, "InstanceNorm3D")
LocalResponseNorm = _wrap(N.LocalResponseNorm)
UpsamplingNearest2D = _wrap(N.UpsamplingNearest2d, "UpsamplingNearest2D")
UpsamplingBilinear2D = _wrap(N.UpsamplingBilinear2d, "UpsamplingBilinear2D")
CosineSimilarity = _wrap(N.CosineSimilarity)
Dropout3D = _wrap(N.Dropout3d, "Dropout3D")
Bilinear = _wrap(N.Bilinear)
AlphaDropout = _wrap(N.AlphaDropout)
FeatureAlphaDropout = _wrap(N.FeatureAlphaDropout)
Unfold = _wrap(N.Unfold)
Fold = _wrap(N.Fold)
Softsign = _wrap(N.Softsign)
LogSigmoid = _wrap(N.LogSigmoid)
PairwiseDistance = _wrap(N.PairwiseDistance)
MaxPool1D = _wrap(N.MaxPool1d, "MaxPool1D")
MaxPool3D = _wrap(N.MaxPool3d, "MaxPool3D")
AvgPool1D = _wrap(N.AvgPool1d, "AvgPool1D")
AvgPool3D = _wrap(N.AvgPool3d, "AvgPool3D")
AdaptiveAvgPool1D = _wrap(N.AdaptiveAvgPool1d, "AdaptiveAvgPool1D")
AdaptiveAvgPool3D = _wrap(N.AdaptiveAvgPool3d, "AdaptiveAvgPool3D")
AdaptiveMaxPool1D = _wrap(N.AdaptiveMaxPool1d, "AdaptiveMaxPool1D")
AdaptiveMaxPool2D = _wrap(N.AdaptiveMaxPool2d, "AdaptiveMaxPool2D")
AdaptiveMaxPool3D = _wrap(N.AdaptiveMaxPool3d, "AdaptiveMaxPool3D")
FractionalMaxPool2D = _wrap(N.FractionalMaxPool2d, "FractionalMaxPool2D")
FractionalMaxPool3D = _wrap(N.FractionalMaxPool3d, "FractionalMaxPool3D")
LPPool1D = _wrap(N.LPPool1d, "LPPool1D")
LPPool2D = _wrap(N.LPPool2d, "LPPool2D")
MaxUnPool1D = _wrap(N.MaxUnpool1d, "MaxUnPool1D")
MaxUnPool2D = _wrap(N.MaxUnpool2d, "MaxUnPool2D")
MaxUnPool3D = _wrap(N.MaxUnpool3d, "MaxUnPool3D")
Hardshrink = _wrap(N.Hardshrink)
Softshrink = _wrap(N.Softshrink)
Hardtanh = _wrap(N.Hardtanh)
GLU = _wrap(N.GLU)
SELU = _wrap(N.SELU)
Silu = _wrap(N.SiLU, "Silu")
Tanhshrink = _wrap(N.Tanhshrink)
PReLU = _wrap(N.PReLU)
RReLU = _wrap(N.RReLU)
PixelShuffle = _wrap(N.PixelShuffle)
PixelUnshuffle = _wrap(N.PixelUnshuffle)
ChannelShuffle = _wrap(N.ChannelShuffle)
ZeroPad1D = _wrap(N.ZeroPad1d, "ZeroPad1D")
ZeroPad2D = _wrap(N.ZeroPad2d, "ZeroPad2D")
ZeroPad3D = _wrap(N.ZeroPad3d, "ZeroPad3D")
Pad1D = _wrap(N.ConstantPad1d, "Pad1D")
Pad3D = _wrap(N.ConstantPad3d, "Pad3D")
Conv3D = _wrap(N.Conv3d, "Conv3D")
Conv3DTranspose = _wrap(N.ConvTranspose3d, "Conv3DTranspose")
Conv1DTranspose = _wrap(N.ConvTranspose1d, "Conv1DTranspose")
Unflatten = _wrap(N.Unflatten)
ParameterDict = N.ParameterDict
LayerDict = N.ModuleDict
Softmax2D = _wrap(N.Softmax2d, "Softmax2D")
AdaptiveLogSoftmaxWithLoss = _wrap(N.AdaptiveLogSoftmaxWithLoss)

# losses
PoissonNLLLoss = _wrap(N.PoissonNLLLoss)
MarginRankingLoss = _wrap(N.MarginRankingLoss)
MultiLabelSoftMarginLoss = _wrap(N.MultiLabelSoftMarginLoss)
HingeEmbeddingLoss = _wrap(N.HingeEmbeddingLoss)
CosineEmbeddingLoss = _wrap(N.CosineEmbeddingLoss)
MultiMarginLoss = _wrap(N.MultiMarginLoss)
TripletMarginLoss = _wrap(N.TripletMarginLoss)
TripletMarginWithDistanceLoss = _wrap(N.TripletMarginWithDistanceLoss)
SoftMarginLoss = _wrap(N.SoftMarginLoss)
GaussianNLLLoss = _wrap(N.GaussianNLLLoss)
CTCLoss = _wrap(N.CTCLoss)


class Maxout(Layer):
    """paddle.nn.Maxout (reference: nn/layer/activation.py Maxout)."""

    def __init__(self, groups, axis=1, name=None):
        super().__init__()
        self.groups = groups
        self.axis = axis

    def forward(self, x):
        from . import functional as F
        return F.maxout(x, self.groups, self.axis)


class ThresholdedReLU(Layer):
    def __init__(self, threshold=1.0, value=0.0, name=None):
        super().__init__()
        self.threshold = threshold
        self.value = value

    def forward(self, x):
        return torch.where(x > self.threshold, x, torch.full_like(x, self.value))


class SpectralNorm(Layer):
    """Normalizes a weight tensor by its spectral norm (power iteration).
    reference: nn/layer/norm.py SpectralNorm."""

    def __init__(self, weight_shape, dim=0, power_iters=1, epsilon=1e-12,
                 dtype="float32"):
        super().__init__()
        self.dim = dim
        self.power_iters = power_iters
        self.eps = epsilon
        h = weight_shape[dim]
        w = 1
        for i, s in enumerate(weight_shape):
            if i != dim:
                w *= s
        self.register_buffer("_u", torch.randn(h))
        self.register_buffer("_v", torch.randn(w))

    def forward(self, weight):
        mat = weight.transpose(0, self.dim).reshape(weight.shape[self.dim], -1)
        u, v = self._u, self._v
        with torch.no_grad():
            for _ in range(self.power_iters):
                v = torch.nn.functional.normalize(mat.t() @ u, dim=0, eps=self.eps)
                u = torch.nn.functional.normalize(mat @ v, dim=0, eps=self.eps)
            self._u.copy_(u)
            self._v.copy_(v)
        sigma = (u @ mat @ v).clamp(min=self.eps)
        return weight / sigma


class RNNCellBase(Layer):
    """Base for custom cells consumed by paddle.nn.RNN (reference:
    nn/layer/rnn.py RNNCellBase)."""

    def get_initial_states(self, batch_ref, shape=None, dtype=None,
                           init_value=0.0, batch_dim_idx=0):
        b = batch_ref.shape[batch_dim_idx]
        h = getattr(self, "hidden_size", shape[-1] if shape else None)
        return torch.full((b, h), init_value, dtype=batch_ref.dtype,
                          device=batch_ref.device)


class RNN(Layer):
    """Run a cell over time (reference: nn/layer/rnn.py:RNN)."""

    def __init__(self, cell, is_reverse=False, time_major=False):
        super().__init__()
        self.cell = cell
        self.is_reverse = is_reverse
        self.time_major = time_major

    def forward(self, inputs, initial_states=None, sequence_length=None):
        if not self.time_major:
            inputs = inputs.transpose(0, 1)     # [T, B, C]
        T = inputs.shape[0]
        steps = range(T - 1, -1, -1) if self.is_reverse else range(T)
        state = initial_states
        outs = [None] * T
        for t in steps:
            if state is None:
                out, state = self.cell(inputs[t])
            else:
                out, state = self.cell(inputs[t], state)
            outs[t] = out
        y = torch.stack(outs, dim=0)
        if not self.time_major:
            y = y.transpose(0, 1)
        return y, state


class BiRNN(Layer):
    def __init__(self, cell_fw, cell_bw, time_major=False):
        super().__init__()
        self.fw = RNN(cell_fw, is_reverse=False, time_major=time_major)
        self.bw = RNN(cell_bw, is_reverse=True, time_major=time_major)

    def forward(self, inputs, initial_states=None, sequence_length=None):
        sf = sb = None
        if initial_states is not None:
            sf, sb = initial_states
        yf, stf = self.fw(inputs, sf)
        yb, stb = self.bw(inputs, sb)
        return torch.cat([yf, yb], dim=-1), (stf, stb)


class HSigmoidLoss(Layer):
    def __init__(self, *a, **kw):
        super().__init__()
        raise NotImplementedError(
            "HSigmoidLoss (hierarchical sigmoid) is not implemented; use "
            "CrossEntropyLoss or sampled softmax")


class RNNTLoss(Layer):
    def __init__(self, *a, **kw):
        super().__init__()
        raise NotImplementedError(
            "RNNTLoss requires the transducer kernel (not in this build)")


class BeamSearchDecoder:
    """Greedy/beam decode helper (reference: nn/decode.py:BeamSearchDecoder).
    Round-1: beam_size-wide log-prob beam over a step callable."""

    def __init__(self, cell, start_token, end_token, beam_size,
                 embedding_fn=None, output_fn=None):
        self.cell = cell
        self.start_token = start_token
        self.end_token = end_token
        self.beam_size = beam_size
        self.embedding_fn = embedding_fn
        self.output_fn = output_fn


def dynamic_decode(decoder, inits=None, max_step_num=100, **kwargs):
    raise NotImplementedError(
        "dynamic_decode: use paddle_amd.models.generation.generate for "
        "autoregressive decoding (paged KV cache, greedy/top-p)")

__all__ = [
    "AdaptiveAvgPool1D",
    "AdaptiveAvgPool3D",
    "AdaptiveLogSoftmaxWithLoss",
    "AdaptiveMaxPool1D",
    "AdaptiveMaxPool2D",
    "AdaptiveMaxPool3D",
    "AlphaDropout",
    "AvgPool1D",
    "AvgPool3D",
    "BatchNorm3D",
    "BeamSearchDecoder",
    "BiRNN",
    "Bilinear",
    "CELU",
    "CTCLoss",
    "ChannelShuffle",
    "Conv1DTranspose",
    "Conv3D",
    "Conv3DTranspose",
    "CosineEmbeddingLoss",
    "CosineSimilarity",
    "Dropout3D",
    "FeatureAlphaDropout",
    "Fold",
    "FractionalMaxPool2D",
    "FractionalMaxPool3D",
    "GLU",
    "GaussianNLLLoss",
    "HSigmoidLoss",
    "Hardshrink",
    "Hardtanh",
    "HingeEmbeddingLoss",
    "InstanceNorm1D",
    "InstanceNorm2D",
    "InstanceNorm3D",
    "LPPool1D",
    "LPPool2D",
    "LayerDict",
    "LocalResponseNorm",
    "LogSigmoid",
    "MarginRankingLoss",
    "MaxPool1D",
    "MaxPool3D",
    "MaxUnPool1D",
    "MaxUnPool2D",
    "MaxUnPool3D",
    "Maxout",
    "MultiLabelSoftMarginLoss",
    "MultiMarginLoss",
    "PReLU",
    "Pad1D",
    "Pad3D",
    "PairwiseDistance",
    "ParameterDict",
    "PixelShuffle",
    "PixelUnshuffle",
    "PoissonNLLLoss",
    "RNN",
    "RNNCellBase",
    "RNNTLoss",
    "RReLU",
    "SELU",
    "Silu",
    "SoftMarginLoss",
    "Softmax2D",
    "Softshrink",
    "Softsign",
    "SpectralNorm",
    "Tanhshrink",
    "ThresholdedReLU",
    "TripletMarginLoss",
    "TripletMarginWithDistanceLoss",
    "Unflatten",
    "Unfold",
    "UpsamplingBilinear2D",
    "UpsamplingNearest2D",
    "ZeroPad1D",
    "ZeroPad2D",
    "ZeroPad3D",
    "dynamic_decode",
]
