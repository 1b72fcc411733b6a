"""paddle.distribution.transform parity (reference:
python/paddle/distribution/transform.py -- 13 Transform classes).

Thin adapters over torch.distributions.transforms exposing paddle's
method names: forward / inverse / forward_log_det_jacobian /
inverse_log_det_jacobian / forward_shape / inverse_shape.
"""
from __future__ import annotations

import torch
import torch.distributions.transforms as tt

__all__ = [
    'Transform', 'AbsTransform', 'AffineTransform', 'ChainTransform',
    'ExpTransform', 'IndependentTransform', 'PowerTransform',
    'ReshapeTransform', 'SigmoidTransform', 'SoftmaxTransform',
    'StackTransform', 'StickBreakingTransform', 'TanhTransform',
]


class Transform:
    _torch_cls = None

    def __init__(self, *args, **kwargs):
        if self._torch_cls is not None:
            self._t = self._torch_cls(*args, **kwargs)

    # paddle method surface -> torch transform
    def forward(self, x):
        return self._t(x)

    __call__ = forward

    def inverse(self, y):
        return self._t.inv(y)

    def forward_log_det_jacobian(self, x):
        return self._t.log_abs_det_jacobian(x, self._t(x))

    def inverse_log_det_jacobian(self, y):
        x = self._t.inv(y)
        return -self._t.log_abs_det_jacobian(x, y)

    def forward_shape(self, shape):
        return tuple(self._t.forward_shape(torch.Size(shape)))

    def inverse_shape(self, shape):
        return tuple(self._t.inverse_shape(torch.Size(shape)))


class AbsTransform(Transform):
    _torch_cls = tt.AbsTransform


class AffineTransform(Transform):
    def __init__(self, loc, scale):
        self._t = tt.AffineTransform(loc, scale)


class ChainTransform(Transform):
    """reference: composes transforms in order (torch ComposeTransform)."""

    def __init__(self, transforms):
        self._t = tt.ComposeTransform([t._t for t in transforms])


class ExpTransform(Transform):
    _torch_cls = tt.ExpTransform


class IndependentTransform(Transform):
    def __init__(self, base, reinterpreted_batch_rank):
        self._t = tt.IndependentTransform(base._t, reinterpreted_batch_rank)


class PowerTransform(Transform):
    def __init__(self, power):
        self._t = tt.PowerTransform(power)


class ReshapeTransform(Transform):
    def __init__(self, in_event_shape, out_event_shape):
        self._t = tt.ReshapeTransform(torch.Size(in_event_shape),
                                      torch.Size(out_event_shape))


class SigmoidTransform(Transform):
    _torch_cls = tt.SigmoidTransform


class SoftmaxTransform(Transform):
    _torch_cls = tt.SoftmaxTransform


class StackTransform(Transform):
    def __init__(self, transforms, axis=0):
        self._t = tt.StackTransform([t._t for t in transforms], dim=axis)


class StickBreakingTransform(Transform):
    _torch_cls = tt.StickBreakingTransform


class TanhTransform(Transform):
    _torch_cls = tt.TanhTransform
