"""paddle.fft parity -- hipFFT via torch.fft on GPU, pocketfft on CPU."""
from __future__ import annotations

import torch.fft as tf


def fft(x, n=None, axis=-1, norm="backward", name=None):
    return tf.fft(x, n=n, dim=axis, norm=norm)


def ifft(x, n=None, axis=-1, norm="backward", name=None):
    return tf.ifft(x, n=n, dim=axis, norm=norm)


def fft2(x, s=None, axes=(-2, -1), norm="backward", name=None):
    return tf.fft2(x, s=s, dim=axes, norm=norm)


def ifft2(x, s=None, axes=(-2, -1), norm="backward", name=None):
    return tf.ifft2(x, s=s, dim=axes, norm=norm)


def fftn(x, s=None, axes=None, norm="backward", name=None):
    return tf.fftn(x, s=s, dim=axes, norm=norm)


def ifftn(x, s=None, axes=None, norm="backward", name=None):
    return tf.ifftn(x, s=s, dim=axes, norm=norm)


def rfft(x, n=None, axis=-1, norm="backward", name=None):
    return tf.rfft(x, n=n, dim=axis, norm=norm)


def irfft(x, n=None, axis=-1, norm="backward", name=None):
    return tf.irfft(x, n=n, dim=axis, norm=norm)


def rfft2(x, s=None, axes=(-2, -1), norm="backward", name=None):
    return tf.rfft2(x, s=s, dim=axes, norm=norm)


def irfft2(x, s=None, axes=(-2, -1), norm="backward", name=None):
    return tf.irfft2(x, s=s, dim=axes, norm=norm)


def rfftn(x, s=None, axes=None, norm="backward", name=None):
    return tf.rfftn(x, s=s, dim=axes, norm=norm)


def irfftn(x, s=None, axes=None, norm="backward", name=None):
    return tf.irfftn(x, s=s, dim=axes, norm=norm)


def hfft(x, n=None, axis=-1, norm="backward", name=None):
    return tf.hfft(x, n=n, dim=axis, norm=norm)


def ihfft(x, n=None, axis=-1, norm="backward", name=None):
    return tf.ihfft(x, n=n, dim=axis, norm=norm)


def fftfreq(n, d=1.0, dtype=None, name=None):
    return tf.fftfreq(n, d=d)


def rfftfreq(n, d=1.0, dtype=None, name=None):
    return tf.rfftfreq(n, d=d)


def fftshift(x, axes=None, name=None):
    return tf.fftshift(x, dim=axes)


def ifftshift(x, axes=None, name=None):
    return tf.ifftshift(x, dim=axes)


def hfft2(x, s=None, axes=(-2, -1), norm="backward", name=None):
    return torch.fft.hfft2(x, s=s, dim=axes, norm=norm)


def ihfft2(x, s=None, axes=(-2, -1), norm="backward", name=None):
    return torch.fft.ihfft2(x, s=s, dim=axes, norm=norm)


def hfftn(x, s=None, axes=None, norm="backward", name=None):
    return torch.fft.hfftn(x, s=s, dim=axes, norm=norm)


def ihfftn(x, s=None, axes=None, norm="backward", name=None):
    return torch.fft.ihfftn(x, s=s, dim=axes, norm=norm)
