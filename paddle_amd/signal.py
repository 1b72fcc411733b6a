"""paddle.signal (reference: python/paddle/signal.py -- stft/istft)."""
from __future__ import annotations

import torch


def stft(x, n_fft, hop_length=None, win_length=None, window=None, center=True,
         pad_mode="reflect", normalized=False, onesided=True, name=None):
    return torch.stft(x, n_fft, hop_length=hop_length, win_length=win_length,
                      window=window, center=center, pad_mode=pad_mode,
                      normalized=normalized, onesided=onesided,
                      return_complex=True)


def istft(x, n_fft, hop_length=None, win_length=None, window=None, center=True,
          normalized=False, onesided=True, length=None, return_complex=False,
          name=None):
    return torch.istft(x, n_fft, hop_length=hop_length, win_length=win_length,
                       window=window, center=center, normalized=normalized,
                       onesided=onesided, length=length,
                       return_complex=return_complex)
