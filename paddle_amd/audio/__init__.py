"""paddle.audio parity subset: spectrogram/mel features over torch ops."""
from __future__ import annotations

import math

import torch


class functional:
    @staticmethod
    def create_dct(n_mfcc, n_mels, norm="ortho"):
        n = torch.arange(float(n_mels))
        k = torch.arange(float(n_mfcc)).unsqueeze(1)
        dct = torch.cos(math.pi / n_mels * (n + 0.5) * k)
        if norm == "ortho":
            dct[0] *= 1.0 / math.sqrt(2.0)
            dct *= math.sqrt(2.0 / n_mels)
        return dct.t()

    @staticmethod
    def hz_to_mel(f, htk=False):
        if htk:
            return 2595.0 * math.log10(1.0 + f / 700.0)
        return 1127.0 * math.log(1.0 + f / 700.0)

    @staticmethod
    def mel_to_hz(m, htk=False):
        if htk:
            return 700.0 * (10.0 ** (m / 2595.0) - 1.0)
        return 700.0 * (math.exp(m / 1127.0) - 1.0)

    @staticmethod
    def compute_fbank_matrix(sr, n_fft, n_mels=64, f_min=0.0, f_max=None):
        f_max = f_max or sr / 2
        m_min = functional.hz_to_mel(f_min)
        m_max = functional.hz_to_mel(f_max)
        m_pts = torch.linspace(m_min, m_max, n_mels + 2)
        f_pts = torch.tensor([functional.mel_to_hz(float(m)) for m in m_pts])
        bins = torch.floor((n_fft + 1) * f_pts / sr).long()
        fb = torch.zeros(n_mels, n_fft // 2 + 1)
        for i in range(n_mels):
            l, c, r = bins[i], bins[i + 1], bins[i + 2]
            for j in range(int(l), int(c)):
                if c > l:
                    fb[i, j] = (j - l) / float(c - l)
            for j in range(int(c), int(r)):
                if r > c:
                    fb[i, j] = (r - j) / float(r - c)
        return fb


class features:
    class Spectrogram(torch.nn.Module):
        def __init__(self, n_fft=512, hop_length=None, win_length=None,
                     window="hann", power=2.0, center=True, pad_mode="reflect"):
            super().__init__()
            self.n_fft = n_fft
            self.hop = hop_length or n_fft // 4
            self.win = win_length or n_fft
            self.power = power
            self.center = center
            self.register_buffer("window", torch.hann_window(self.win))

        def forward(self, x):
            spec = torch.stft(x, self.n_fft, self.hop, self.win, self.window,
                              center=self.center, return_complex=True)
            return spec.abs() ** self.power

    class MelSpectrogram(torch.nn.Module):
        def __init__(self, sr=22050, n_fft=512, hop_length=None, n_mels=64,
                     f_min=50.0, f_max=None, **kw):
            super().__init__()
            self.spec = features.Spectrogram(n_fft, hop_length)
            self.register_buffer(
                "fbank", functional.compute_fbank_matrix(sr, n_fft, n_mels, f_min, f_max))

        def forward(self, x):
            return torch.matmul(self.fbank, self.spec(x))
