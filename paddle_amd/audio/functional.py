"""paddle.audio.functional (reference: python/paddle/audio/functional/
{functional,window}.py)."""
from __future__ import annotations

import math

import torch


def hz_to_mel(freq, htk=False):
    t = torch.as_tensor(freq, dtype=torch.float64)
    if htk:
        out = 2595.0 * torch.log10(1.0 + t / 700.0)
    else:
        f_min, f_sp = 0.0, 200.0 / 3
        mels = (t - f_min) / f_sp
        min_log_hz = 1000.0
        min_log_mel = (min_log_hz - f_min) / f_sp
        logstep = math.log(6.4) / 27.0
        out = torch.where(t >= min_log_hz,
                          min_log_mel + torch.log(t / min_log_hz) / logstep,
                          mels)
    return out.item() if out.dim() == 0 and not torch.is_tensor(freq) else out


def mel_to_hz(mel, htk=False):
    t = torch.as_tensor(mel, dtype=torch.float64)
    if htk:
        out = 700.0 * (10.0 ** (t / 2595.0) - 1.0)
    else:
        f_min, f_sp = 0.0, 200.0 / 3
        freqs = f_min + f_sp * t
        min_log_hz = 1000.0
        min_log_mel = (min_log_hz - f_min) / f_sp
        logstep = math.log(6.4) / 27.0
        out = torch.where(t >= min_log_mel,
                          min_log_hz * torch.exp(logstep * (t - min_log_mel)),
                          freqs)
    return out.item() if out.dim() == 0 and not torch.is_tensor(mel) else out


def mel_frequencies(n_mels=64, f_min=0.0, f_max=11025.0, htk=False,
                    dtype="float32"):
    lo = hz_to_mel(torch.tensor(f_min), htk)
    hi = hz_to_mel(torch.tensor(f_max), htk)
    mels = torch.linspace(float(lo), float(hi), n_mels, dtype=torch.float64)
    from .. import framework
    return mel_to_hz(mels, htk).to(framework.convert_dtype(dtype))


def fft_frequencies(sr, n_fft, dtype="float32"):
    from .. import framework
    return torch.linspace(0, sr / 2, n_fft // 2 + 1,
                          dtype=framework.convert_dtype(dtype))


def compute_fbank_matrix(sr, n_fft, n_mels=64, f_min=0.0, f_max=None,
                         htk=False, norm="slaney", dtype="float32"):
    f_max = f_max or sr / 2
    fft_f = fft_frequencies(sr, n_fft, "float64").double()
    mel_f = mel_frequencies(n_mels + 2, f_min, f_max, htk, "float64").double()
    fdiff = mel_f[1:] - mel_f[:-1]
    ramps = mel_f.unsqueeze(1) - fft_f.unsqueeze(0)
    lower = -ramps[:-2] / fdiff[:-1].unsqueeze(1)
    upper = ramps[2:] / fdiff[1:].unsqueeze(1)
    fb = torch.clamp(torch.minimum(lower, upper), min=0)
    if norm == "slaney":
        enorm = 2.0 / (mel_f[2:n_mels + 2] - mel_f[:n_mels])
        fb = fb * enorm.unsqueeze(1)
    from .. import framework
    return fb.to(framework.convert_dtype(dtype))


def create_dct(n_mfcc, n_mels, norm="ortho", dtype="float32"):
    n = torch.arange(float(n_mels))
    k = torch.arange(float(n_mfcc)).unsqueeze(1)
    dct = torch.cos(math.pi / n_mels * (n + 0.5) * k)
    if norm == "ortho":
        dct[0] *= 1.0 / math.sqrt(2.0)
        dct *= math.sqrt(2.0 / n_mels)
    from .. import framework
    return dct.t().to(framework.convert_dtype(dtype))


def power_to_db(spect, ref_value=1.0, amin=1e-10, top_db=80.0):
    s = torch.as_tensor(spect)
    log_spec = 10.0 * torch.log10(torch.clamp(s, min=amin))
    log_spec -= 10.0 * math.log10(max(ref_value, amin))
    if top_db is not None:
        log_spec = torch.clamp(log_spec, min=float(log_spec.max()) - top_db)
    return log_spec


def get_window(window, win_length, fftbins=True, dtype="float32"):
    from .. import framework
    dt = framework.convert_dtype(dtype)
    periodic = fftbins
    name = window[0] if isinstance(window, tuple) else window
    fns = {"hann": torch.hann_window, "hamming": torch.hamming_window,
           "blackman": torch.blackman_window, "bartlett": torch.bartlett_window}
    if name in fns:
        return fns[name](win_length, periodic=periodic, dtype=dt)
    if name in ("boxcar", "rect", "rectangular"):
        return torch.ones(win_length, dtype=dt)
    if name == "kaiser":
        beta = window[1] if isinstance(window, tuple) else 12.0
        return torch.kaiser_window(win_length, periodic=periodic, beta=beta,
                                   dtype=dt)
    if name == "gaussian":
        std = window[1] if isinstance(window, tuple) else 7.0
        n = torch.arange(win_length, dtype=dt) - (win_length - 1) / 2
        return torch.exp(-0.5 * (n / std) ** 2)
    raise ValueError(f"unknown window {window}")
