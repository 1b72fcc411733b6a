"""paddle.audio.backends (reference: audio/backends).  No audio codec in
this image; the wave backend handles PCM .wav via the stdlib."""
_backend = "wave"


def list_available_backends():
    return ["wave"]


def get_current_backend():
    return _backend


def set_backend(backend):
    global _backend
    assert backend in list_available_backends(), backend
    _backend = backend


def load(filepath, frame_offset=0, num_frames=-1, normalize=True):
    import wave

    import numpy as np
    import torch
    with wave.open(filepath, "rb") as w:
        sr = w.getframerate()
        w.setpos(frame_offset)
        n = num_frames if num_frames > 0 else w.getnframes() - frame_offset
        data = np.frombuffer(w.readframes(n), dtype=np.int16)
        data = data.reshape(-1, w.getnchannels()).T
    t = torch.from_numpy(data.astype("float32"))
    if normalize:
        t = t / 32768.0
    return t, sr


def info(filepath):
    """Audio metadata (reference audio/backends wave backend)."""
    import wave
    from dataclasses import dataclass

    @dataclass
    class AudioInfo:
        sample_rate: int
        num_samples: int
        num_channels: int
        bits_per_sample: int
        encoding: str = "PCM_S"

    with wave.open(filepath, "rb") as w:
        return AudioInfo(w.getframerate(), w.getnframes(), w.getnchannels(),
                         w.getsampwidth() * 8)


def save(filepath, src, sample_rate, channels_first=True,
         encoding="PCM_S", bits_per_sample=16):
    """Write a [C, N] (or [N, C]) float tensor as 16-bit PCM WAV."""
    import wave

    import numpy as np
    import torch
    t = src.detach().cpu()
    if not channels_first:
        t = t.t()
    data = (t.clamp(-1, 1).numpy().T * 32767.0).astype(np.int16)
    with wave.open(filepath, "wb") as w:
        w.setnchannels(data.shape[1] if data.ndim > 1 else 1)
        w.setsampwidth(2)
        w.setframerate(int(sample_rate))
        w.writeframes(data.tobytes())
