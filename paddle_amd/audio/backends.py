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
