"""paddle.audio (reference: python/paddle/audio/__init__.py --
functional/features/backends/datasets subpackages; spectrogram and mel
features computed with torch.stft on the HIP FFT path)."""
from . import backends  # noqa: F401
from . import datasets  # noqa: F401
from . import features  # noqa: F401
from . import functional  # noqa: F401

from .backends import info, load, save  # noqa: F401,E402
