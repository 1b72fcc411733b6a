"""paddle.audio.datasets (reference: audio/datasets -- TESS/ESC50):
download-backed; gated without network egress."""
from ..io import Dataset


class _NoNetworkAudioDataset(Dataset):
    def __init__(self, *a, **kw):
        raise RuntimeError(f"{type(self).__name__}: dataset download needs "
                           "network egress; point paddle.io.Dataset at "
                           "local files instead")


class TESS(_NoNetworkAudioDataset):
    pass


class ESC50(_NoNetworkAudioDataset):
    pass
