"""paddle.text parity surface: dataset shells (no network in this
environment -- datasets construct from local files or synthetic data)."""
from __future__ import annotations

from ..io import Dataset


class Imdb(Dataset):
    def __init__(self, data_file=None, mode="train", cutoff=150):
        raise NotImplementedError(
            "dataset downloads need network; pass local files via data_file "
            "to a custom paddle.io.Dataset")


class Conll05st(Dataset):
    def __init__(self, *a, **k):
        raise NotImplementedError("no network for dataset download")


class UCIHousing(Dataset):
    def __init__(self, *a, **k):
        raise NotImplementedError("no network for dataset download")


class ViterbiDecoder:
    def __init__(self, transitions, include_bos_eos_tag=True):
        import torch
        self.trans = torch.as_tensor(transitions)
        self.with_tag = include_bos_eos_tag

    def __call__(self, potentials, lengths):
        import torch
        b, s, n = potentials.shape
        scores = potentials[:, 0]
        hist = []
        for t in range(1, s):
            prev = scores.unsqueeze(2) + self.trans.unsqueeze(0)
            best, idx = prev.max(1)
            scores = best + potentials[:, t]
            hist.append(idx)
        best_final, last = scores.max(-1)
        paths = [last]
        for idx in reversed(hist):
            last = idx.gather(1, last.unsqueeze(1)).squeeze(1)
            paths.append(last)
        paths.reverse()
        return best_final, torch.stack(paths, dim=1)


# dataset classes are download-backed in the reference; no egress here, so
# they raise with instructions (reference: text/datasets/*.py)
class _NoNetworkDataset:
    def __init__(self, *a, **kw):
        raise RuntimeError(f"{type(self).__name__}: dataset download needs "
                           "network egress; place files locally and use "
                           "paddle.io.Dataset")


class Imikolov(_NoNetworkDataset):
    pass


class Movielens(_NoNetworkDataset):
    pass


class WMT14(_NoNetworkDataset):
    pass


class WMT16(_NoNetworkDataset):
    pass


def viterbi_decode(potentials, transition_params, lengths=None,
                   include_bos_eos_tag=True, name=None):
    """CRF Viterbi decoding (reference: text/viterbi_decode.py -> phi
    viterbi_decode kernel).  potentials [B,T,N], transition [N,N]."""
    import torch
    B, T, N = potentials.shape
    if lengths is None:
        lengths = torch.full((B,), T, dtype=torch.int64)
    scores = potentials[:, 0].clone()
    history = []
    for t in range(1, T):
        m = scores.unsqueeze(2) + transition_params.t().unsqueeze(0)
        best, idx = m.max(dim=1)
        scores = best + potentials[:, t]
        history.append(idx)
    best_final, last = scores.max(-1)
    paths = [last]
    for idx in reversed(history):
        last = idx.gather(1, last.unsqueeze(1)).squeeze(1)
        paths.append(last)
    paths.reverse()
    return best_final, torch.stack(paths, dim=1)
