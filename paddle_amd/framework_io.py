"""paddle.save / paddle.load -- .pdparams/.pdopt checkpoint format.

Reference: python/paddle/framework/io.py:773 (save), :1020 (load),
_pickle_save:413.  Format: a pickle (protocol 2..4) of the state_dict
with every tensor reduced to a numpy ndarray -- interchangeable with
the reference's files for fp32/fp16/int dtypes.  bf16 tensors are
stored as numpy uint16 (raw bits) exactly like the reference's
convert_to_numpy path, wrapped so load() restores bfloat16.
"""
from __future__ import annotations

import os
import pickle
import threading
from typing import Any

import numpy as np
import torch

_PROTOCOL = 2


class _BF16Array:
    """Marker wrapper: bf16 bits as uint16 ndarray (unpickles to bf16)."""

    def __init__(self, bits: np.ndarray):
        self.bits = bits


def _to_serializable(obj: Any) -> Any:
    if isinstance(obj, torch.Tensor):
        t = obj.detach().cpu()
        if t.dtype == torch.bfloat16:
            return _BF16Array(t.view(torch.uint16).numpy())
        return t.numpy()
    if isinstance(obj, dict):
        return {k: _to_serializable(v) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        conv = [_to_serializable(v) for v in obj]
        return type(obj)(conv) if not isinstance(obj, tuple) else tuple(conv)
    return obj


def _from_serializable(obj: Any, return_numpy=False) -> Any:
    if isinstance(obj, _BF16Array):
        t = torch.from_numpy(obj.bits.copy()).view(torch.bfloat16)
        return t.float().numpy() if return_numpy else t
    if isinstance(obj, np.ndarray):
        return obj if return_numpy else torch.from_numpy(obj.copy())
    if isinstance(obj, dict):
        return {k: _from_serializable(v, return_numpy) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        conv = [_from_serializable(v, return_numpy) for v in obj]
        return conv if isinstance(obj, list) else tuple(conv)
    return obj


def save(obj, path, protocol=_PROTOCOL, **configs):
    d = os.path.dirname(path)
    if d:
        os.makedirs(d, exist_ok=True)
    data = _to_serializable(obj)
    with open(path, "wb") as f:
        pickler = pickle.Pickler(f, protocol=max(protocol, 2))
        pickler.dump(data)


def load(path, return_numpy=False, **configs):
    with open(path, "rb") as f:
        data = pickle.load(f, encoding="latin1")
    return _from_serializable(data, return_numpy=return_numpy)


_async_threads = []


def async_save(obj, path, protocol=_PROTOCOL, sync_other_task=False, **configs):
    """Snapshot to CPU synchronously, write on a thread (io.py:94)."""
    snapshot = _to_serializable(obj)

    def _write():
        d = os.path.dirname(path)
        if d:
            os.makedirs(d, exist_ok=True)
        with open(path, "wb") as f:
            pickle.dump(snapshot, f, protocol=max(protocol, 2))

    t = threading.Thread(target=_write, daemon=True)
    t.start()
    _async_threads.append(t)
    return t


def clear_async_save_task_queue():
    for t in _async_threads:
        t.join()
    _async_threads.clear()


def save_safetensors(state_dict, path, metadata=None):
    """Save a flat {name: tensor} dict as .safetensors (ecosystem interchange;
    HF loaders read these directly).  Non-tensor entries are rejected --
    use paddle.save's pickle format for optimizer state."""
    from safetensors.torch import save_file
    flat = {}
    for k, v in state_dict.items():
        if not isinstance(v, torch.Tensor):
            raise ValueError(f"save_safetensors: '{k}' is not a tensor; "
                             "use paddle.save for mixed state")
        flat[k] = v.detach().cpu().contiguous()
    save_file(flat, path, metadata=metadata or {"format": "paddle_amd"})


def load_safetensors(path, device="cpu"):
    from safetensors.torch import load_file
    return load_file(path, device=device)
