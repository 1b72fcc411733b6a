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
    """Legacy round-1 wrapper kept ONLY so old repo-saved files still load.
    New files use the reference's own convention (raw uint16 ndarray)."""

    def __init__(self, bits: np.ndarray):
        self.bits = bits


_name_counter = [0]


def _tensor_to_reference_form(obj: torch.Tensor):
    """Reference reduce_varbase (io.py:425): a tensor pickles as the plain
    tuple (name, ndarray).  bf16 has no numpy dtype; the reference's
    Tensor.numpy() emits raw-bits uint16 and paddle.to_tensor maps uint16
    back to bfloat16 (the uint16-means-bf16 convention)."""
    t = obj.detach().cpu().contiguous()
    if t.dtype == torch.bfloat16:
        data = t.view(torch.uint16).numpy()
    else:
        data = t.numpy()
    name = getattr(obj, "name", None)
    if not name:
        _name_counter[0] += 1
        name = f"generated_tensor_{_name_counter[0]}"
    return (name, data)


def _to_serializable(obj: Any) -> Any:
    if isinstance(obj, torch.Tensor):
        return _tensor_to_reference_form(obj)
    if isinstance(obj, dict):
        return {k: _to_serializable(v) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        conv = [_to_serializable(v) for v in obj]
        return type(obj)(conv) if not isinstance(obj, tuple) else tuple(conv)
    return obj


def _is_varbase_tuple(obj) -> bool:
    # reference _transformed_from_varbase (io.py:549)
    return (isinstance(obj, tuple) and len(obj) == 2
            and isinstance(obj[0], str) and isinstance(obj[1], np.ndarray))


def _ndarray_to_tensor(arr: np.ndarray, return_numpy: bool):
    if return_numpy:
        return arr
    if arr.dtype == np.uint16:  # uint16-means-bf16 convention
        return torch.from_numpy(arr.copy()).view(torch.bfloat16)
    return torch.from_numpy(arr.copy())


def _from_serializable(obj: Any, return_numpy=False) -> Any:
    if isinstance(obj, _BF16Array):  # legacy round-1 files
        t = torch.from_numpy(obj.bits.copy()).view(torch.bfloat16)
        return t.float().numpy() if return_numpy else t
    if _is_varbase_tuple(obj):
        t = _ndarray_to_tensor(obj[1], return_numpy)
        if not return_numpy:
            try:
                t.name = obj[0]
            except (AttributeError, RuntimeError):
                pass
        return t
    if isinstance(obj, np.ndarray):
        return _ndarray_to_tensor(obj, return_numpy)
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
