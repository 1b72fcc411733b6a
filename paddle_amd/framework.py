"""Core framework types: dtypes, places, flags, RNG.

MI355X-native rebuild of the reference's L0 layer
(paddle/common/flags.h, paddle/phi/common/place.h). One GPU backend
(gfx950), one CPU backend for plumbing tests -- no multi-backend Place
dispatch (SURVEY.md L1 note).
"""
from __future__ import annotations

import os
from typing import Any, Dict

import torch

# ---------------------------------------------------------------------------
# dtypes: paddle-style names aliased to torch dtypes
# ---------------------------------------------------------------------------
bool_ = torch.bool
uint8 = torch.uint8
int8 = torch.int8
int16 = torch.int16
int32 = torch.int32
int64 = torch.int64
float16 = torch.float16
bfloat16 = torch.bfloat16
float32 = torch.float32
float64 = torch.float64
complex64 = torch.complex64
complex128 = torch.complex128
float8_e4m3fn = torch.float8_e4m3fn  # OCP e4m3 (gfx950 MFMA fp8 format)
float8_e5m2 = torch.float8_e5m2

_STR2DTYPE: Dict[str, torch.dtype] = {
    "bool": torch.bool,
    "uint8": torch.uint8,
    "int8": torch.int8,
    "int16": torch.int16,
    "int32": torch.int32,
    "int64": torch.int64,
    "float16": torch.float16,
    "bfloat16": torch.bfloat16,
    "float32": torch.float32,
    "float64": torch.float64,
    "complex64": torch.complex64,
    "complex128": torch.complex128,
    "float8_e4m3fn": torch.float8_e4m3fn,
    "float8_e5m2": torch.float8_e5m2,
}

_DTYPE2STR = {v: k for k, v in _STR2DTYPE.items()}


def convert_dtype(dtype: Any) -> torch.dtype:
    """Accept a paddle-style dtype string, numpy dtype, or torch dtype."""
    if isinstance(dtype, torch.dtype):
        return dtype
    if isinstance(dtype, str):
        if dtype in _STR2DTYPE:
            return _STR2DTYPE[dtype]
        raise ValueError(f"unknown dtype string: {dtype!r}")
    # numpy dtype / type object
    import numpy as np

    npdt = np.dtype(dtype)
    name = npdt.name
    if name in _STR2DTYPE:
        return _STR2DTYPE[name]
    raise ValueError(f"cannot convert dtype {dtype!r}")


def dtype_name(dtype: torch.dtype) -> str:
    return _DTYPE2STR.get(dtype, str(dtype))


# ---------------------------------------------------------------------------
# Places.  GPUPlace == one MI355X (HIP device); CPUPlace for plumbing.
# Mirrors paddle.CUDAPlace/paddle.CPUPlace API shape.
# ---------------------------------------------------------------------------
class Place:
    def torch_device(self) -> torch.device:  # pragma: no cover - abstract
        raise NotImplementedError


class CPUPlace(Place):
    def torch_device(self) -> torch.device:
        return torch.device("cpu")

    def __repr__(self) -> str:
        return "Place(cpu)"

    def __eq__(self, other):
        return isinstance(other, CPUPlace)

    def __hash__(self):
        return hash("cpu")


class GPUPlace(Place):
    """A single MI355X. torch calls it "cuda" on ROCm; we keep one name."""

    def __init__(self, device_id: int = 0):
        self.device_id = int(device_id)

    def torch_device(self) -> torch.device:
        return torch.device("cuda", self.device_id)

    def get_device_id(self) -> int:
        return self.device_id

    def __repr__(self) -> str:
        return f"Place(gpu:{self.device_id})"

    def __eq__(self, other):
        return isinstance(other, GPUPlace) and other.device_id == self.device_id

    def __hash__(self):
        return hash(("gpu", self.device_id))


# Paddle-compat alias: CUDAPlace name is part of the public API surface.
CUDAPlace = GPUPlace


def _place_from_any(place) -> torch.device:
    if place is None:
        return get_default_device()
    if isinstance(place, Place):
        return place.torch_device()
    if isinstance(place, torch.device):
        return place
    if isinstance(place, str):
        p = place.replace("gpu", "cuda")
        return torch.device(p)
    raise ValueError(f"bad place {place!r}")


_default_device: torch.device | None = None


def set_device(device) -> torch.device:
    """paddle.set_device("gpu:0"/"cpu")."""
    global _default_device
    _default_device = _place_from_any(device)
    if _default_device.type == "cuda":
        torch.cuda.set_device(_default_device)
    return _default_device


def get_device() -> str:
    d = get_default_device()
    if d.type == "cuda":
        return f"gpu:{d.index if d.index is not None else torch.cuda.current_device()}"
    return "cpu"


def get_default_device() -> torch.device:
    global _default_device
    if _default_device is None:
        if torch.cuda.is_available():
            _default_device = torch.device("cuda", torch.cuda.current_device())
        else:
            _default_device = torch.device("cpu")
    return _default_device


def is_compiled_with_cuda() -> bool:
    # ROCm builds report True here (HIP is the CUDA-namespace on torch).
    return torch.backends.cuda.is_built()


def is_compiled_with_rocm() -> bool:
    return torch.version.hip is not None


# ---------------------------------------------------------------------------
# Flags registry (reference: paddle/common/flags.h:337 GetExportedFlagInfoMap,
# python set_flags/get_flags in python/paddle/base/framework.py:132).
# Env override: FLAGS_<name>=value at import time.
# ---------------------------------------------------------------------------
class _FlagInfo:
    __slots__ = ("name", "value", "default", "doc", "typ")

    def __init__(self, name, default, doc):
        self.name = name
        self.default = default
        self.doc = doc
        self.typ = type(default)
        self.value = default


_FLAGS: Dict[str, _FlagInfo] = {}


def _coerce(typ, raw):
    if typ is bool:
        if isinstance(raw, str):
            return raw.lower() in ("1", "true", "yes", "on")
        return bool(raw)
    return typ(raw)


def define_flag(name: str, default, doc: str = ""):
    info = _FlagInfo(name, default, doc)
    env = os.environ.get(name)
    if env is not None:
        info.value = _coerce(info.typ, env)
    _FLAGS[name] = info
    return info


def set_flags(flags: Dict[str, Any]):
    for k, v in flags.items():
        if k not in _FLAGS:
            raise KeyError(f"flag {k!r} not registered")
        _FLAGS[k].value = _coerce(_FLAGS[k].typ, v)


def get_flags(flags):
    if isinstance(flags, str):
        flags = [flags]
    return {k: _FLAGS[k].value for k in flags}


def get_flag(name: str):
    return _FLAGS[name].value


# Core flags (subset of the reference's 184; added as subsystems need them)
define_flag("FLAGS_check_nan_inf", False, "check every op output for nan/inf")
define_flag("FLAGS_benchmark", False, "sync after every op for timing")
define_flag("FLAGS_sharding_bucket_mb", 128.0,
            "bucket size (MB) for sharding reduce-scatter over xGMI")
define_flag("FLAGS_dp_bucket_mb", 128.0,
            "bucket size (MB) for data-parallel allreduce over xGMI "
            "(xGMI links are 153 GB/s point-to-point; larger buckets than "
            "the reference's 25MB amortise launch latency)")
define_flag("FLAGS_use_native_kernels", True,
            "use the gfx950 HIP extension for hot ops (fail loudly if "
            "missing on GPU); False falls back to torch composites")
define_flag("FLAGS_enable_async_trace", False, "collective watchdog")
define_flag("FLAGS_eager_communication_connection", False,
            "pre-build comms at group creation")


# ---------------------------------------------------------------------------
# RNG.  paddle.seed seeds every device generator (reference
# phi/core/generator.cc); the TP RNGStatesTracker lives in
# distributed/fleet/random.py.
# ---------------------------------------------------------------------------
def seed(s: int):
    torch.manual_seed(s)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(s)
    return s


def get_rng_state(place=None):
    d = _place_from_any(place) if place is not None else get_default_device()
    if d.type == "cuda":
        return torch.cuda.get_rng_state(d)
    return torch.get_rng_state()


def set_rng_state(state, place=None):
    d = _place_from_any(place) if place is not None else get_default_device()
    if d.type == "cuda":
        torch.cuda.set_rng_state(state, d)
    else:
        torch.set_rng_state(state)
