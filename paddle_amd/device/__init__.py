"""paddle.device parity: HIP streams/events/graphs + memory stats.

Reference: python/paddle/device/ (cuda streams/events, graphs.py,
cuda/__init__.py:233 memory stats).  All of it maps onto torch's HIP
runtime ("cuda" namespace on ROCm).
"""
from __future__ import annotations

import torch

from .. import framework


def set_device(device):
    return framework.set_device(device)


def get_device():
    return framework.get_device()


def is_compiled_with_cuda():
    return framework.is_compiled_with_cuda()


def is_compiled_with_rocm():
    return framework.is_compiled_with_rocm()


def device_count():
    return torch.cuda.device_count() if torch.cuda.is_available() else 0


class Stream:
    """HIP stream (reference: phi GPUContext stream; python device/cuda/streams.py)."""

    def __init__(self, device=None, priority=2, stream_base=None):
        if stream_base is not None:
            self._s = stream_base
        else:
            # paddle priority: 1 = high, 2 = normal -> torch: -1 high, 0 normal
            self._s = torch.cuda.Stream(device=device, priority=-1 if priority == 1 else 0)

    @property
    def stream_base(self):
        return self._s

    def wait_event(self, event):
        self._s.wait_event(event._e if isinstance(event, Event) else event)

    def wait_stream(self, stream):
        self._s.wait_stream(stream._s if isinstance(stream, Stream) else stream)

    def record_event(self, event=None):
        e = event or Event()
        (e._e if isinstance(e, Event) else e).record(self._s)
        return e

    def synchronize(self):
        self._s.synchronize()

    def query(self):
        return self._s.query()


class Event:
    def __init__(self, enable_timing=False, blocking=False, interprocess=False):
        self._e = torch.cuda.Event(enable_timing=enable_timing, blocking=blocking,
                                   interprocess=interprocess)

    def record(self, stream=None):
        self._e.record(stream._s if isinstance(stream, Stream) else stream)

    def query(self):
        return self._e.query()

    def synchronize(self):
        self._e.synchronize()

    def elapsed_time(self, other):
        return self._e.elapsed_time(other._e if isinstance(other, Event) else other)


def current_stream(device=None):
    return Stream(stream_base=torch.cuda.current_stream(device))


def stream_guard(stream):
    return torch.cuda.stream(stream._s if isinstance(stream, Stream) else stream)


def synchronize(device=None):
    if torch.cuda.is_available():
        torch.cuda.synchronize(device)


class cuda:
    """paddle.device.cuda namespace."""

    Stream = Stream
    Event = Event

    @staticmethod
    def device_count():
        return device_count()

    @staticmethod
    def current_stream(device=None):
        return current_stream(device)

    @staticmethod
    def stream_guard(stream):
        return stream_guard(stream)

    @staticmethod
    def synchronize(device=None):
        synchronize(device)

    @staticmethod
    def memory_allocated(device=None):
        return torch.cuda.memory_allocated(device)

    @staticmethod
    def max_memory_allocated(device=None):
        return torch.cuda.max_memory_allocated(device)

    @staticmethod
    def memory_reserved(device=None):
        return torch.cuda.memory_reserved(device)

    @staticmethod
    def max_memory_reserved(device=None):
        return torch.cuda.max_memory_reserved(device)

    @staticmethod
    def reset_peak_memory_stats(device=None):
        torch.cuda.reset_peak_memory_stats(device)

    @staticmethod
    def empty_cache():
        torch.cuda.empty_cache()

    @staticmethod
    def get_device_properties(device=None):
        return torch.cuda.get_device_properties(device)

    @staticmethod
    def get_device_name(device=None):
        return torch.cuda.get_device_name(device)

    class graphs:
        """hipGraph capture (reference: python/paddle/device/cuda/graphs.py).
        torch.cuda.CUDAGraph on ROCm is hipGraph."""

        CUDAGraph = torch.cuda.CUDAGraph

        @staticmethod
        def graph(g, pool=None, stream=None, capture_error_mode="global"):
            return torch.cuda.graph(g, pool=pool, stream=stream,
                                    capture_error_mode=capture_error_mode)


class CUDAGraph:
    """paddle.device.cuda.graphs.CUDAGraph parity wrapper over hipGraph."""

    def __init__(self, place=None, mode="thread_local"):
        self._g = torch.cuda.CUDAGraph()

    def capture_begin(self):
        self._ctx = torch.cuda.graph(self._g)
        self._ctx.__enter__()

    def capture_end(self):
        self._ctx.__exit__(None, None, None)

    def replay(self):
        self._g.replay()

    def reset(self):
        self._g.reset()


# breadth parity (reference: device/__init__.py __all__)
def get_cudnn_version():
    import torch
    v = torch.backends.cudnn.version()  # MIOpen version on ROCm
    return v


class XPUPlace:
    def __init__(self, *a):
        raise NotImplementedError("XPU is not a target of this MI355X build")


class IPUPlace:
    def __init__(self, *a):
        raise NotImplementedError("IPU is not a target of this MI355X build")


def is_compiled_with_xpu():
    return False


def is_compiled_with_ipu():
    return False


def is_compiled_with_cinn():
    return False  # by design: hand-fused HIP kernels, no IR compiler


def is_compiled_with_distribute():
    return True


def is_compiled_with_custom_device(device_type=None):
    return False


def get_all_device_type():
    import torch
    return ["cpu", "gpu"] if torch.cuda.is_available() else ["cpu"]


def get_all_custom_device_type():
    return []


def get_available_device():
    import torch
    return ([f"gpu:{i}" for i in range(torch.cuda.device_count())]
            if torch.cuda.is_available() else ["cpu"])


def get_available_custom_device():
    return []


def set_stream(stream):
    import torch
    torch.cuda.set_stream(stream._raw if hasattr(stream, "_raw") else stream)
    return stream
