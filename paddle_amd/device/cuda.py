"""paddle.device.cuda as an importable submodule (reference:
python/paddle/device/cuda/__init__.py).  The same surface is also
reachable as attributes of paddle.device.cuda via the namespace class in
device/__init__.py; this module form supports
`import paddle.device.cuda` / `from paddle.device.cuda import graphs`."""
import torch

from . import (  # noqa: F401
    Event,
    Stream,
    current_stream,
    device_count,
    stream_guard,
    synchronize,
)

memory_allocated = torch.cuda.memory_allocated
max_memory_allocated = torch.cuda.max_memory_allocated
memory_reserved = torch.cuda.memory_reserved
max_memory_reserved = torch.cuda.max_memory_reserved
reset_peak_memory_stats = torch.cuda.reset_peak_memory_stats
empty_cache = torch.cuda.empty_cache
get_device_properties = torch.cuda.get_device_properties
get_device_name = torch.cuda.get_device_name


def get_device_capability(device=None):
    return torch.cuda.get_device_capability(device)


class graphs:
    """hipGraph capture (reference: device/cuda/graphs.py)."""

    CUDAGraph = torch.cuda.CUDAGraph

    @staticmethod
    def graph(g, pool=None, stream=None, capture_error_mode="global"):
        return torch.cuda.graph(g, pool=pool, stream=stream,
                                capture_error_mode=capture_error_mode)
