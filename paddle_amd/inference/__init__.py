"""paddle.inference facade (reference: paddle/fluid/inference/
AnalysisPredictor + python/paddle/inference/).

Round-1 scope: inference = Layer.eval() + jit.save/load;
the Predictor wraps a loaded Layer (or a user-provided one) and exposes
the ZeroCopy-style handle API.  TensorRT-analog passes are out of scope
-- the same gfx950 kernels serve training and inference.
"""
from __future__ import annotations

import numpy as np
import torch


class Config:
    def __init__(self, model_path=None, params_path=None):
        self.model_path = model_path
        self.params_path = params_path
        self._device = "gpu" if torch.cuda.is_available() else "cpu"
        self._layer = None

    def enable_use_gpu(self, memory_pool_init_size_mb=100, device_id=0):
        self._device = f"gpu:{device_id}"

    def disable_gpu(self):
        self._device = "cpu"

    def set_layer(self, layer):
        """attach an in-memory Layer (dygraph-first deployment path)"""
        self._layer = layer

    def switch_ir_optim(self, flag=True):
        pass

    def enable_memory_optim(self):
        pass


class _Handle:
    def __init__(self, name):
        self.name = name
        self._value = None

    def copy_from_cpu(self, arr):
        self._value = torch.as_tensor(np.asarray(arr))

    def copy_to_cpu(self):
        return self._value.detach().cpu().numpy()

    def reshape(self, shape):
        pass


class Predictor:
    def __init__(self, config: Config):
        self.config = config
        self._layer = config._layer
        if self._layer is None and config.model_path:
            from .. import jit
            prefix = config.model_path
            for suf in (".pdmodel",):
                if prefix.endswith(suf):
                    prefix = prefix[: -len(suf)]
            self._layer = jit.load(prefix)
        self._inputs = {}
        self._outputs = {}
        if self._layer is not None and hasattr(self._layer, "eval"):
            self._layer.eval()

    def get_input_names(self):
        return ["input_0"]

    def get_input_handle(self, name):
        h = self._inputs.setdefault(name, _Handle(name))
        return h

    def get_output_names(self):
        return list(self._outputs.keys()) or ["output_0"]

    def get_output_handle(self, name):
        return self._outputs.setdefault(name, _Handle(name))

    def run(self, inputs=None):
        dev = torch.device("cuda") if "gpu" in self.config._device and \
            torch.cuda.is_available() else torch.device("cpu")
        if inputs is not None:
            args = [torch.as_tensor(np.asarray(a)).to(dev) for a in inputs]
        else:
            args = [h._value.to(dev) for h in self._inputs.values()]
        with torch.no_grad():
            out = self._layer(*args)
        outs = out if isinstance(out, (list, tuple)) else [out]
        res = []
        for i, o in enumerate(outs):
            h = self.get_output_handle(f"output_{i}")
            h._value = o
            res.append(o.detach().cpu().numpy())
        return res


def create_predictor(config: Config):
    return Predictor(config)


# breadth parity (reference: python/paddle/inference/__init__.py __all__)
import enum as _enum

import torch as _torch


class DataType(_enum.Enum):
    FLOAT32 = 0
    INT64 = 1
    INT32 = 2
    UINT8 = 3
    INT8 = 4
    FLOAT16 = 5
    BFLOAT16 = 6
    FLOAT64 = 7
    BOOL = 8


class PlaceType(_enum.Enum):
    UNK = -1
    CPU = 0
    GPU = 1


class PrecisionType(_enum.Enum):
    Float32 = 0
    Half = 1
    Int8 = 2
    Bfloat16 = 3


Tensor = _torch.Tensor


def get_version():
    from .. import __version__
    return f"paddle_amd inference {__version__}"


def get_num_bytes_of_data_type(dtype):
    return {DataType.FLOAT32: 4, DataType.INT64: 8, DataType.INT32: 4,
            DataType.UINT8: 1, DataType.INT8: 1, DataType.FLOAT16: 2,
            DataType.BFLOAT16: 2, DataType.FLOAT64: 8, DataType.BOOL: 1}[dtype]


def get_trt_compile_version():
    return (0, 0, 0)  # TensorRT does not exist on ROCm; MIGraphX is round 2


def get_trt_runtime_version():
    return (0, 0, 0)


def _get_phi_kernel_name(op_name):
    return op_name


def convert_to_mixed_precision(model_file, params_file, mixed_model_file,
                               mixed_params_file, mixed_precision, backend,
                               **kwargs):
    raise NotImplementedError(
        "convert_to_mixed_precision operates on .pdmodel graphs; use "
        "paddle.amp at runtime instead")


class PredictorPool:
    def __init__(self, config, size=1):
        self._preds = [create_predictor(config) for _ in range(size)]

    def retrieve(self, idx):
        return self._preds[idx]


class XpuConfig:
    def __init__(self, *a, **kw):
        raise NotImplementedError("XPU is not a target of this MI355X build")
