"""paddle.onnx parity (reference: paddle2onnx integration in
python/paddle/onnx/export.py) -- exports a Layer via torch.onnx."""
from __future__ import annotations

import os

import torch


def export(layer, path, input_spec=None, opset_version=17, **configs):
    """paddle.onnx.export(layer, path, input_spec=[InputSpec|Tensor,...]).
    Writes `<path>.onnx`."""
    if input_spec is None:
        raise ValueError("input_spec (list of example tensors or InputSpec) required")
    examples = []
    for spec in input_spec:
        if isinstance(spec, torch.Tensor):
            examples.append(spec)
        else:  # static.InputSpec
            from . import framework
            shape = [s if (s is not None and s != -1) else 1 for s in spec.shape]
            examples.append(torch.zeros(shape, dtype=framework.convert_dtype(spec.dtype)))
    out = path if path.endswith(".onnx") else path + ".onnx"
    d = os.path.dirname(out)
    if d:
        os.makedirs(d, exist_ok=True)
    layer.eval()
    try:
        import onnx as _onnx  # noqa: F401
    except ImportError as e:
        raise RuntimeError(
            "paddle.onnx.export requires the `onnx` package, which is not "
            "installed in this offline environment") from e
    torch.onnx.export(layer, tuple(examples), out, opset_version=opset_version,
                      dynamo=False)
    return out
