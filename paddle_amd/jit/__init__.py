"""paddle.jit parity surface (reference: python/paddle/jit/).

save/load: the layer's forward is captured by tracing through the torch
substrate with the HIP-native dispatch disabled (so the recorded graph
contains only substrate ops and is executable anywhere), serialized as a
TorchScript archive in the reference's file layout (path.pdmodel program
meta + path.pdiparams reference-form weights + path.pdscript program).
jit.load returns an EXECUTABLE TranslatedLayer -- no model class needed
(reference python/paddle/jit/api.py save/load, static/io.py:513 layout).
to_static remains an eager-preserving wrapper (no SOT/AST bytecode pass;
SURVEY §7 scopes the compiler out).
"""
from __future__ import annotations

import contextlib
import os
import pickle

import torch


class StaticFunction:
    """Trace-based program capture (the MI355X stand-in for the
    reference's AST/SOT to_static, dygraph_to_static/program_translator.py):
    the first call with a given input signature traces the function
    through the substrate (op wrappers in trace mode, jit/__init__.py
    _substrate_only) into a TorchScript graph; matching calls replay the
    captured program.  Data-dependent control flow is baked at trace
    time -- calls whose tensor ranks/dtypes differ re-trace, and
    non-traceable calls fall back to eager."""

    def __init__(self, fn, input_spec=None, full_graph=False):
        self._fn = fn
        self.input_spec = input_spec
        self._traces = {}
        self._trace_failed = False

    @staticmethod
    def _sig(args):
        parts = []
        for a in args:
            if isinstance(a, torch.Tensor):
                parts.append(("T", tuple(a.shape), str(a.dtype), str(a.device)))
            else:
                return None
        return tuple(parts)

    def __call__(self, *args, **kwargs):
        sig = None if (kwargs or self._trace_failed or not args) \
            else self._sig(args)
        if sig is None:
            return self._fn(*args, **kwargs)
        mod = self._traces.get(sig)
        if mod is None:
            try:
                with _substrate_only():
                    mod = torch.jit.trace(self._fn, args, check_trace=False,
                                          strict=False)
                self._traces[sig] = mod
            except Exception:
                self._trace_failed = True
                return self._fn(*args)
        return mod(*args)

    @property
    def dygraph_function(self):
        return self._fn

    @property
    def concrete_program(self):
        """Last captured TorchScript program (None before first call)."""
        return next(reversed(self._traces.values()), None) \
            if self._traces else None


def to_static(function=None, input_spec=None, full_graph=False, backend=None, **kwargs):
    def deco(fn):
        import functools
        if isinstance(fn, torch.nn.Module):
            return fn  # layers stay eager; train loop unchanged
        wrapped = StaticFunction(fn, input_spec)
        functools.update_wrapper(wrapped, fn, updated=[])
        return wrapped

    if function is not None:
        return deco(function)
    return deco


def not_to_static(fn=None):
    return fn if fn is not None else (lambda f: f)


def ignore_module(modules):
    pass


@contextlib.contextmanager
def _substrate_only():
    """Trace with HIP-native dispatch off AND the op wrappers in trace
    mode (plain substrate expressions; autograd.Function wrappers are not
    TorchScript-exportable)."""
    from .. import framework
    from ..ops import functional as hot
    old = framework.get_flag("FLAGS_use_native_kernels")
    framework.set_flags({"FLAGS_use_native_kernels": False})
    hot._TRACE_SUBSTRATE = True
    try:
        yield
    finally:
        hot._TRACE_SUBSTRATE = False
        framework.set_flags({"FLAGS_use_native_kernels": old})


def _example_inputs(layer, input_spec):
    if input_spec is None:
        raise ValueError(
            "jit.save needs input_spec (paddle.static.InputSpec list or "
            "example tensors) to capture the program")
    ex = []
    from ..static import InputSpec
    from .. import framework
    dev = next((p.device for p in layer.parameters()), torch.device("cpu")) \
        if hasattr(layer, "parameters") else torch.device("cpu")
    dtype0 = next((p.dtype for p in layer.parameters()), torch.float32) \
        if hasattr(layer, "parameters") else torch.float32
    for spec in input_spec:
        if isinstance(spec, torch.Tensor):
            ex.append(spec)
            continue
        shape = [1 if (d is None or d < 0) else d for d in spec.shape]
        dt = framework.convert_dtype(spec.dtype) if isinstance(spec.dtype, str) \
            else spec.dtype
        if dt in (torch.int64, torch.int32):
            ex.append(torch.zeros(shape, dtype=dt, device=dev))
        else:
            ex.append(torch.randn(shape, device=dev).to(
                dt if dt is not None else dtype0))
    return tuple(ex)


def save(layer, path, input_spec=None, **configs):
    """Capture + serialize a Layer (or function) for class-free reload."""
    from ..framework_io import save as fsave
    os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
    target = layer._fn if isinstance(layer, StaticFunction) else layer
    if isinstance(layer, StaticFunction) and input_spec is None:
        input_spec = layer.input_spec
    state = target.state_dict() if hasattr(target, "state_dict") else {}
    fsave(state, path + ".pdiparams")
    meta = {
        "class_module": type(target).__module__,
        "class_name": type(target).__name__,
        "format": "torchscript",
        "program_file": os.path.basename(path) + ".pdscript",
    }
    # capture the program: trace through the substrate
    was_training = getattr(target, "training", False)
    try:
        if hasattr(target, "eval"):
            target.eval()
        ex = _example_inputs(target, input_spec)
        with _substrate_only(), torch.no_grad():
            traced = torch.jit.trace(target, ex, strict=False, check_trace=False)
        torch.jit.save(traced, path + ".pdscript")
    except Exception as e:  # capture failure: keep state-only save
        meta["format"] = "state_only"
        meta["capture_error"] = repr(e)
    finally:
        if was_training and hasattr(target, "train"):
            target.train()
    with open(path + ".pdmodel", "wb") as f:
        pickle.dump(meta, f, protocol=2)


class TranslatedLayer(torch.nn.Module):
    """Executable reloaded program (reference jit/translated_layer.py)."""

    def __init__(self, state, meta, program=None):
        super().__init__()
        self._state = state
        self._meta = meta
        self._program = program

    def forward(self, *args, **kwargs):
        if self._program is None:
            raise RuntimeError(
                "this jit.save archive has no captured program "
                f"(capture_error={self._meta.get('capture_error')!r}); "
                "reconstruct the Layer class and set_state_dict(layer_state())")
        return self._program(*args, **kwargs)

    def layer_state(self):
        return self._state

    def eval(self):
        if self._program is not None:
            self._program.eval()
        return super().eval()



def load(path, **configs):
    from ..framework_io import load as fload
    state = fload(path + ".pdiparams")
    with open(path + ".pdmodel", "rb") as f:
        meta = pickle.load(f)
    program = None
    if meta.get("format") == "torchscript":
        pfile = os.path.join(os.path.dirname(path) or ".",
                             meta["program_file"])
        if os.path.exists(pfile):
            program = torch.jit.load(pfile, map_location="cpu")
    return TranslatedLayer(state, meta, program)


def enable_to_static(flag=True):
    pass


def set_code_level(level=100, also_to_stdout=False):
    """dy2static debug verbosity (reference: jit/dy2static/logging_utils.py);
    the deferred-graph build has no transformed code to print."""


def set_verbosity(level=0, also_to_stdout=False):
    pass
