"""paddle.static facade (reference: python/paddle/static/, base/executor.py).

Round-1 scope: an executor that runs captured dygraph programs (the
jit.to_static capture path) plus the data/feed API shape.  The full
PIR-style instruction scheduler is tracked in SURVEY.md §2 as phase-7
work; training uses the dygraph path.
"""
from __future__ import annotations

import torch

from .. import framework

_static_mode = False


def enable_static():
    global _static_mode
    _static_mode = True


def disable_static():
    global _static_mode
    _static_mode = False


def in_static_mode():
    return _static_mode


class Program:
    """Minimal Program facade: records a traced callable + params."""

    def __init__(self):
        self.fn = None
        self.feed_names = []
        self.fetch_names = []

    def clone(self, for_test=False):
        import copy
        return copy.copy(self)


_default_main = Program()
_default_startup = Program()


def default_main_program():
    return _default_main


def default_startup_program():
    return _default_startup


def data(name, shape, dtype="float32", lod_level=0):
    class _Var:
        def __init__(self, name, shape, dtype):
            self.name, self.shape, self.dtype = name, shape, dtype

    return _Var(name, shape, dtype)


class Executor:
    """reference: python/paddle/base/executor.py:1234.  Runs a captured
    TracedProgram (from paddle.jit) or a plain callable with feeds."""

    def __init__(self, place=None):
        self.place = place

    def run(self, program=None, feed=None, fetch_list=None, return_numpy=True):
        feed = feed or {}
        prog = program or _default_main
        fn = getattr(prog, "fn", None)
        if fn is None:
            raise RuntimeError(
                "static Executor.run needs a program captured via paddle.jit "
                "(dygraph-first build; see SURVEY.md §7 step 7)")
        tensors = {k: (torch.as_tensor(v) if not isinstance(v, torch.Tensor) else v)
                   for k, v in feed.items()}
        outs = fn(**tensors)
        if not isinstance(outs, (list, tuple)):
            outs = [outs]
        if return_numpy:
            outs = [o.detach().cpu().numpy() if isinstance(o, torch.Tensor) else o for o in outs]
        return list(outs)


class InputSpec:
    def __init__(self, shape=None, dtype="float32", name=None, stop_gradient=True):
        self.shape = shape
        self.dtype = dtype
        self.name = name
        self.stop_gradient = stop_gradient

    @classmethod
    def from_tensor(cls, tensor, name=None):
        return cls(list(tensor.shape), str(tensor.dtype), name)


def save_inference_model(path_prefix, feed_vars, fetch_vars, executor, program=None,
                         **kwargs):
    """Minimal .pdmodel save: stores the jit-scripted program if available."""
    from .. import jit
    raise NotImplementedError(
        "save_inference_model requires the jit capture path; use paddle.jit.save")


def load_inference_model(path_prefix, executor, **kwargs):
    raise NotImplementedError("use paddle.jit.load")


def gradients(targets, inputs, target_gradients=None):
    return torch.autograd.grad(targets, inputs, target_gradients, allow_unused=True)


class amp:
    pass
