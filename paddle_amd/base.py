"""paddle.base compatibility shim (reference: python/paddle/base/ -- the
legacy `fluid` namespace many downstream scripts still import).

Provides the commonly-touched entry points (framework mode checks,
dygraph guard, Executor, a `core` stub with helpful errors); the legacy
ProgramDesc machinery is replaced by the deferred-graph static module.
"""
from __future__ import annotations

import types as _types

import torch as _torch

from . import framework as _fw
from . import static as _static


class _CoreStub(_types.ModuleType):
    """`paddle.base.core` touchpoints; anything unported raises with a
    pointer to the replacement."""

    def is_compiled_with_cuda(self):
        return _torch.cuda.is_available()  # HIP reports as cuda on ROCm

    def is_compiled_with_rocm(self):
        return True

    def is_compiled_with_xpu(self):
        return False

    def get_cuda_device_count(self):
        return _torch.cuda.device_count() if _torch.cuda.is_available() else 0

    def globals(self):
        return _fw._FLAGS

    def __getattr__(self, name):
        raise AttributeError(
            f"paddle.base.core.{name}: legacy C++ binding not ported; the "
            "MI355X build replaces ProgramDesc/PIR internals (see "
            "paddle_amd.static / paddle_amd.framework)")


core = _CoreStub("paddle_amd.base.core")


class framework:
    @staticmethod
    def in_dygraph_mode():
        return not _static._static_mode

    @staticmethod
    def in_pir_mode():
        return False

    Program = _static.Program
    Variable = _static.Var
    default_main_program = staticmethod(_static.default_main_program)
    default_startup_program = staticmethod(_static.default_startup_program)


class dygraph:
    class guard:
        def __init__(self, place=None):
            pass

        def __enter__(self):
            _static.disable_static()
            return self

        def __exit__(self, *a):
            return False

    @staticmethod
    def to_variable(value, name=None, zero_copy=None):
        return _torch.as_tensor(value)


class executor:
    Executor = _static.Executor
    global_scope = staticmethod(_static.global_scope)
    scope_guard = staticmethod(_static.scope_guard)


Executor = _static.Executor
CPUPlace = _fw.CPUPlace
CUDAPlace = _fw.CUDAPlace if hasattr(_fw, "CUDAPlace") else _fw.GPUPlace
