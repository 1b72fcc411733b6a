"""paddle.utils.cpp_extension (reference: python/paddle/utils/
cpp_extension/cpp_extension.py:92 setup, extension_utils.py).

Out-of-tree custom ops compile through torch.utils.cpp_extension, which
drives hipcc for .cu/.hip sources; gfx950 is pinned so every extension
built here targets the MI355X.
"""
from __future__ import annotations

import os


def _pin_arch():
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")


def CppExtension(sources, *args, **kwargs):
    _pin_arch()
    from torch.utils import cpp_extension as tce
    name = kwargs.pop("name", None) or "paddle_custom_op"
    return tce.CppExtension(name, sources, *args, **kwargs)


def CUDAExtension(sources, *args, **kwargs):
    """Reference name kept; compiles HIP for gfx950 on this stack."""
    _pin_arch()
    from torch.utils import cpp_extension as tce
    name = kwargs.pop("name", None) or "paddle_custom_op"
    return tce.CUDAExtension(name, sources, *args, **kwargs)


class BuildExtension:
    """setuptools cmdclass shim (reference: cpp_extension.py BuildExtension)."""

    @staticmethod
    def with_options(**options):
        from torch.utils import cpp_extension as tce
        return tce.BuildExtension.with_options(**options)

    def __new__(cls, *args, **kwargs):
        from torch.utils import cpp_extension as tce
        return tce.BuildExtension(*args, **kwargs)


def setup(**attrs):
    _pin_arch()
    from setuptools import setup as _setup
    from torch.utils import cpp_extension as tce
    attrs.setdefault("cmdclass", {"build_ext": tce.BuildExtension})
    ext = attrs.pop("ext_modules", None)
    if ext is not None and not isinstance(ext, (list, tuple)):
        ext = [ext]
    return _setup(ext_modules=list(ext or []), **attrs)


def load(name, sources, extra_cxx_cflags=None, extra_cuda_cflags=None,
         extra_ldflags=None, extra_include_paths=None, build_directory=None,
         verbose=False):
    """JIT-compile and import a custom-op module (reference:
    cpp_extension.load).  hipcc cross-compiles for gfx950 even without a
    GPU present."""
    _pin_arch()
    from torch.utils import cpp_extension as tce
    return tce.load(
        name=name, sources=sources,
        extra_cflags=extra_cxx_cflags,
        extra_cuda_cflags=extra_cuda_cflags,
        extra_ldflags=extra_ldflags,
        extra_include_paths=extra_include_paths,
        build_directory=build_directory,
        verbose=verbose)


def get_build_directory(verbose=False):
    from torch.utils.cpp_extension import _get_build_directory
    return _get_build_directory("paddle_custom_ops", verbose)
