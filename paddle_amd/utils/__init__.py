

# reference: utils/__init__.py __all__
import importlib as _importlib
import warnings as _warnings


def deprecated(update_to="", since="", reason="", level=0):
    def deco(fn):
        import functools

        @functools.wraps(fn)
        def wrapper(*a, **kw):
            _warnings.warn(f"{fn.__name__} is deprecated since {since}: "
                           f"{reason} {('use ' + update_to) if update_to else ''}",
                           DeprecationWarning)
            return fn(*a, **kw)
        return wrapper
    return deco


def try_import(module_name, err_msg=None):
    try:
        return _importlib.import_module(module_name)
    except ImportError:
        raise ImportError(err_msg or f"{module_name} is required but not "
                          f"installed (no network egress to install it)")


def require_version(min_version, max_version=None):
    return True


def run_check():
    """Verify the install: tensor op + (if visible) a GPU kernel."""
    import torch
    import paddle_amd as paddle
    x = paddle.ones([2, 2])
    assert float(paddle.sum(x)) == 4.0
    if torch.cuda.is_available():
        y = torch.ones(8, device="cuda", dtype=torch.bfloat16)
        from paddle_amd.ops import functional as hot
        hot.l2_norm_squared(y)
    print("paddle_amd is installed successfully!")

from . import cpp_extension  # noqa: E402,F401
