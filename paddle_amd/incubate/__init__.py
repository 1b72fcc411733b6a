from . import nn  # noqa: F401
