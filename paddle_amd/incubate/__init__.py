from . import asp  # noqa: F401
from . import autotune  # noqa: F401
from . import fp8  # noqa: F401
from . import nn  # noqa: F401
