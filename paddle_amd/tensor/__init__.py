from . import creation, manipulation, math, linalg, search, random, einsum  # noqa: F401
