"""paddle.incubate.distributed (reference import-path parity; the MoE
model family lives here in the reference tree)."""
from . import models  # noqa: F401
