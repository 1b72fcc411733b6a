from . import gpt  # noqa: F401
from .gpt import (  # noqa: F401
    GPTConfig,
    GPTForPretraining,
    GPTModel,
    GPTPretrainingCriterion,
    build_gpt,
)
