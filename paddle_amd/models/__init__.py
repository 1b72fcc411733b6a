from . import bert, gpt, llama, moe  # noqa: F401
from .bert import BertConfig, BertForPretraining, BertModel, build_bert  # noqa: F401
from .gpt import (  # noqa: F401
    GPTConfig,
    GPTForPretraining,
    GPTModel,
    GPTPretrainingCriterion,
    build_gpt,
)
from .llama import (  # noqa: F401
    LlamaConfig,
    LlamaForCausalLM,
    LlamaPretrainingCriterion,
    build_llama,
)
from .moe import GPTMoEForPretraining, MoELayer, TopKGate  # noqa: F401
