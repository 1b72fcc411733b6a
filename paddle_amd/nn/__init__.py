"""paddle.nn parity surface (reference: python/paddle/nn/__init__.py)."""
from . import functional  # noqa: F401
from . import initializer  # noqa: F401
from . import utils  # noqa: F401
from .layer import Layer, LayerList, ParameterList, Sequential  # noqa: F401
from .common import (  # noqa: F401
    Dropout,
    Dropout2D,
    Embedding,
    Flatten,
    Identity,
    LayerNorm,
    Linear,
    Pad2D,
    RMSNorm,
    Upsample,
)
from .activation import (  # noqa: F401
    ELU,
    GELU,
    Hardsigmoid,
    Hardswish,
    LeakyReLU,
    LogSoftmax,
    Mish,
    ReLU,
    ReLU6,
    Sigmoid,
    SiLU,
    Softmax,
    Softplus,
    Swish,
    Tanh,
)
from .conv import (  # noqa: F401
    AdaptiveAvgPool2D,
    AvgPool2D,
    BatchNorm,
    BatchNorm1D,
    BatchNorm2D,
    Conv1D,
    Conv2D,
    Conv2DTranspose,
    GroupNorm,
    MaxPool2D,
    SyncBatchNorm,
)
from .loss import (  # noqa: F401
    BCELoss,
    BCEWithLogitsLoss,
    CrossEntropyLoss,
    KLDivLoss,
    L1Loss,
    MSELoss,
    NLLLoss,
    SmoothL1Loss,
)
from .rnn import (  # noqa: F401
    GRU,
    GRUCell,
    LSTM,
    LSTMCell,
    SimpleRNN,
    SimpleRNNCell,
)
from .torch_wrap import *  # noqa: F401,F403
from .transformer import (  # noqa: F401
    MultiHeadAttention,
    Transformer,
    TransformerDecoder,
    TransformerDecoderLayer,
    TransformerEncoder,
    TransformerEncoderLayer,
)


class ClipGradByGlobalNorm:
    """paddle.nn.ClipGradByGlobalNorm parity -- consumed by optimizers."""

    def __init__(self, clip_norm=1.0, group_name="default_group", auto_skip_clip=False):
        self.clip_norm = clip_norm


class ClipGradByNorm:
    def __init__(self, clip_norm=1.0):
        self.clip_norm = clip_norm


class ClipGradByValue:
    def __init__(self, max, min=None):
        self.max = max
        self.min = -max if min is None else min
