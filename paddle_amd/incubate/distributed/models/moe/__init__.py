"""paddle.incubate.distributed.models.moe parity (reference
incubate/distributed/models/moe/moe_layer.py:263 MoELayer +
gate/{gshard,switch}_gate.py): re-exports the MI355X implementations
from paddle_amd.models.moe."""
from paddle_amd.models.moe import (  # noqa: F401
    ExpertMLP,
    GroupedExperts,
    MoELayer,
    TopKGate,
)

# reference gate-class names
GShardGate = TopKGate
SwitchGate = TopKGate
