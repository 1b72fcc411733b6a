"""fleet.meta_parallel (reference: python/paddle/distributed/fleet/
meta_parallel/__init__.py) -- re-exports the hybrid-parallel building
blocks: TP layers, pipeline schedules, sharding stages, TP RNG tracker."""
from .mpu import (  # noqa: F401
    ColumnParallelLinear,
    ParallelCrossEntropy,
    RowParallelLinear,
    VocabParallelEmbedding,
)
from .pipeline import (  # noqa: F401
    InterleavedPipelineParallel,
    LayerDesc,
    PipelineLayer,
    PipelineParallel,
    SharedLayerDesc,
    VirtualPipelineLayer,
    ZeroBubblePipelineParallel,
    ZeroBubbleInterleavedPipelineParallel,
)
from .random import RNGStatesTracker, get_rng_state_tracker  # noqa: F401
from .sharding import (  # noqa: F401
    DygraphShardingOptimizer,
    GroupShardedStage2,
    GroupShardedStage3,
    ShardedAdamW,
    group_sharded_parallel,
)


class TensorParallel:
    """DataParallel-style wrapper marker for TP models (reference:
    meta_parallel/tensor_parallel.py) -- TP layers here sync via their own
    collectives, so the wrapper is the identity plus broadcast of
    non-TP params at construction."""

    def __new__(cls, layers, hcg=None, **kwargs):
        return layers
