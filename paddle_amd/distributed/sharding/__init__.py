"""paddle.distributed.sharding namespace (reference:
python/paddle/distributed/sharding/group_sharded.py)."""
from ..fleet.sharding import (  # noqa: F401
    GroupShardedStage2,
    GroupShardedStage3,
    ShardedAdamW,
    group_sharded_parallel,
)


def save_group_sharded_model(model, output, optimizer=None):
    from ...framework_io import save
    import os
    os.makedirs(output, exist_ok=True)
    sd = model.state_dict()
    save(sd, os.path.join(output, "model.pdparams"))
    if optimizer is not None:
        save(optimizer.state_dict(), os.path.join(output, "model.pdopt"))
