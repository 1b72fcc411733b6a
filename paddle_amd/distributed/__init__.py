"""paddle.distributed parity surface (reference: python/paddle/distributed/)."""
from . import collective  # noqa: F401
from .collective import (  # noqa: F401
    Group,
    P2POp,
    ReduceOp,
    all_gather,
    all_gather_into_tensor,
    all_gather_object,
    all_reduce,
    alltoall,
    alltoall_single,
    barrier,
    batch_isend_irecv,
    broadcast,
    broadcast_object_list,
    destroy_process_group,
    get_group,
    irecv,
    isend,
    is_initialized,
    new_group,
    recv,
    reduce,
    reduce_scatter,
    reduce_scatter_tensor,
    scatter,
    send,
    stream,
)
from .parallel import (  # noqa: F401
    DataParallel,
    ParallelEnv,
    get_rank,
    get_world_size,
    init_parallel_env,
    sync_gradients,
)
from . import fleet  # noqa: F401
from . import sharding  # noqa: F401
from . import checkpoint  # noqa: F401
from . import auto_parallel  # noqa: F401
from .auto_parallel import (  # noqa: F401
    Partial,
    ProcessMesh,
    Replicate,
    Shard,
    dtensor_from_fn,
    reshard,
    shard_layer,
    shard_optimizer,
    shard_tensor,
)


def spawn(func, args=(), nprocs=-1, join=True, daemon=False, **options):
    """paddle.distributed.spawn parity over torch.multiprocessing."""
    import os

    import torch
    import torch.multiprocessing as mp

    if nprocs == -1:
        nprocs = torch.cuda.device_count() if torch.cuda.is_available() else 1

    def _entry(rank, n, fn, fn_args):
        os.environ["RANK"] = str(rank)
        os.environ["LOCAL_RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(n)
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29600")
        fn(*fn_args)

    return mp.start_processes(_entry, args=(nprocs, func, args), nprocs=nprocs,
                              join=join, daemon=daemon, start_method="spawn")

# remaining comm API names
def wait(tensor, group=None, use_calc_stream=True):
    """Block until async work on `tensor` is visible (stream-sync model)."""
    import torch
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    return tensor


def gather(tensor, gather_list=None, dst=0, group=None, sync_op=True):
    from . import collective as C
    return C.gather(tensor, gather_list, dst=dst, group=group)


from . import checkpoint as io  # noqa: E402,F401  (distributed.io save/load tier)
from .auto_parallel import DistAttr, Placement, Strategy  # noqa: E402,F401
from .checkpoint import load_state_dict, save_state_dict  # noqa: E402,F401
from .extras import (  # noqa: E402,F401
    CountFilterEntry,
    DistModel,
    InMemoryDataset,
    ProbabilityEntry,
    QueueDataset,
    ShowClickEntry,
    ParallelMode,
    ReduceType,
    ShardDataloader,
    ShardingStage1,
    ShardingStage2,
    ShardingStage3,
    get_backend,
    gloo_barrier,
    gloo_init_parallel_env,
    gloo_release,
    is_available,
    scatter_object_list,
    shard_dataloader,
    shard_scaler,
    split,
    to_static,
    unshard_dtensor,
)
from . import launch  # noqa: E402,F401
from . import passes  # noqa: E402,F401
