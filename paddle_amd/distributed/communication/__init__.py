"""paddle.distributed.communication (reference: python/paddle/distributed/
communication/) -- the collective API also re-exported at
paddle.distributed top level."""
from ..collective import (  # noqa: F401
    all_gather,
    all_gather_object,
    all_reduce,
    alltoall,
    alltoall_single,
    barrier,
    broadcast,
    broadcast_object_list,
    irecv,
    isend,
    new_group,
    recv,
    reduce,
    reduce_scatter,
    scatter,
    send,
)
from . import stream  # noqa: F401
