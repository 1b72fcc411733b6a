"""paddle.distributed.communication.stream (reference:
communication/stream/__init__.py) -- stream-variant collectives; on this
stack every collective already runs on the comm stream, so these alias
the task-returning forms."""
from ..collective import (  # noqa: F401
    all_gather,
    all_reduce,
    alltoall,
    alltoall_single,
    broadcast,
    recv,
    reduce,
    reduce_scatter,
    scatter,
    send,
)
