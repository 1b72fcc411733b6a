"""Group management + collectives (reference: python/paddle/distributed/
communication/, collective.py:194 new_group, group.py:29 Group).

MI355X mapping: torch.distributed with backend "nccl" IS RCCL over xGMI
on ROCm; CPU tests use gloo.  Each Group wraps one torch ProcessGroup;
RCCL comms are created lazily on first collective exactly like the
reference.
"""
from __future__ import annotations

import datetime
import os
from typing import List, Optional

import torch
import torch.distributed as dist

_default_timeout = datetime.timedelta(minutes=30)  # process_group_nccl.h:86


class ReduceOp:
    SUM = dist.ReduceOp.SUM
    MAX = dist.ReduceOp.MAX
    MIN = dist.ReduceOp.MIN
    PROD = dist.ReduceOp.PRODUCT
    AVG = getattr(dist.ReduceOp, "AVG", dist.ReduceOp.SUM)


class Group:
    """paddle Group: ranks + id + underlying torch ProcessGroup."""

    def __init__(self, rank_in_group, gid, ranks, pg=None, name=None):
        self.rank = rank_in_group
        self.id = gid
        self.ranks = ranks
        self.pg = pg
        self.name = name or f"group_{gid}"

    @property
    def nranks(self):
        return len(self.ranks)

    @property
    def world_size(self):
        return len(self.ranks)

    @property
    def process_group(self):
        return self.pg

    def is_member(self):
        return self.rank >= 0

    def get_group_rank(self, rank):
        return self.ranks.index(rank) if rank in self.ranks else -1

    def __repr__(self):
        return f"Group(id={self.id}, ranks={self.ranks})"


_GROUP_COUNT = 0
_global_group: Optional[Group] = None


def _backend():
    if torch.cuda.is_available():
        return "nccl"  # RCCL on ROCm
    return "gloo"


def is_initialized():
    return dist.is_available() and dist.is_initialized()


def _ensure_global_group() -> Group:
    global _global_group
    if _global_group is None:
        if not is_initialized():
            raise RuntimeError("call paddle.distributed.init_parallel_env() first")
        w = dist.get_world_size()
        _global_group = Group(dist.get_rank(), 0, list(range(w)), dist.group.WORLD)
    return _global_group


def _get_group(group) -> Group:
    if group is None:
        return _ensure_global_group()
    return group


def new_group(ranks=None, backend=None, timeout=_default_timeout):
    """collective.py:194 parity.  Must be called by ALL ranks."""
    global _GROUP_COUNT
    _GROUP_COUNT += 1
    gid = _GROUP_COUNT
    if ranks is None:
        ranks = list(range(dist.get_world_size()))
    ranks = sorted(ranks)
    pg = dist.new_group(ranks=ranks, backend=backend or _backend(),
                        timeout=timeout if isinstance(timeout, datetime.timedelta)
                        else datetime.timedelta(milliseconds=timeout))
    me = dist.get_rank()
    rank_in = ranks.index(me) if me in ranks else -1
    return Group(rank_in, gid, ranks, pg)


def get_group(gid=0):
    return _ensure_global_group() if gid == 0 else None


def destroy_process_group(group=None):
    if group is None:
        dist.destroy_process_group()
    else:
        dist.destroy_process_group(group.pg)


# ---------------------------------------------------------------------------
# collectives (communication/*.py parity; sync_op semantics: default True)
# ---------------------------------------------------------------------------
class _Task:
    def __init__(self, work):
        self._work = work

    def wait(self):
        if self._work is not None:
            self._work.wait()

    def is_completed(self):
        return self._work.is_completed() if self._work is not None else True


def all_reduce(tensor, op=ReduceOp.SUM, group=None, sync_op=True):
    from . import comm_check as _cc
    _cc.record("all_reduce", tensor)
    g = _get_group(group)
    work = dist.all_reduce(tensor, op=op, group=g.pg, async_op=not sync_op)
    return _Task(work) if not sync_op else None


def all_gather(tensor_list, tensor, group=None, sync_op=True):
    from . import comm_check as _cc
    _cc.record("all_gather", tensor_list)
    g = _get_group(group)
    if isinstance(tensor_list, list) and len(tensor_list) == 0:
        tensor_list.extend(torch.empty_like(tensor) for _ in range(g.nranks))
    work = dist.all_gather(tensor_list, tensor.contiguous(), group=g.pg,
                           async_op=not sync_op)
    return _Task(work) if not sync_op else None


def all_gather_into_tensor(out, tensor, group=None, sync_op=True):
    from . import comm_check as _cc
    _cc.record("all_gather_into_tensor", out)
    g = _get_group(group)
    work = dist.all_gather_into_tensor(out, tensor.contiguous(), group=g.pg,
                                       async_op=not sync_op)
    return _Task(work) if not sync_op else None


def all_gather_object(object_list, obj, group=None):
    g = _get_group(group)
    if len(object_list) == 0:
        object_list.extend([None] * g.nranks)
    dist.all_gather_object(object_list, obj, group=g.pg)


def reduce_scatter(tensor, tensor_list, op=ReduceOp.SUM, group=None, sync_op=True):
    from . import comm_check as _cc
    _cc.record("reduce_scatter", tensor)
    g = _get_group(group)
    if g.pg is not None and dist.get_backend(g.pg) == "gloo":
        # gloo has no reduce_scatter: emulate (CPU test path only)
        stacked = torch.stack(list(tensor_list))
        dist.all_reduce(stacked, op=op, group=g.pg)
        tensor.copy_(stacked[g.rank])
        return None
    work = dist.reduce_scatter(tensor, list(tensor_list), op=op, group=g.pg,
                               async_op=not sync_op)
    return _Task(work) if not sync_op else None


def reduce_scatter_tensor(out, tensor, op=ReduceOp.SUM, group=None, sync_op=True):
    from . import comm_check as _cc
    _cc.record("reduce_scatter_tensor", out)
    g = _get_group(group)
    if g.pg is not None and dist.get_backend(g.pg) == "gloo":
        t = tensor.clone()
        dist.all_reduce(t, op=op, group=g.pg)
        n = t.numel() // g.nranks
        out.copy_(t.view(-1)[g.rank * n:(g.rank + 1) * n].view(out.shape))
        return None
    work = dist.reduce_scatter_tensor(out, tensor.contiguous(), op=op, group=g.pg,
                                      async_op=not sync_op)
    return _Task(work) if not sync_op else None


def broadcast(tensor, src, group=None, sync_op=True):
    from . import comm_check as _cc
    _cc.record("broadcast", tensor)
    g = _get_group(group)
    work = dist.broadcast(tensor, src=src, group=g.pg, async_op=not sync_op)
    return _Task(work) if not sync_op else None


def broadcast_object_list(object_list, src, group=None):
    g = _get_group(group)
    dist.broadcast_object_list(object_list, src=src, group=g.pg)


def reduce(tensor, dst, op=ReduceOp.SUM, group=None, sync_op=True):
    from . import comm_check as _cc
    _cc.record("reduce", tensor)
    g = _get_group(group)
    work = dist.reduce(tensor, dst=dst, op=op, group=g.pg, async_op=not sync_op)
    return _Task(work) if not sync_op else None


def scatter(tensor, tensor_list=None, src=0, group=None, sync_op=True):
    from . import comm_check as _cc
    _cc.record("scatter", tensor)
    g = _get_group(group)
    work = dist.scatter(tensor, scatter_list=tensor_list, src=src, group=g.pg,
                        async_op=not sync_op)
    return _Task(work) if not sync_op else None


def _is_gloo(g: Group) -> bool:
    return g.pg is not None and dist.get_backend(g.pg) == "gloo"


def alltoall(in_tensor_list, out_tensor_list, group=None, sync_op=True):
    from . import comm_check as _cc
    _cc.record("alltoall", in_tensor_list)
    g = _get_group(group)
    if isinstance(out_tensor_list, list) and len(out_tensor_list) == 0:
        out_tensor_list.extend(torch.empty_like(t) for t in in_tensor_list)
    if _is_gloo(g):
        # gloo has no alltoall (CPU test path): emulate via all_gather of the
        # stacked inputs, then select column my_rank.
        stacked = torch.stack(list(in_tensor_list))
        gathered = [torch.empty_like(stacked) for _ in range(g.nranks)]
        dist.all_gather(gathered, stacked, group=g.pg)
        for src in range(g.nranks):
            out_tensor_list[src].copy_(gathered[src][g.rank])
        return None
    work = dist.all_to_all(out_tensor_list, list(in_tensor_list), group=g.pg,
                           async_op=not sync_op)
    return _Task(work) if not sync_op else None


def alltoall_single(in_tensor, out_tensor, in_split_sizes=None, out_split_sizes=None,
                    group=None, sync_op=True):
    from . import comm_check as _cc
    _cc.record("alltoall_single", in_tensor)
    g = _get_group(group)
    if _is_gloo(g):
        w = g.nranks
        ins = list(in_tensor.chunk(w)) if in_split_sizes is None else \
            list(in_tensor.split(in_split_sizes))
        outs = list(out_tensor.chunk(w)) if out_split_sizes is None else \
            list(out_tensor.split(out_split_sizes))
        tmp = [o.contiguous() for o in outs]
        alltoall(ins, tmp, group=g)
        for o, t in zip(outs, tmp):
            o.copy_(t)
        return None
    work = dist.all_to_all_single(out_tensor, in_tensor, out_split_sizes,
                                  in_split_sizes, group=g.pg, async_op=not sync_op)
    return _Task(work) if not sync_op else None


def send(tensor, dst=0, group=None, sync_op=True):
    from . import comm_check as _cc
    _cc.record("send", tensor)
    g = _get_group(group)
    if sync_op:
        dist.send(tensor, dst=dst, group=g.pg)
        return None
    return _Task(dist.isend(tensor, dst=dst, group=g.pg))


def recv(tensor, src=0, group=None, sync_op=True):
    from . import comm_check as _cc
    _cc.record("recv", tensor)
    g = _get_group(group)
    if sync_op:
        dist.recv(tensor, src=src, group=g.pg)
        return None
    return _Task(dist.irecv(tensor, src=src, group=g.pg))


def isend(tensor, dst, group=None):
    g = _get_group(group)
    return _Task(dist.isend(tensor, dst=dst, group=g.pg))


def irecv(tensor, src, group=None):
    g = _get_group(group)
    return _Task(dist.irecv(tensor, src=src, group=g.pg))


class P2POp:
    def __init__(self, op, tensor, peer, group=None):
        g = _get_group(group)
        torch_op = dist.isend if op in (isend, "isend", send) else dist.irecv
        self._op = dist.P2POp(torch_op, tensor, peer, group=g.pg)


def batch_isend_irecv(p2p_op_list):
    """batch_isend_irecv.py parity: one RCCL group for the batch."""
    works = dist.batch_isend_irecv([p._op for p in p2p_op_list])
    return [_Task(w) for w in works]


def barrier(group=None):
    g = _get_group(group)
    dist.barrier(group=g.pg)


class stream:
    """paddle.distributed.communication.stream variants."""

    @staticmethod
    def all_reduce(tensor, op=ReduceOp.SUM, group=None, sync_op=True,
                   use_calc_stream=False):
        return all_reduce(tensor, op, group, sync_op)

    @staticmethod
    def all_gather(tensor_or_list, tensor, group=None, sync_op=True,
                   use_calc_stream=False):
        if isinstance(tensor_or_list, list):
            return all_gather(tensor_or_list, tensor, group, sync_op)
        return all_gather_into_tensor(tensor_or_list, tensor, group, sync_op)

    @staticmethod
    def reduce_scatter(tensor, tensor_or_list, op=ReduceOp.SUM, group=None,
                       sync_op=True, use_calc_stream=False):
        if isinstance(tensor_or_list, list):
            return reduce_scatter(tensor, tensor_or_list, op, group, sync_op)
        return reduce_scatter_tensor(tensor, tensor_or_list, op, group, sync_op)

    @staticmethod
    def broadcast(tensor, src, group=None, sync_op=True, use_calc_stream=False):
        return broadcast(tensor, src, group, sync_op)

    @staticmethod
    def alltoall(out_list, in_list, group=None, sync_op=True, use_calc_stream=False):
        return alltoall(in_list, out_list, group, sync_op)

    @staticmethod
    def send(tensor, dst, group=None, sync_op=True, use_calc_stream=False):
        return send(tensor, dst, group, sync_op)

    @staticmethod
    def recv(tensor, src, group=None, sync_op=True, use_calc_stream=False):
        return recv(tensor, src, group, sync_op)


def gather(tensor, gather_list=None, dst=0, group=None, sync_op=True):
    """Gather tensors to dst (reference: communication/gather.py).
    gloo/rccl portable: implemented over all_gather (rccl has no native
    gather; the extra traffic is one ring pass)."""
    world = get_world_size(group)
    parts = [torch.empty_like(tensor) for _ in range(world)]
    all_gather(parts, tensor, group=group)
    if get_rank(group) == (dst if group is None else dst) and gather_list is not None:
        for g, p in zip(gather_list, parts):
            g.copy_(p)
    return gather_list
