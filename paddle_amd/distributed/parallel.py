"""init_parallel_env / ParallelEnv / DataParallel.

Reference: python/paddle/distributed/parallel.py (init_parallel_env,
ParallelEnv:677, DataParallel:219 with 25MB buckets at :377).

MI355X: one process per GPU; RCCL over xGMI via torch.distributed init
(env:// rendezvous -- the launcher sets MASTER_ADDR/PORT/RANK/WORLD_SIZE,
or paddle-style PADDLE_TRAINER_* which we translate).  DP gradient
bucketing defaults to FLAGS_dp_bucket_mb = 128 MB: xGMI links are
153 GB/s point-to-point, so the reference's 25 MB buckets leave launch
latency on the table (SURVEY.md §2.3 reducer note).
"""
from __future__ import annotations

import datetime
import os
from typing import List, Optional

import torch
import torch.distributed as dist

from .. import framework
from . import collective as C


class ParallelEnv:
    @property
    def rank(self):
        return get_rank()

    @property
    def world_size(self):
        return get_world_size()

    @property
    def local_rank(self):
        return int(os.environ.get("LOCAL_RANK",
                                  os.environ.get("PADDLE_RANK_IN_NODE", "0")))

    @property
    def device_id(self):
        return self.local_rank

    @property
    def nranks(self):
        return get_world_size()

    @property
    def dev_id(self):
        return self.local_rank


def _translate_paddle_env():
    """Accept PADDLE_TRAINER_* env (our launcher sets both)."""
    env = os.environ
    if "RANK" not in env and "PADDLE_TRAINER_ID" in env:
        env["RANK"] = env["PADDLE_TRAINER_ID"]
    if "WORLD_SIZE" not in env and "PADDLE_TRAINERS_NUM" in env:
        env["WORLD_SIZE"] = env["PADDLE_TRAINERS_NUM"]
    if "MASTER_ADDR" not in env and "PADDLE_MASTER" in env:
        addr, port = env["PADDLE_MASTER"].rsplit(":", 1)
        env["MASTER_ADDR"] = addr
        env["MASTER_PORT"] = port
    env.setdefault("MASTER_ADDR", "127.0.0.1")
    env.setdefault("MASTER_PORT", "29500")
    env.setdefault("RANK", "0")
    env.setdefault("WORLD_SIZE", "1")


def init_parallel_env(backend=None, timeout=None):
    if C.is_initialized():
        return ParallelEnv()
    _translate_paddle_env()
    backend = backend or C._backend()
    local_rank = int(os.environ.get("LOCAL_RANK",
                                    os.environ.get("PADDLE_RANK_IN_NODE",
                                                   os.environ.get("RANK", "0"))))
    if torch.cuda.is_available():
        ndev = torch.cuda.device_count()
        torch.cuda.set_device(local_rank % ndev)
        framework.set_device(f"gpu:{local_rank % ndev}")
    dist.init_process_group(
        backend=backend,
        timeout=timeout or C._default_timeout,
        device_id=(torch.device("cuda", local_rank % torch.cuda.device_count())
                   if torch.cuda.is_available() and backend == "nccl" else None),
    )
    C._ensure_global_group()
    return ParallelEnv()


def get_rank(group=None):
    if not C.is_initialized():
        return int(os.environ.get("RANK", os.environ.get("PADDLE_TRAINER_ID", "0")))
    if group is not None:
        return group.rank
    return dist.get_rank()


def get_world_size(group=None):
    if not C.is_initialized():
        return int(os.environ.get("WORLD_SIZE", os.environ.get("PADDLE_TRAINERS_NUM", "1")))
    if group is not None:
        return group.nranks
    return dist.get_world_size()


def is_initialized():
    return C.is_initialized()


# ---------------------------------------------------------------------------
# DataParallel: bucketed grad all-reduce overlapped with backward.
# Re-derived EagerReducer (collective/reducer.cc:794,1086): buckets are
# filled in reverse parameter order as grads become ready via
# post-accumulate hooks; a full bucket launches one async all-reduce on
# the comm stream; step() waits for all buckets.
# ---------------------------------------------------------------------------
class _Bucket:
    __slots__ = ("params", "numel", "flat", "work", "ready")

    def __init__(self):
        self.params = []
        self.numel = 0
        self.flat = None
        self.work = None
        self.ready = 0


class DataParallel(torch.nn.Module):
    def __init__(self, layers, strategy=None, comm_buffer_size_MB=None,
                 last_comm_buffer_size=1, find_unused_parameters=False,
                 group=None, process_group=None):
        super().__init__()
        self._layers = layers
        self.group = group
        self.world = get_world_size(group)
        if comm_buffer_size_MB is None:
            comm_buffer_size_MB = framework.get_flag("FLAGS_dp_bucket_mb")
        self.bucket_bytes = int(comm_buffer_size_MB * 1024 * 1024)
        self._grad_sync_enabled = True
        self._final_cb_queued = False
        self._build_buckets()
        self._hooks = []
        self._register_hooks()
        self._sync_params()

    # paddle API passthrough
    def forward(self, *a, **kw):
        return self._layers(*a, **kw)

    def state_dict(self, *a, **kw):
        return self._layers.state_dict(*a, **kw)

    def set_state_dict(self, sd, **kw):
        return self._layers.set_state_dict(sd, **kw)

    def parameters(self, *a, **kw):
        return self._layers.parameters(*a, **kw)

    def named_parameters(self, *a, **kw):
        return self._layers.named_parameters(*a, **kw)

    def _sync_params(self):
        if self.world <= 1:
            return
        with torch.no_grad():
            for p in self._layers.parameters():
                C.broadcast(p.data, src=(self.group.ranks[0] if self.group else 0),
                            group=self.group)

    def _build_buckets(self):
        params = [p for p in self._layers.parameters() if p.requires_grad]
        # reverse order: grads become ready roughly in reverse forward order
        self.buckets: List[_Bucket] = []
        cur = _Bucket()
        for p in reversed(params):
            cur.params.append(p)
            cur.numel += p.numel()
            if cur.numel * p.element_size() >= self.bucket_bytes:
                self.buckets.append(cur)
                cur = _Bucket()
        if cur.params:
            self.buckets.append(cur)
        self._p2bucket = {}
        for b in self.buckets:
            for p in b.params:
                self._p2bucket[id(p)] = b

    def _register_hooks(self):
        if self.world <= 1:
            return
        for p in self._layers.parameters():
            if p.requires_grad:
                h = p.register_post_accumulate_grad_hook(self._on_grad_ready)
                self._hooks.append(h)

    def _on_grad_ready(self, p):
        if not self._grad_sync_enabled or self.world <= 1:
            return
        if not self._final_cb_queued:
            # run _finalize_grads when this backward pass completes (the
            # reference finalizes in EagerReducer at backward end)
            torch.autograd.Variable._execution_engine.queue_callback(self._on_backward_done)
            self._final_cb_queued = True
        b = self._p2bucket[id(p)]
        b.ready += 1
        if b.ready == len(b.params):
            grads = [q.grad for q in b.params]
            flat = torch._utils._flatten_dense_tensors(grads)
            flat.div_(self.world)
            task = C.all_reduce(flat, group=self.group, sync_op=False)
            b.flat, b.work = flat, task
            b.ready = 0

    def _on_backward_done(self):
        self._final_cb_queued = False
        self._finalize_grads()

    def _finalize_grads(self):
        for b in self.buckets:
            if b.work is not None:
                b.work.wait()
                outs = torch._utils._unflatten_dense_tensors(b.flat, [q.grad for q in b.params])
                for q, o in zip(b.params, outs):
                    q.grad.copy_(o)
                b.work, b.flat = None, None
            elif b.ready > 0:
                # partial bucket (grads missing for some params): sync what we have
                b.ready = 0

    def no_sync(self):
        import contextlib

        @contextlib.contextmanager
        def ctx():
            self._grad_sync_enabled = False
            try:
                yield
            finally:
                self._grad_sync_enabled = True

        return ctx()

    def __getattr__(self, name):
        try:
            return super().__getattr__(name)
        except AttributeError:
            return getattr(self._layers, name)


def sync_gradients(params, group=None):
    """fused_allreduce_gradients parity (hybrid_parallel_util.py:249)."""
    world = get_world_size(group)
    if world <= 1:
        return
    grads = [p.grad for p in params if p.grad is not None]
    if not grads:
        return
    flat = torch._utils._flatten_dense_tensors(grads)
    flat.div_(world)
    C.all_reduce(flat, group=group)
    outs = torch._utils._unflatten_dense_tensors(flat, grads)
    for g, o in zip(grads, outs):
        g.copy_(o)
