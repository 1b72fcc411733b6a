"""Distributed (sharded) checkpoints -- distcp-style.

Reference: python/paddle/distributed/checkpoint/{save_state_dict,
load_state_dict,metadata}.py: every rank writes rank_i.distcp plus a
global metadata file (offsets/shapes); load reshards onto the new
layout.  Round-1 scope: per-rank files + metadata for same-topology
resume (resharding on load: flat-shard concat path for the sharding-3
optimizer)."""
from __future__ import annotations

import os
import pickle

import torch

from .. import collective as C
from ..parallel import get_rank, get_world_size


def save_state_dict(state_dict, path, process_group=None, coordinator_rank=0):
    os.makedirs(path, exist_ok=True)
    rank = get_rank()
    cpu_sd = {}
    meta = {}
    for k, v in state_dict.items():
        if isinstance(v, torch.Tensor):
            cpu_sd[k] = v.detach().cpu()
            meta[k] = {"shape": list(v.shape), "dtype": str(v.dtype)}
        else:
            cpu_sd[k] = v
    with open(os.path.join(path, f"rank_{rank}.distcp"), "wb") as f:
        pickle.dump(cpu_sd, f, protocol=4)
    metas = [None] * get_world_size()
    if C.is_initialized() and get_world_size() > 1:
        C.all_gather_object(metas, {str(rank): meta})
    else:
        metas = [{str(rank): meta}]
    if rank == coordinator_rank:
        merged = {}
        for m in metas:
            if m:
                merged.update(m)
        with open(os.path.join(path, "0.metadata"), "wb") as f:
            pickle.dump(merged, f, protocol=4)
    if C.is_initialized() and get_world_size() > 1:
        C.barrier()


def load_state_dict(state_dict, path, process_group=None):
    rank = get_rank()
    fp = os.path.join(path, f"rank_{rank}.distcp")
    with open(fp, "rb") as f:
        loaded = pickle.load(f)
    for k, v in state_dict.items():
        if k in loaded:
            lv = loaded[k]
            if isinstance(v, torch.Tensor) and isinstance(lv, torch.Tensor):
                v.copy_(lv.to(v.device, v.dtype))
            else:
                state_dict[k] = lv
    return state_dict
