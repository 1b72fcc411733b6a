"""Distributed (sharded) checkpoints -- distcp-style, with resharding.

Reference: python/paddle/distributed/checkpoint/{save_state_dict,
load_state_dict,metadata}.py: every rank writes rank_i.distcp plus a
global metadata file (offsets/shapes); load reshards onto the new
layout.

Resharding model: a checkpoint key may carry `shard_info`
{"global_numel": N, "offset": o} meaning the saved tensor is the flat
slice [o, o+numel) of a global 1-D buffer of N elements (exactly the
ZeRO-3 flat-shard layout of GroupShardedStage3 / ShardedAdamW).  On
load, the target's own shard_info selects the needed range, which is
reassembled from every saved rank whose slice overlaps -- so a job
saved on W ranks restores on W' != W ranks.  Keys without shard_info
are replicated: any rank file containing the key serves it.
"""
from __future__ import annotations

import os
import pickle

import torch

from .. import collective as C
from ..parallel import get_rank, get_world_size


def save_state_dict(state_dict, path, process_group=None, coordinator_rank=0,
                    shard_info=None):
    os.makedirs(path, exist_ok=True)
    rank = get_rank()
    shard_info = shard_info or {}
    cpu_sd = {}
    meta = {}
    for k, v in state_dict.items():
        if isinstance(v, torch.Tensor):
            cpu_sd[k] = v.detach().cpu()
            m = {"shape": list(v.shape), "dtype": str(v.dtype)}
            if k in shard_info:
                m["shard"] = {"global_numel": int(shard_info[k]["global_numel"]),
                              "offset": int(shard_info[k]["offset"]),
                              "numel": int(v.numel())}
            meta[k] = m
        else:
            cpu_sd[k] = v
            meta[k] = {"shape": None, "dtype": None}
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
            pickle.dump({"world_size": len(metas), "ranks": merged}, f, protocol=4)
    if C.is_initialized() and get_world_size() > 1:
        C.barrier()


def _read_meta(path):
    with open(os.path.join(path, "0.metadata"), "rb") as f:
        m = pickle.load(f)
    if "ranks" not in m:  # legacy format
        return {"world_size": len(m), "ranks": m}
    return m


def load_state_dict(state_dict, path, process_group=None, shard_info=None):
    rank = get_rank()
    shard_info = shard_info or {}
    meta = _read_meta(path)
    ranks_meta = meta["ranks"]
    file_cache: dict[str, dict] = {}

    def rank_file(r):
        if r not in file_cache:
            with open(os.path.join(path, f"rank_{r}.distcp"), "rb") as f:
                file_cache[r] = pickle.load(f)
        return file_cache[r]

    for k, v in state_dict.items():
        if k in shard_info and isinstance(v, torch.Tensor):
            # reassemble my flat range from every overlapping saved slice
            need_off = int(shard_info[k]["offset"])
            need_n = v.numel()
            flat = v.view(-1)
            for r, rmeta in ranks_meta.items():
                km = rmeta.get(k)
                if not km or "shard" not in km:
                    continue
                so = km["shard"]["offset"]
                sn = km["shard"]["numel"]
                lo = max(need_off, so)
                hi = min(need_off + need_n, so + sn)
                if lo >= hi:
                    continue
                src = rank_file(r)[k].view(-1)
                flat[lo - need_off:hi - need_off].copy_(
                    src[lo - so:hi - so].to(flat.device, flat.dtype))
            continue
        # replicated key: prefer my own rank file, else any rank that has it
        src_rank = None
        if str(rank) in ranks_meta and k in ranks_meta[str(rank)]:
            src_rank = str(rank)
        else:
            for r, rmeta in ranks_meta.items():
                if k in rmeta:
                    src_rank = r
                    break
        if src_rank is None:
            continue
        lv = rank_file(src_rank)[k]
        if isinstance(v, torch.Tensor) and isinstance(lv, torch.Tensor):
            v.copy_(lv.to(v.device, v.dtype))
        else:
            state_dict[k] = lv
    return state_dict


def merge_sharded(path):
    """Offline utility: reassemble a sharded (flat-slice) checkpoint into
    full per-key tensors without a process group (reference:
    checkpoint/utils.py merge behavior).  Returns {key: tensor} with every
    shard_info key rebuilt at its global size."""
    meta = _read_meta(path)
    ranks_meta = meta["ranks"]
    files = {}

    def rf(r):
        if r not in files:
            with open(os.path.join(path, f"rank_{r}.distcp"), "rb") as f:
                files[r] = pickle.load(f)
        return files[r]

    out = {}
    for r, rmeta in ranks_meta.items():
        for k, km in rmeta.items():
            if km.get("shape") is None:      # non-tensor
                out.setdefault(k, rf(r)[k])
                continue
            if "shard" in km:
                g = km["shard"]["global_numel"]
                if k not in out:
                    out[k] = torch.zeros(g, dtype=rf(r)[k].dtype)
                o = km["shard"]["offset"]
                n = km["shard"]["numel"]
                out[k][o:o + n] = rf(r)[k].view(-1)
            else:
                out.setdefault(k, rf(r)[k])
    return out
