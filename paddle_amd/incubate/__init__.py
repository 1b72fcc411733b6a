from . import asp  # noqa: F401
from . import autograd  # noqa: F401
from . import multiprocessing  # noqa: F401
from . import autotune  # noqa: F401
from . import fp8  # noqa: F401
from . import nn  # noqa: F401

# graph / segment ops + optimizer wrappers (reference: incubate/__init__.py)
import torch as _t

from . import optimizer as optimizer  # noqa: F401
from .optimizer import LookAhead, ModelAverage  # noqa: F401


def segment_sum(data, segment_ids, name=None):
    n = int(segment_ids.max()) + 1 if segment_ids.numel() else 0
    out = _t.zeros(n, *data.shape[1:], dtype=data.dtype, device=data.device)
    return out.index_add_(0, segment_ids.long(), data)


def segment_mean(data, segment_ids, name=None):
    s = segment_sum(data, segment_ids)
    cnt = _t.zeros(s.shape[0], dtype=data.dtype, device=data.device)
    cnt.index_add_(0, segment_ids.long(), _t.ones_like(segment_ids, dtype=data.dtype))
    return s / cnt.clamp(min=1).reshape(-1, *([1] * (data.dim() - 1)))


def _segment_reduce(data, segment_ids, mode):
    n = int(segment_ids.max()) + 1 if segment_ids.numel() else 0
    out = _t.zeros(n, *data.shape[1:], dtype=data.dtype, device=data.device)
    idx = segment_ids.long().reshape(-1, *([1] * (data.dim() - 1))).expand_as(data)
    return out.scatter_reduce_(0, idx, data, mode, include_self=False)


def segment_max(data, segment_ids, name=None):
    return _segment_reduce(data, segment_ids, "amax")


def segment_min(data, segment_ids, name=None):
    return _segment_reduce(data, segment_ids, "amin")


def graph_send_recv(x, src_index, dst_index, pool_type="sum", out_size=None,
                    name=None):
    gathered = x.index_select(0, src_index.long())
    n = out_size or x.shape[0]
    out = _t.zeros(n, *x.shape[1:], dtype=x.dtype, device=x.device)
    if pool_type in ("sum", "mean"):
        out.index_add_(0, dst_index.long(), gathered)
        if pool_type == "mean":
            cnt = _t.zeros(n, dtype=x.dtype, device=x.device)
            cnt.index_add_(0, dst_index.long(),
                           _t.ones_like(dst_index, dtype=x.dtype))
            out = out / cnt.clamp(min=1).reshape(-1, *([1] * (x.dim() - 1)))
    else:
        idx = dst_index.long().reshape(-1, *([1] * (x.dim() - 1))).expand_as(gathered)
        out.scatter_reduce_(0, idx, gathered,
                            {"max": "amax", "min": "amin"}[pool_type],
                            include_self=False)
    return out


def graph_reindex(x, neighbors, count, value_buffer=None, index_buffer=None,
                  flag_buffer=None, name=None):
    nodes = _t.cat([x, neighbors])
    uniq, inv = _t.unique(nodes, return_inverse=True)
    reindex_src = inv[x.numel():]
    # dst expanded by count per source node
    dst = _t.repeat_interleave(inv[:x.numel()], count.long())
    return reindex_src, dst, uniq


def graph_sample_neighbors(row, colptr, input_nodes, sample_size=-1,
                           eids=None, return_eids=False, perm_buffer=None,
                           flag_perm_buffer=False, name=None):
    out_n, out_count = [], []
    for node in input_nodes.tolist():
        lo, hi = int(colptr[node]), int(colptr[node + 1])
        neigh = row[lo:hi]
        if 0 < sample_size < neigh.numel():
            sel = _t.randperm(neigh.numel())[:sample_size]
            neigh = neigh[sel]
        out_n.append(neigh)
        out_count.append(neigh.numel())
    return (_t.cat(out_n) if out_n else row.new_empty(0),
            _t.tensor(out_count, dtype=_t.int64))


def graph_khop_sampler(row, colptr, input_nodes, sample_sizes, sorted_eids=None,
                       return_eids=False, name=None):
    cur = input_nodes
    all_n = [input_nodes]
    for sz in sample_sizes:
        neigh, _ = graph_sample_neighbors(row, colptr, cur, sz)
        all_n.append(neigh)
        cur = _t.unique(neigh)
    edge_src = _t.cat(all_n[1:]) if len(all_n) > 1 else row.new_empty(0)
    uniq = _t.unique(_t.cat(all_n))
    return edge_src, uniq


def softmax_mask_fuse(x, mask, name=None):
    return _t.softmax(x.float() + mask.float(), dim=-1).to(x.dtype)


def softmax_mask_fuse_upper_triangle(x):
    s = x.shape[-1]
    mask = _t.triu(_t.full((s, s), float("-inf"), device=x.device), diagonal=1)
    return _t.softmax(x.float() + mask, dim=-1).to(x.dtype)


def identity_loss(x, reduction="none"):
    if reduction in (0, "sum"):
        return x.sum()
    if reduction in (1, "mean"):
        return x.mean()
    return x


class inference:
    pass

from . import distributed  # noqa: F401,E402
