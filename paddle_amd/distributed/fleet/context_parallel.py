"""Context parallelism for long sequences -- NEW work beyond the
reference snapshot (the reference has only the `sep` axis
plumbing; ring/Ulysses live downstream in PaddleNLP).

Ulysses-style attention: activations sharded over seq; before attention
an all-to-all swaps seq-shards for head-shards (each rank gets ALL
tokens of H/cp heads), the flash-attention kernel runs on full
sequences, and a second all-to-all swaps back.  On the 8-GPU xGMI full
mesh all-to-all is single-hop over all 7 links -- the cheapest
collective on this topology.
"""
from __future__ import annotations

import math

import torch

from .. import collective as C


class _SeqHeadAllToAll(torch.autograd.Function):
    """[b, s_local, h, d] -> [b, s_full, h_local, d] (scatter heads,
    gather seq).  Inverse when `inverse`=True."""

    @staticmethod
    def forward(ctx, x, group, inverse):
        ctx.group = group
        ctx.inverse = inverse
        return _a2a(x, group, inverse)

    @staticmethod
    def backward(ctx, dy):
        return _a2a(dy.contiguous(), ctx.group, not ctx.inverse), None, None


def _a2a(x, group, inverse):
    w = group.nranks if group else 1
    if w == 1:
        return x
    b = x.shape[0]
    if not inverse:
        # x: [b, s_loc, h, d] -> out [b, s_loc*w, h/w, d]
        b, s, h, d = x.shape
        assert h % w == 0
        # reorganize to [w, b, s, h/w, d] chunks by head
        xs = x.reshape(b, s, w, h // w, d).permute(2, 0, 1, 3, 4).contiguous()
        out = torch.empty_like(xs)
        C.alltoall_single(xs.view(w, -1), out.view(w, -1), group=group)
        # out[w_src] = tokens of MY heads from rank w_src: cat over seq
        return out.permute(1, 0, 2, 3, 4).reshape(b, w * s, h // w, d)
    else:
        # x: [b, s_full, h_loc, d] -> [b, s_full/w, h_loc*w, d]
        b, s, h, d = x.shape
        assert s % w == 0
        xs = x.reshape(b, w, s // w, h, d).permute(1, 0, 2, 3, 4).contiguous()
        out = torch.empty_like(xs)
        C.alltoall_single(xs.view(w, -1), out.view(w, -1), group=group)
        return out.permute(1, 2, 0, 3, 4).reshape(b, s // w, w * h, d)


def ulysses_attention(q, k, v, cp_group=None, causal=True, scale=None):
    """q/k/v: [b, s_local, h, d] sharded over seq on the cp group.
    Returns [b, s_local, h, d]."""
    from ...ops import functional as hot
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if cp_group is not None and cp_group.nranks > 1:
        q = _SeqHeadAllToAll.apply(q, cp_group, False)
        k = _SeqHeadAllToAll.apply(k, cp_group, False)
        v = _SeqHeadAllToAll.apply(v, cp_group, False)
    out, _ = hot.flash_attention(q, k, v, causal=causal, scale=scale)
    if cp_group is not None and cp_group.nranks > 1:
        out = _SeqHeadAllToAll.apply(out.contiguous(), cp_group, True)
    return out
