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


# ---------------------------------------------------------------------------
# Ring attention (phase 2 of SURVEY §5): KV rotates around the cp ring via
# p2p while each rank's Q stays put; partial outputs merge online with the
# LSE correction.  Causal uses ZIGZAG sharding (rank r owns chunks r and
# 2w-1-r of 2w) so every rank does equal work on the causal triangle.
# xGMI note: ring p2p rides one dedicated 153 GB/s link per neighbor pair.
# ---------------------------------------------------------------------------
def zigzag_split(x, w, r, dim=1):
    """Global sequence -> rank r's zigzag shard (chunks r, 2w-1-r of 2w)."""
    ch = x.chunk(2 * w, dim)
    return torch.cat([ch[r], ch[2 * w - 1 - r]], dim).contiguous()


def zigzag_merge(shards, w, dim=1):
    """List of per-rank shards -> global sequence."""
    ch = [None] * (2 * w)
    for r, s in enumerate(shards):
        lo, hi = s.chunk(2, dim)
        ch[r], ch[2 * w - 1 - r] = lo, hi
    return torch.cat(ch, dim)


def _ring_next_prev(group):
    w, idx = group.nranks, group.rank
    nxt = group.ranks[(idx + 1) % w]
    prv = group.ranks[(idx - 1) % w]
    return nxt, prv


def _ring_rotate(t, group):
    """Send `t` to the next rank, receive the previous rank's (blocking)."""
    import torch.distributed as dist
    nxt, prv = _ring_next_prev(group)
    out = torch.empty_like(t)
    sreq = dist.isend(t.contiguous(), nxt, group=group.pg)
    rreq = dist.irecv(out, prv, group=group.pg)
    sreq.wait()
    rreq.wait()
    return out


def _blk_fwd(q, k, v, scale, causal):
    """One attention block: q,k,v [b,s,h,d] -> (o fp32 [b,s,h,d], lse [b,h,s])."""
    from ...ops import functional as hot
    o, lse = hot.flash_attention(q, k, v, causal=causal, scale=scale,
                                 return_softmax_lse=True)
    return o.float(), lse.float()


def _blk_bwd(do, q, k, v, o, lse, delta, scale, causal):
    """Block backward with GLOBAL lse/delta (the FA2 decomposition: the
    per-row stats of the merged output make per-block grads exact)."""
    from ... import _ext
    qt = q.transpose(1, 2).contiguous()
    kt = k.transpose(1, 2).contiguous()
    vt = v.transpose(1, 2).contiguous()
    if _ext.use_native(q) and q.dtype == torch.bfloat16 and q.shape[-1] in (64, 128):
        C = _ext.get_ext()
        dot = do.to(q.dtype).transpose(1, 2).contiguous()
        ot = o.to(q.dtype).transpose(1, 2).contiguous()
        dq, dk, dv = C.flash_attn_bwd(dot, qt, kt, vt, ot, lse.contiguous(),
                                      None, None, None, scale, causal)
        return (dq.transpose(1, 2), dk.transpose(1, 2), dv.transpose(1, 2))
    # CPU oracle path: dS = P (dP - delta); P from the GLOBAL lse
    qf = qt.float()
    kf = kt.float()
    vf = vt.float()
    dof = do.transpose(1, 2).float()
    s = torch.matmul(qf, kf.transpose(-1, -2)) * scale
    if causal:
        sq, skv = s.shape[-2], s.shape[-1]
        mask = torch.ones(sq, skv, dtype=torch.bool, device=s.device).tril(skv - sq)
        s = s.masked_fill(~mask, float("-inf"))
    p = torch.exp(s - lse.unsqueeze(-1))
    dv_ = torch.matmul(p.transpose(-1, -2), dof)
    dp = torch.matmul(dof, vf.transpose(-1, -2))
    ds = p * (dp - delta.unsqueeze(-1)) * scale
    dq_ = torch.matmul(ds, kf)
    dk_ = torch.matmul(ds.transpose(-1, -2), qf)
    return (dq_.transpose(1, 2).to(q.dtype), dk_.transpose(1, 2).to(q.dtype),
            dv_.transpose(1, 2).to(q.dtype))


def _merge(o_acc, lse_acc, o_blk, lse_blk):
    """Online LSE-corrected merge; o [b,s,h,d] fp32, lse [b,h,s] fp32."""
    lse_new = torch.logaddexp(lse_acc, lse_blk)
    wa = torch.exp(lse_acc - lse_new).transpose(1, 2).unsqueeze(-1)
    wb = torch.exp(lse_blk - lse_new).transpose(1, 2).unsqueeze(-1)
    return o_acc * wa + o_blk * wb, lse_new


class _RingAttn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, group, causal, scale):
        w, r = group.nranks, group.rank
        b, s, h, d = q.shape
        c = s // 2
        kv = torch.stack([k, v]).contiguous()
        o = torch.zeros(b, s, h, d, dtype=torch.float32, device=q.device)
        lse = torch.full((b, h, s), float("-inf"), dtype=torch.float32,
                         device=q.device)
        for t in range(w):
            j = (r - t) % w
            kt, vt = kv[0], kv[1]
            if not causal:
                ob, lb = _blk_fwd(q, kt, vt, scale, False)
                o, lse = _merge(o, lse, ob, lb)
            elif j == r:
                # local shard: zigzag concat order == plain causal order
                ob, lb = _blk_fwd(q, kt, vt, scale, True)
                o, lse = _merge(o, lse, ob, lb)
            elif j < r:
                # KV-low chunk is entirely in the past of both Q chunks
                ob, lb = _blk_fwd(q, kt[:, :c], vt[:, :c], scale, False)
                o, lse = _merge(o, lse, ob, lb)
            else:
                # only the high Q chunk sees this shard (fully)
                ob, lb = _blk_fwd(q[:, c:], kt, vt, scale, False)
                oh, lh = _merge(o[:, c:], lse[:, :, c:], ob, lb)
                o = torch.cat([o[:, :c], oh], dim=1)
                lse = torch.cat([lse[:, :, :c], lh], dim=2)
            if t + 1 < w:
                kv = _ring_rotate(kv, group)
        out = o.to(q.dtype)
        ctx.save_for_backward(q, k, v, out, lse)
        ctx.group, ctx.causal, ctx.scale = group, causal, scale
        return out, lse

    @staticmethod
    def backward(ctx, do, _dlse):
        q, k, v, out, lse = ctx.saved_tensors
        group, causal, scale = ctx.group, ctx.causal, ctx.scale
        w, r = group.nranks, group.rank
        b, s, h, d = q.shape
        c = s // 2
        do = do.contiguous()
        # global per-row delta = rowsum(do * o)  [b, h, s]
        delta = (do.float() * out.float()).sum(-1).transpose(1, 2).contiguous()
        dq = torch.zeros_like(q, dtype=torch.float32)
        # kv and its grad accumulator rotate together; after w rotations
        # the accumulated dk/dv are home at their owning rank
        kv = torch.stack([k, v]).contiguous()
        dkv = torch.zeros(2, b, s, h, d, dtype=torch.float32, device=q.device)
        for t in range(w):
            j = (r - t) % w
            kt, vt = kv[0], kv[1]
            if not causal:
                dq_b, dk_b, dv_b = _blk_bwd(do, q, kt, vt, out, lse, delta,
                                            scale, False)
                dq += dq_b.float()
                dkv[0] += dk_b.float()
                dkv[1] += dv_b.float()
            elif j == r:
                dq_b, dk_b, dv_b = _blk_bwd(do, q, kt, vt, out, lse, delta,
                                            scale, True)
                dq += dq_b.float()
                dkv[0] += dk_b.float()
                dkv[1] += dv_b.float()
            elif j < r:
                dq_b, dk_b, dv_b = _blk_bwd(do, q, kt[:, :c].contiguous(),
                                            vt[:, :c].contiguous(), out, lse,
                                            delta, scale, False)
                dq += dq_b.float()
                dkv[0][:, :c] += dk_b.float()
                dkv[1][:, :c] += dv_b.float()
            else:
                dq_b, dk_b, dv_b = _blk_bwd(do[:, c:].contiguous(),
                                            q[:, c:].contiguous(), kt, vt,
                                            out[:, c:].contiguous(),
                                            lse[:, :, c:].contiguous(),
                                            delta[:, :, c:].contiguous(),
                                            scale, False)
                dq[:, c:] += dq_b.float()
                dkv[0] += dk_b.float()
                dkv[1] += dv_b.float()
            # rotate kv+grad every step (w total) so dkv lands back home
            stacked = torch.cat([kv.float(), dkv], dim=0)
            stacked = _ring_rotate(stacked, group)
            kv = stacked[:2].to(q.dtype).contiguous()
            dkv = stacked[2:].contiguous()
        return (dq.to(q.dtype), dkv[0].to(q.dtype), dkv[1].to(q.dtype),
                None, None, None)


def ring_attention(q, k, v, cp_group=None, causal=True, scale=None):
    """Ring (context-parallel) attention over zigzag seq shards.

    q/k/v: [b, s_local, h, d] where the global sequence is zigzag-sharded
    (`zigzag_split`) over the cp group.  Returns (out [b, s_local, h, d],
    lse [b, h, s_local]).  Reference: new work per SURVEY §5 (ring + LSE
    online merge); reference snapshot has only the sep-axis plumbing.
    """
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if cp_group is None or cp_group.nranks == 1:
        from ...ops import functional as hot
        return hot.flash_attention(q, k, v, causal=causal, scale=scale,
                                   return_softmax_lse=True)
    assert q.shape[1] % 2 == 0, "ring_attention needs an even local seq (zigzag)"
    return _RingAttn.apply(q, k, v, cp_group, causal, scale)
