"""Mixture-of-Experts GPT with expert parallelism over xGMI all-to-all.

Reference parity: python/paddle/incubate/distributed/models/moe/
moe_layer.py:263 (MoELayer; MoEScatter/MoEGather PyLayers over
global_scatter/global_gather) and gate/{gshard,switch}_gate.py.

MI355X design: EP all-to-all is the best case on the 8-GPU xGMI full
mesh (single hop, all 7 links concurrent -- SURVEY.md §5); dispatch
uses a single alltoall_single on a packed [capacity-padded] buffer.
"""
from __future__ import annotations

from dataclasses import dataclass

import torch

from .. import nn
from ..distributed import collective as C
from ..nn.initializer import Normal, _apply_initializer
from ..ops import functional as hot
from .gpt import GPTAttention, GPTConfig, GPTEmbeddings


class _AllToAll(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        if group is None or group.nranks == 1:
            return x
        out = torch.empty_like(x)
        C.alltoall_single(x.contiguous(), out, group=group)
        return out

    @staticmethod
    def backward(ctx, dy):
        if ctx.group is None or ctx.group.nranks == 1:
            return dy, None
        out = torch.empty_like(dy)
        C.alltoall_single(dy.contiguous(), out, group=ctx.group)
        return out, None


class _FusedGate(torch.autograd.Function):
    """Fused softmax+top-k+stats on the gfx950 kernel (csrc/kernels/moe.hip,
    reference gshard_gate.py math); backward re-derives the softmax
    jacobian for the selected probs and the aux-loss me-term."""

    @staticmethod
    def forward(ctx, logits, k, num_experts):
        from .. import _ext
        C = _ext.get_ext()
        topv, topi, me, ce = C.moe_gate_topk(logits.contiguous(), k)
        ctx.save_for_backward(logits, topi, ce)
        ctx.k, ctx.E = k, num_experts
        ctx.mark_non_differentiable(topi)
        aux = (me * ce).sum() * num_experts
        return topv, topi, aux

    @staticmethod
    def backward(ctx, dtopv, _dtopi, daux):
        logits, topi, ce = ctx.saved_tensors
        T = logits.shape[0]
        p = torch.softmax(logits, -1)
        # dL/dp = scatter(dtopv at topi) + daux * E/T * ce  (me = mean p)
        g = torch.zeros_like(p).scatter_(1, topi.long(), dtopv)
        if daux is not None:
            g = g + daux * (ctx.E / T) * ce.unsqueeze(0)
        dlogits = p * (g - (g * p).sum(-1, keepdim=True))
        return dlogits, None, None


class TopKGate(nn.Layer):
    """Switch (k=1) / GShard (k=2) style gate with capacity + aux loss."""

    def __init__(self, hidden_size, num_experts, k=2, capacity_factor=1.25):
        super().__init__()
        self.num_experts = num_experts
        self.k = k
        self.capacity_factor = capacity_factor
        self.wg = nn.Linear(hidden_size, num_experts, bias_attr=False)
        _apply_initializer(Normal(0.0, 0.02), self.wg.weight)

    def forward(self, x):
        from .. import _ext
        # x: [tokens, h]; gate math in fp32 regardless of model dtype
        logits = torch.matmul(x.float(), self.wg.weight.float())
        if (_ext.use_native(x) and self.num_experts <= 64 and self.k <= 4):
            topv, topi, aux = _FusedGate.apply(logits, self.k, self.num_experts)
            return topv, topi.long(), aux
        probs = torch.softmax(logits, -1)
        topv, topi = probs.topk(self.k, dim=-1)            # [T, k]
        # aux load-balance loss (gshard): num_experts * sum(me * ce)
        me = probs.mean(0)
        ce = torch.zeros_like(me).scatter_add_(
            0, topi[:, 0], torch.ones_like(topi[:, 0], dtype=me.dtype))
        ce = ce / x.shape[0]
        aux = (me * ce).sum() * self.num_experts
        return topv, topi, aux


class ExpertMLP(nn.Layer):
    def __init__(self, hidden_size, inter_size):
        super().__init__()
        self.fc1 = nn.Linear(hidden_size, inter_size)
        self.fc2 = nn.Linear(inter_size, hidden_size)

    def forward(self, x):
        return self.fc2(hot.bias_gelu(torch.matmul(x, self.fc1.weight), self.fc1.bias))


class GroupedExperts(nn.Layer):
    """All local experts as stacked parameters -- ONE batched GEMM pair per
    layer (hipBLASLt strided-batched) instead of a Python loop of tiny
    per-expert GEMMs (64 experts x 5 launches was launch-bound on MI355X:
    the per-expert path measured 390 ms/step on the 350M/64e config).
    Weights: w1 [E, h, I], b1 [E, I], w2 [E, I, h], b2 [E, h]."""

    def __init__(self, num_local, hidden_size, inter_size):
        super().__init__()
        import torch as _t
        self.num_local = num_local
        self.w1 = self.create_parameter([num_local, hidden_size, inter_size])
        self.b1 = self.create_parameter([num_local, inter_size], is_bias=True)
        self.w2 = self.create_parameter([num_local, inter_size, hidden_size])
        self.b2 = self.create_parameter([num_local, hidden_size], is_bias=True)
        with _t.no_grad():
            for w, fan_in in ((self.w1, hidden_size), (self.w2, inter_size)):
                w.normal_(0.0, (2.0 / (fan_in + w.shape[-1])) ** 0.5)

    def forward(self, x):
        # x [E, N, h] -> [E, N, h].  Explicit backward: the autograd-
        # generated baddbmm backward issues strided-batched bf16 GEMMs
        # with a TRANSPOSED B operand, which memory-faults in this
        # ROCm/hipBLASLt build at power-of-two token counts (bisect:
        # tools/moe_prof.py); _GroupedFFN materializes the transposes.
        return _GroupedFFN.apply(x, self.w1, self.b1, self.w2, self.b2)


class _GroupedFFN(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w1, b1, w2, b2):
        z = torch.baddbmm(b1.unsqueeze(1), x, w1)
        g = hot.bias_gelu(z, None)
        out = torch.baddbmm(b2.unsqueeze(1), g, w2)
        ctx.save_for_backward(x, z, g, w1, w2)
        return out

    @staticmethod
    def backward(ctx, dy):
        from .. import _ext
        x, z, g, w1, w2 = ctx.saved_tensors
        dyc = dy.contiguous()
        native = _ext.use_native(z) and w1.dtype == torch.bfloat16
        if native:
            # own batched NT MFMA GEMM takes the stored [E, rows, K] weights
            # directly (dg = dy @ w2^T, dx = dz @ w1^T) -- no transB bmm
            # (faults in this hipBLASLt build) and no transpose copies
            C = _ext.get_ext()
            dg = C.gemm_bf16_nt_batched(dyc, w2)
        else:
            dg = torch.bmm(dyc, w2.transpose(1, 2).contiguous())
        dw2 = torch.bmm(g.transpose(1, 2), dyc)
        db2 = dyc.sum(1, dtype=torch.float32).to(w2.dtype)
        if _ext.use_native(z):
            dz = _ext.get_ext().bias_gelu_bwd(dg.contiguous(), z, None)
        else:
            zf = z.float()
            cdf = 0.5 * (1 + torch.erf(zf * 0.7071067811865476))
            pdf = 0.3989422804014327 * torch.exp(-0.5 * zf * zf)
            dz = (dg.float() * (cdf + zf * pdf)).to(z.dtype)
        if native:
            dx = C.gemm_bf16_nt_batched(dz, w1)
        else:
            dx = torch.bmm(dz, w1.transpose(1, 2).contiguous())
        dw1 = torch.bmm(x.transpose(1, 2), dz)
        db1 = dz.sum(1, dtype=torch.float32).to(w1.dtype)
        return dx, dw1, db1, dw2, db2


class MoELayer(nn.Layer):
    """Token dispatch -> EP all-to-all -> local experts -> combine."""

    def __init__(self, hidden_size, inter_size, num_experts, k=2,
                 capacity_factor=2.0, ep_group=None, grouped=True):
        super().__init__()
        self.ep_group = ep_group
        self.ep_size = ep_group.nranks if ep_group else 1
        assert num_experts % self.ep_size == 0
        self.num_experts = num_experts
        self.local_experts = num_experts // self.ep_size
        self.k = k
        self.capacity_factor = capacity_factor
        self.gate = TopKGate(hidden_size, num_experts, k, capacity_factor)
        self.grouped = grouped
        if grouped:
            self.experts = GroupedExperts(self.local_experts, hidden_size, inter_size)
        else:
            self.experts = nn.LayerList([ExpertMLP(hidden_size, inter_size)
                                         for _ in range(self.local_experts)])
        self.aux_loss = None

    def forward(self, x):
        b, s, h = x.shape
        tokens = x.reshape(-1, h)
        T = tokens.shape[0]
        topv, topi, aux = self.gate(tokens)
        self.aux_loss = aux
        cap = int(self.capacity_factor * T * self.k / self.num_experts) + 1
        cap = max(cap, 4)

        # build per-expert capacity-padded dispatch buffer [E, cap, h]
        dispatch = torch.zeros(self.num_experts, cap, h, dtype=x.dtype, device=x.device)
        combine_w = torch.zeros(T, self.k, dtype=x.dtype, device=x.device)
        slot_of = torch.full((T, self.k), -1, dtype=torch.long, device=x.device)
        from .. import _ext
        with torch.no_grad():
            if _ext.use_native(x) and self.num_experts <= 64:
                # one ballot-prefix kernel (assign_pos/number_count parity)
                C = _ext.get_ext()
                slots, _counts = C.moe_assign_slots(
                    topi.to(torch.int32).contiguous(), self.num_experts, cap)
                slot_of = slots.long()
            else:
                for kk in range(self.k):
                    e = topi[:, kk]
                    # position of each token within its expert queue
                    pos = torch.zeros_like(e)
                    order = torch.argsort(e, stable=True)
                    sorted_e = e[order]
                    seg_start = torch.searchsorted(sorted_e, torch.arange(
                        self.num_experts, device=x.device))
                    idx_in_seg = torch.arange(T, device=x.device) - seg_start[sorted_e]
                    pos[order] = idx_in_seg
                    keep = pos < cap
                    slot = e * cap + pos
                    slot_of[:, kk] = torch.where(keep, slot,
                                                 torch.full_like(slot, -1))
        flat_dispatch = dispatch.reshape(-1, h)
        for kk in range(self.k):
            valid = slot_of[:, kk] >= 0
            idx = slot_of[valid, kk]
            flat_dispatch.index_copy_(0, idx, tokens[valid])
            combine_w[:, kk] = torch.where(valid, topv[:, kk].to(x.dtype),
                                           torch.zeros_like(combine_w[:, kk]))

        # EP all-to-all: [E, cap, h] -> experts-local [ep, local_E, cap, h]
        dd = _AllToAll.apply(dispatch.reshape(self.ep_size, self.local_experts, cap, h)
                             .contiguous().view(-1, h), self.ep_group)
        dd = dd.view(self.ep_size, self.local_experts, cap, h)

        if self.grouped:
            # [ep, local_E, cap, h] -> [local_E, ep*cap, h] -> batched GEMMs
            xe = dd.permute(1, 0, 2, 3).reshape(self.local_experts, -1, h)
            expert_out = (self.experts(xe)
                          .view(self.local_experts, self.ep_size, cap, h)
                          .permute(1, 0, 2, 3))  # [ep, local_E, cap, h]
        else:
            outs = []
            for i, expert in enumerate(self.experts):
                outs.append(expert(dd[:, i].reshape(-1, h)).view(self.ep_size, cap, h))
            expert_out = torch.stack(outs, dim=1)  # [ep, local_E, cap, h]

        back = _AllToAll.apply(expert_out.contiguous().view(-1, h), self.ep_group)
        back = back.view(self.num_experts * cap, h)

        out = torch.zeros_like(tokens)
        for kk in range(self.k):
            valid = slot_of[:, kk] >= 0
            idx = slot_of[valid, kk]
            out[valid] += back[idx] * combine_w[valid, kk].unsqueeze(-1)
        return out.view(b, s, h)


class MoEDecoderLayer(nn.Layer):
    def __init__(self, cfg: GPTConfig, num_experts, k=2, ep_group=None):
        super().__init__()
        self.ln1 = nn.LayerNorm(cfg.hidden_size)
        self.attn = GPTAttention(cfg)
        self.ln2 = nn.LayerNorm(cfg.hidden_size)
        self.moe = MoELayer(cfg.hidden_size, cfg.intermediate_size, num_experts,
                            k=k, ep_group=ep_group)

    def forward(self, x):
        h = x + self.attn(self.ln1(x))
        return h + self.moe(self.ln2(h))


class GPTMoEForPretraining(nn.Layer):
    def __init__(self, cfg: GPTConfig, num_experts=64, k=2, ep_group=None,
                 aux_weight=0.01):
        super().__init__()
        self.cfg = cfg
        self.aux_weight = aux_weight
        self.embeddings = GPTEmbeddings(cfg)
        self.layers = nn.LayerList([MoEDecoderLayer(cfg, num_experts, k, ep_group)
                                    for _ in range(cfg.num_layers)])
        self.final_norm = nn.LayerNorm(cfg.hidden_size)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias_attr=False)

    def forward(self, input_ids):
        x = self.embeddings(input_ids)
        for l in self.layers:
            x = l(x)
        return self.lm_head(self.final_norm(x))

    def aux_loss(self):
        total = 0.0
        for l in self.layers:
            if l.moe.aux_loss is not None:
                total = total + l.moe.aux_loss
        return self.aux_weight * total

    def sharding_units(self):
        return [self.embeddings, *self.layers, self.final_norm, self.lm_head]
