"""Tensor-parallel layers + primitives (Megatron MPU style).

Reference: fleet/layers/mpu/mp_layers.py (VocabParallelEmbedding:49,
ColumnParallelLinear:336, RowParallelLinear:543), mp_ops.py
(_c_identity:91, _c_concat:134, _c_split:196, _mp_allreduce:293,
_c_softmax_with_cross_entropy:414).

MI355X: TP collectives run over the mp group on RCCL/xGMI; the
vocab-parallel cross-entropy uses the 3-collective algorithm
(max -> sum -> label-logit gather) from
c_softmax_with_cross_entropy_op.cu, expressed over our comm layer.
"""
from __future__ import annotations

import torch

from ...nn.initializer import Constant, XavierNormal
from ...nn.layer import Layer
from .. import collective as C


# -- autograd-correct TP primitives -----------------------------------------
class _IdentityFwdAllreduceBwd(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return x

    @staticmethod
    def backward(ctx, dy):
        d = dy.contiguous()
        C.all_reduce(d, group=ctx.group)
        return d, None


class _AllreduceFwdIdentityBwd(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        x = x.contiguous()
        C.all_reduce(x, group=group)
        return x

    @staticmethod
    def backward(ctx, dy):
        return dy, None


def _c_identity(x, group):
    return _IdentityFwdAllreduceBwd.apply(x, group)


def _mp_allreduce(x, group):
    return _AllreduceFwdIdentityBwd.apply(x, group)


class _ScatterLastDim(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        w = group.nranks if group else 1
        r = group.rank if group else 0
        n = x.shape[-1] // w
        return x[..., r * n:(r + 1) * n].contiguous()

    @staticmethod
    def backward(ctx, dy):
        g = ctx.group
        w = g.nranks if g else 1
        if w == 1:
            return dy, None
        outs = [torch.empty_like(dy) for _ in range(w)]
        C.all_gather(outs, dy.contiguous(), group=g)
        return torch.cat(outs, dim=-1), None


class _GatherLastDim(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        w = group.nranks if group else 1
        if w == 1:
            return x
        outs = [torch.empty_like(x) for _ in range(w)]
        C.all_gather(outs, x.contiguous(), group=group)
        return torch.cat(outs, dim=-1)

    @staticmethod
    def backward(ctx, dy):
        g = ctx.group
        w = g.nranks if g else 1
        if w == 1:
            return dy, None
        r = g.rank
        n = dy.shape[-1] // w
        return dy[..., r * n:(r + 1) * n].contiguous(), None


def _c_split(x, group):
    return _ScatterLastDim.apply(x, group)


def _c_concat(x, group):
    return _GatherLastDim.apply(x, group)


# -- layers ------------------------------------------------------------------
class VocabParallelEmbedding(Layer):
    def __init__(self, num_embeddings, embedding_dim, weight_attr=None, mp_group=None,
                 name=None):
        super().__init__()
        from . import get_hybrid_communicate_group
        hcg = get_hybrid_communicate_group()
        self.group = mp_group or (hcg.get_model_parallel_group() if hcg else None)
        self.world = self.group.nranks if self.group else 1
        self.rank = self.group.rank if self.group else 0
        assert num_embeddings % self.world == 0
        self.per_part = num_embeddings // self.world
        self.vocab_start = self.rank * self.per_part
        self.num_embeddings = num_embeddings
        self.weight = self.create_parameter([self.per_part, embedding_dim],
                                            attr=weight_attr,
                                            default_initializer=XavierNormal())
        self.weight.is_distributed = self.world > 1

    def forward(self, x):
        if self.world == 1:
            from ...ops import functional as hot
            return hot.embedding(x, self.weight)
        masked = x - self.vocab_start
        oob = (masked < 0) | (masked >= self.per_part)
        masked = masked.clamp(0, self.per_part - 1)
        from ...ops import functional as hot
        out = hot.embedding(masked, self.weight)
        out = out * (~oob).unsqueeze(-1).to(out.dtype)
        return _mp_allreduce(out, self.group)


class ColumnParallelLinear(Layer):
    def __init__(self, in_features, out_features, weight_attr=None, has_bias=None,
                 gather_output=True, fuse_matmul_bias=False, mp_group=None, name=None):
        super().__init__()
        from . import get_hybrid_communicate_group
        hcg = get_hybrid_communicate_group()
        self.group = mp_group or (hcg.get_model_parallel_group() if hcg else None)
        self.world = self.group.nranks if self.group else 1
        assert out_features % self.world == 0
        self.out_per_part = out_features // self.world
        self.gather_output = gather_output
        self.weight = self.create_parameter([in_features, self.out_per_part],
                                            attr=weight_attr,
                                            default_initializer=XavierNormal(fan_in=in_features, fan_out=out_features))
        self.weight.is_distributed = self.world > 1
        self.bias = None
        if has_bias:
            self.bias = self.create_parameter([self.out_per_part], is_bias=True)
            self.bias.is_distributed = self.world > 1

    def forward(self, x):
        x = _c_identity(x, self.group) if self.world > 1 else x
        out = torch.matmul(x, self.weight)
        if self.bias is not None:
            out = out + self.bias
        if self.gather_output and self.world > 1:
            out = _c_concat(out, self.group)
        return out


class RowParallelLinear(Layer):
    def __init__(self, in_features, out_features, weight_attr=None, has_bias=True,
                 input_is_parallel=False, fuse_matmul_bias=False, mp_group=None,
                 name=None):
        super().__init__()
        from . import get_hybrid_communicate_group
        hcg = get_hybrid_communicate_group()
        self.group = mp_group or (hcg.get_model_parallel_group() if hcg else None)
        self.world = self.group.nranks if self.group else 1
        assert in_features % self.world == 0
        self.in_per_part = in_features // self.world
        self.input_is_parallel = input_is_parallel
        self.weight = self.create_parameter([self.in_per_part, out_features],
                                            attr=weight_attr,
                                            default_initializer=XavierNormal(fan_in=in_features, fan_out=out_features))
        self.weight.is_distributed = self.world > 1
        self.bias = None
        if has_bias:
            self.bias = self.create_parameter([out_features], is_bias=True)

    def forward(self, x):
        if not self.input_is_parallel and self.world > 1:
            x = _c_split(x, self.group)
        out = torch.matmul(x, self.weight)
        if self.world > 1:
            out = _mp_allreduce(out, self.group)
        if self.bias is not None:
            out = out + self.bias
        return out


class _VocabParallelCE(torch.autograd.Function):
    """c_softmax_with_cross_entropy: local max ->
    allreduce(max) -> exp-sum -> allreduce(sum) -> gather label logit."""

    @staticmethod
    def forward(ctx, logits, label, group, vocab_start, ignore_index):
        lf = logits.float()
        lmax = lf.amax(-1, keepdim=True)
        if group is not None and group.nranks > 1:
            C.all_reduce(lmax, op=C.ReduceOp.MAX, group=group)
        shifted = lf - lmax
        sumexp = shifted.exp().sum(-1, keepdim=True)
        if group is not None and group.nranks > 1:
            C.all_reduce(sumexp, group=group)
        logsumexp = sumexp.log() + lmax
        per_part = logits.shape[-1]
        local_lab = label - vocab_start
        in_range = (local_lab >= 0) & (local_lab < per_part)
        safe = local_lab.clamp(0, per_part - 1)
        picked = lf.gather(-1, safe.unsqueeze(-1)).squeeze(-1)
        picked = torch.where(in_range, picked, torch.zeros_like(picked))
        if group is not None and group.nranks > 1:
            C.all_reduce(picked, group=group)
        loss = logsumexp.squeeze(-1) - picked
        valid = label != ignore_index
        loss = torch.where(valid, loss, torch.zeros_like(loss))
        ctx.save_for_backward(lf, logsumexp, safe, in_range, valid)
        ctx.ignore_index = ignore_index
        return loss

    @staticmethod
    def backward(ctx, dloss):
        lf, logsumexp, safe, in_range, valid = ctx.saved_tensors
        p = (lf - logsumexp).exp()
        onehot = torch.zeros_like(p)
        onehot.scatter_(-1, safe.unsqueeze(-1),
                        in_range.unsqueeze(-1).to(p.dtype))
        g = (dloss * valid.to(dloss.dtype)).unsqueeze(-1)
        return (g * (p - onehot)).to(lf.dtype), None, None, None, None


class ParallelCrossEntropy(Layer):
    def __init__(self, mp_group=None, name=None, ignore_index=-100):
        super().__init__()
        from . import get_hybrid_communicate_group
        hcg = get_hybrid_communicate_group()
        self.group = mp_group or (hcg.get_model_parallel_group() if hcg else None)
        self.ignore_index = ignore_index
        self.world = self.group.nranks if self.group else 1
        self.rank = self.group.rank if self.group else 0

    def forward(self, input, label):
        per_part = input.shape[-1]
        vocab_start = self.rank * per_part
        if label.dim() == input.dim():
            label = label.squeeze(-1)
        loss = _VocabParallelCE.apply(input, label, self.group, vocab_start,
                                      self.ignore_index)
        return loss.unsqueeze(-1)


def split(x, group=None, axis=-1):
    return _c_split(x, group)
