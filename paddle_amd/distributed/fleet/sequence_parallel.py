"""Megatron-style sequence parallelism (TP-SP).

Reference: fleet/utils/sequence_parallel_utils.py (ScatterOp:85,
GatherOp:97, AllGatherOp:111, ReduceScatterOp:127,
ColumnSequenceParallelLinear:429, RowSequenceParallelLinear,
register_sequence_parallel_allreduce_hooks:192).

Activations are sharded along the sequence dim outside the TP block;
inside, all-gather replaces the TP identity and reduce-scatter replaces
the TP all-reduce (saves hidden-size bandwidth on xGMI).
"""
from __future__ import annotations

import torch

from ...nn.initializer import XavierNormal
from ...nn.layer import Layer
from .. import collective as C


def _mp_group():
    from . import get_hybrid_communicate_group
    hcg = get_hybrid_communicate_group()
    return hcg.get_model_parallel_group() if hcg else None


class _Scatter(torch.autograd.Function):
    """split along seq (dim 0 of [s, b, h] or dim 1 of [b, s, h]; paddle
    uses dim 0 flat) -- here: dim 0."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        w = group.nranks if group else 1
        if w == 1:
            return x
        r = group.rank
        n = x.shape[0] // w
        return x[r * n:(r + 1) * n].contiguous()

    @staticmethod
    def backward(ctx, dy):
        g = ctx.group
        if g is None or g.nranks == 1:
            return dy, None
        outs = [torch.empty_like(dy) for _ in range(g.nranks)]
        C.all_gather(outs, dy.contiguous(), group=g)
        return torch.cat(outs, dim=0), None


class _Gather(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        if group is None or group.nranks == 1:
            return x
        outs = [torch.empty_like(x) for _ in range(group.nranks)]
        C.all_gather(outs, x.contiguous(), group=group)
        return torch.cat(outs, dim=0)

    @staticmethod
    def backward(ctx, dy):
        g = ctx.group
        if g is None or g.nranks == 1:
            return dy, None
        n = dy.shape[0] // g.nranks
        return dy[g.rank * n:(g.rank + 1) * n].contiguous(), None


class _AllGatherFwdReduceScatterBwd(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        if group is None or group.nranks == 1:
            return x
        out_shape = list(x.shape)
        out_shape[0] *= group.nranks
        out = torch.empty(out_shape, dtype=x.dtype, device=x.device)
        C.all_gather_into_tensor(out, x.contiguous(), group=group)
        return out

    @staticmethod
    def backward(ctx, dy):
        g = ctx.group
        if g is None or g.nranks == 1:
            return dy, None
        out_shape = list(dy.shape)
        out_shape[0] //= g.nranks
        out = torch.empty(out_shape, dtype=dy.dtype, device=dy.device)
        C.reduce_scatter_tensor(out, dy.contiguous(), group=g)
        return out, None


class _ReduceScatterFwdAllGatherBwd(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        if group is None or group.nranks == 1:
            return x
        out_shape = list(x.shape)
        out_shape[0] //= group.nranks
        out = torch.empty(out_shape, dtype=x.dtype, device=x.device)
        C.reduce_scatter_tensor(out, x.contiguous(), group=group)
        return out

    @staticmethod
    def backward(ctx, dy):
        g = ctx.group
        if g is None or g.nranks == 1:
            return dy, None
        out_shape = list(dy.shape)
        out_shape[0] *= g.nranks
        out = torch.empty(out_shape, dtype=dy.dtype, device=dy.device)
        C.all_gather_into_tensor(out, dy.contiguous(), group=g)
        return out, None


def scatter(x, group=None):
    return _Scatter.apply(x, group or _mp_group())


def gather(x, group=None):
    return _Gather.apply(x, group or _mp_group())


class ScatterOp(_Scatter):
    pass


class GatherOp(_Gather):
    pass


class AllGatherOp(_AllGatherFwdReduceScatterBwd):
    pass


class ReduceScatterOp(_ReduceScatterFwdAllGatherBwd):
    pass


def mark_as_sequence_parallel_parameter(param):
    param.sequence_parallel = True


def is_sequence_parallel_parameter(param):
    return getattr(param, "sequence_parallel", False)


def register_sequence_parallel_allreduce_hooks(model, accumulation_steps=1,
                                               fuse_sequence_parallel_allreduce=False):
    """LN/bias params outside the linear shards are replicated along mp --
    their grads must be all-reduced over the mp group (reference :192)."""
    group = _mp_group()
    if group is None or group.nranks == 1:
        return
    hooks = []
    for p in model.parameters():
        if is_sequence_parallel_parameter(p) and p.requires_grad:
            def hook(param):
                C.all_reduce(param.grad, group=group)
            hooks.append(p.register_post_accumulate_grad_hook(hook))
    return hooks


class ColumnSequenceParallelLinear(Layer):
    """all-gather(x over seq) -> x @ W_col_shard (reference :429)."""

    def __init__(self, in_features, out_features, weight_attr=None, has_bias=None,
                 gather_output=False, mp_group=None, name=None):
        super().__init__()
        self.group = mp_group or _mp_group()
        w = self.group.nranks if self.group else 1
        assert out_features % w == 0
        self.weight = self.create_parameter([in_features, out_features // w],
                                            attr=weight_attr,
                                            default_initializer=XavierNormal(
                                                fan_in=in_features, fan_out=out_features))
        self.weight.is_distributed = w > 1
        self.bias = None
        if has_bias:
            self.bias = self.create_parameter([out_features // w], is_bias=True)
            self.bias.is_distributed = w > 1

    def forward(self, x):
        x = _AllGatherFwdReduceScatterBwd.apply(x, self.group)
        out = torch.matmul(x, self.weight)
        if self.bias is not None:
            out = out + self.bias
        return out


class RowSequenceParallelLinear(Layer):
    """x_parallel @ W_row_shard -> reduce-scatter(out over seq)."""

    def __init__(self, in_features, out_features, weight_attr=None, has_bias=True,
                 input_is_parallel=True, mp_group=None, name=None):
        super().__init__()
        self.group = mp_group or _mp_group()
        w = self.group.nranks if self.group else 1
        assert in_features % w == 0
        self.weight = self.create_parameter([in_features // w, out_features],
                                            attr=weight_attr,
                                            default_initializer=XavierNormal(
                                                fan_in=in_features, fan_out=out_features))
        self.weight.is_distributed = w > 1
        self.bias = None
        if has_bias:
            self.bias = self.create_parameter([out_features], is_bias=True)
            mark_as_sequence_parallel_parameter(self.bias)

    def forward(self, x):
        out = torch.matmul(x, self.weight)
        out = _ReduceScatterFwdAllGatherBwd.apply(out, self.group)
        if self.bias is not None:
            out = out + self.bias
        return out


class _SPOverlapLinear(torch.autograd.Function):
    """Column-parallel SP linear with comm/GEMM overlap (reference
    sequence_parallel_utils.py:257 SPInnerOverlapLinear).

    fwd: async all-gather of the seq shard overlaps with the LOCAL
    chunk's GEMM (its rows need no communication); the remaining chunks
    run after the gather lands.
    bwd: dx = dout @ W^T needs no comm, so its reduce-scatter is issued
    async and overlaps with the weight-grad GEMM, which itself waits on
    an async re-gather of x (cheaper than saving the gathered copy).
    """

    @staticmethod
    def forward(ctx, xs, weight, bias, group):
        import torch.distributed as dist
        ctx.group = group
        w = group.nranks if group else 1
        if w == 1:
            ctx.save_for_backward(xs, weight)
            ctx.has_bias = bias is not None
            out = torch.matmul(xs, weight)
            return out + bias if bias is not None else out
        pg = group.pg
        r = group.rank
        xs = xs.contiguous()
        full = torch.empty((xs.shape[0] * w,) + tuple(xs.shape[1:]),
                           dtype=xs.dtype, device=xs.device)
        work = dist.all_gather_into_tensor(full, xs, group=pg, async_op=True)
        n = xs.shape[0]
        out = torch.empty(full.shape[:-1] + (weight.shape[-1],),
                          dtype=xs.dtype, device=xs.device)
        torch.matmul(xs, weight, out=out[r * n:(r + 1) * n])  # overlapped
        work.wait()
        for c in range(w):
            if c != r:
                torch.matmul(full[c * n:(c + 1) * n], weight,
                             out=out[c * n:(c + 1) * n])
        ctx.save_for_backward(xs, weight)
        ctx.has_bias = bias is not None
        return out + bias if bias is not None else out

    @staticmethod
    def backward(ctx, dout):
        import torch.distributed as dist
        xs, weight = ctx.saved_tensors
        g = ctx.group
        w = g.nranks if g else 1
        dout = dout.contiguous()
        if w == 1:
            dx = torch.matmul(dout, weight.t())
            dw = torch.matmul(xs.reshape(-1, xs.shape[-1]).t(),
                              dout.reshape(-1, dout.shape[-1]))
            db = dout.reshape(-1, dout.shape[-1]).sum(0) if ctx.has_bias else None
            return dx, dw, db, None
        pg = g.pg
        # re-gather x asynchronously; dx GEMM needs no comm and overlaps
        full = torch.empty((xs.shape[0] * w,) + tuple(xs.shape[1:]),
                           dtype=xs.dtype, device=xs.device)
        gwork = dist.all_gather_into_tensor(full, xs, group=pg, async_op=True)
        dx_full = torch.matmul(dout, weight.t())
        dxs = torch.empty_like(xs)
        swork = dist.reduce_scatter_tensor(dxs, dx_full.contiguous(), group=pg,
                                           async_op=True)
        gwork.wait()
        dw = torch.matmul(full.reshape(-1, full.shape[-1]).t(),
                          dout.reshape(-1, dout.shape[-1]))
        db = dout.reshape(-1, dout.shape[-1]).sum(0) if ctx.has_bias else None
        swork.wait()
        return dxs, dw, db, None


class SPInnerOverlapLinear(Layer):
    """ColumnSequenceParallelLinear with the all-gather overlapped against
    the GEMM chunks (reference :257); identical math, lower exposed comm."""

    def __init__(self, in_features, out_features, weight_attr=None,
                 has_bias=None, mp_group=None, name=None):
        super().__init__()
        self.group = mp_group or _mp_group()
        w = self.group.nranks if self.group else 1
        assert out_features % w == 0
        self.weight = self.create_parameter([in_features, out_features // w],
                                            attr=weight_attr,
                                            default_initializer=XavierNormal(
                                                fan_in=in_features, fan_out=out_features))
        self.weight.is_distributed = w > 1
        self.bias = None
        if has_bias:
            self.bias = self.create_parameter([out_features // w], is_bias=True)
            self.bias.is_distributed = w > 1

    def forward(self, x):
        return _SPOverlapLinear.apply(x, self.weight, self.bias, self.group)
