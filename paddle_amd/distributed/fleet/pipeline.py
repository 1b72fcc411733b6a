"""Pipeline parallelism: PipelineLayer + 1F1B schedule over RCCL p2p.

Reference: fleet/meta_parallel/parallel_layers/pp_layers.py (LayerDesc:56,
SharedLayerDesc:76, PipelineLayer:257) and pipeline_parallel.py
(forward_backward_pipeline:575 -- 1F1B), pp_utils/p2p_communication.py
(SendRecvMeta:52 shape handshake).

MI355X: stage boundaries are single xGMI links (153 GB/s point-to-point,
SURVEY.md §5) so activations move cheaply; p2p tensors are sent with a
one-time [ndim, shape..., dtype] int64 header, cached afterwards.
"""
from __future__ import annotations

from typing import Callable, List, Optional

import torch

from .. import collective as C

_DTYPE_IDS = [torch.float32, torch.bfloat16, torch.float16, torch.int64, torch.int32]


class LayerDesc:
    def __init__(self, layer_cls, *args, **kwargs):
        self.layer_cls = layer_cls
        self.args = args
        self.kwargs = kwargs

    def build_layer(self):
        return self.layer_cls(*self.args, **self.kwargs)


class SharedLayerDesc(LayerDesc):
    """Tied layers (e.g. embedding/lm-head) -- built on every owning stage;
    grads all-reduced across the shared group after each step."""

    def __init__(self, key, layer_cls, *args, forward_func=None, shared_weight_attr="weight",
                 **kwargs):
        super().__init__(layer_cls, *args, **kwargs)
        self.layer_name = key
        self.forward_func = forward_func
        self.shared_weight_attr = shared_weight_attr


class PipelineLayer(torch.nn.Module):
    """Build from a list of LayerDesc; each PP stage owns a contiguous
    segment (uniform segmentation; param-count mode later)."""

    def __init__(self, layers, num_stages=None, topology=None, loss_fn=None,
                 seg_method="uniform", recompute_interval=0, recompute_ctx=None,
                 hcg=None):
        super().__init__()
        from . import get_hybrid_communicate_group
        self._hcg = hcg or get_hybrid_communicate_group()
        self.num_stages = num_stages or (self._hcg.get_pipe_parallel_world_size()
                                         if self._hcg else 1)
        self.stage_id = self._hcg.get_pipe_parallel_rank() if self._hcg else 0
        self.loss_fn = loss_fn
        self.descs = list(layers)
        self._recompute_interval = recompute_interval
        n = len(self.descs)
        per = [n // self.num_stages + (1 if i < n % self.num_stages else 0)
               for i in range(self.num_stages)]
        starts = [sum(per[:i]) for i in range(self.num_stages + 1)]
        self.seg_starts = starts
        self.local_start = starts[self.stage_id]
        self.local_end = starts[self.stage_id + 1]
        self.run_funcs = []
        self.shared_layers = {}
        mods = []
        for i in range(self.local_start, self.local_end):
            d = self.descs[i]
            if isinstance(d, SharedLayerDesc):
                if d.layer_name not in self.shared_layers:
                    self.shared_layers[d.layer_name] = d.build_layer()
                layer = self.shared_layers[d.layer_name]
                fwd = d.forward_func
                self.run_funcs.append((layer, fwd))
                mods.append(layer)
            elif isinstance(d, LayerDesc):
                layer = d.build_layer()
                self.run_funcs.append((layer, None))
                mods.append(layer)
            else:
                self.run_funcs.append((d, None))
                if isinstance(d, torch.nn.Module):
                    mods.append(d)
        self.local_layers = torch.nn.ModuleList(mods)

    def forward(self, x):
        from .recompute import recompute
        for idx, (layer, fwd) in enumerate(self.run_funcs):
            use_rc = (self._recompute_interval > 0 and self.training and
                      torch.is_grad_enabled() and
                      idx % self._recompute_interval == 0 and
                      isinstance(x, torch.Tensor) and x.requires_grad)
            fn = (lambda inp, l=layer, f=fwd: f(l, inp) if f else l(inp))
            x = recompute(fn, x) if use_rc else fn(x)
        return x

    def allreduce_shared_weight_gradients(self):
        # tied weights: all-reduce grads across pp group (stages sharing key)
        if not self.shared_layers or self._hcg is None:
            return
        pp_group = self._hcg.get_pipe_parallel_group()
        if pp_group is None:
            return
        for key, layer in self.shared_layers.items():
            for p in layer.parameters():
                if p.grad is not None:
                    C.all_reduce(p.grad, group=pp_group)

    def sharding_units(self):
        return list(self.local_layers)


class _P2P:
    """shape/dtype handshake + cached send/recv (p2p_communication.py)."""

    def __init__(self, hcg):
        self.hcg = hcg
        self.group = hcg.get_pipe_parallel_group()
        self.sent_meta = False
        self.recv_shape = None
        self.recv_dtype = None
        self._sent_meta_to = set()   # peer-aware path (ZB-V placement)
        self._recv_meta_from = {}

    def _send_meta(self, t, dst):
        hdr = torch.tensor([t.dim(), *t.shape, _DTYPE_IDS.index(t.dtype)],
                           dtype=torch.int64, device=t.device)
        n = torch.tensor([hdr.numel()], dtype=torch.int64, device=t.device)
        C.send(n, dst=dst, group=self.group)
        C.send(hdr, dst=dst, group=self.group)

    def _recv_meta(self, src, device):
        n = torch.zeros(1, dtype=torch.int64, device=device)
        C.recv(n, src=src, group=self.group)
        hdr = torch.zeros(int(n.item()), dtype=torch.int64, device=device)
        C.recv(hdr, src=src, group=self.group)
        vals = hdr.tolist()
        ndim = vals[0]
        shape = vals[1:1 + ndim]
        dtype = _DTYPE_IDS[vals[1 + ndim]]
        return shape, dtype

    def send_forward(self, t, dst=None):
        explicit = dst is not None
        if dst is None:
            dst = self.hcg.get_p2p_next_rank()
        if explicit:
            # ZB-V placement: forward peers vary by stage (up-leg -> next,
            # down-leg -> prev); meta handshake cached per peer
            if dst not in self._sent_meta_to:
                self._send_meta(t, dst)
                self._sent_meta_to.add(dst)
        elif not self.sent_meta:
            self._send_meta(t, dst)
            self.sent_meta = True
        C.send(t.contiguous(), dst=dst, group=self.group)

    def recv_forward(self, device, src=None):
        explicit = src is not None
        if src is None:
            src = self.hcg.get_p2p_prev_rank()
        if explicit:
            if src not in self._recv_meta_from:
                self._recv_meta_from[src] = self._recv_meta(src, device)
            shape, dtype = self._recv_meta_from[src]
        else:
            if self.recv_shape is None:
                self.recv_shape, self.recv_dtype = self._recv_meta(src, device)
            shape, dtype = self.recv_shape, self.recv_dtype
        t = torch.zeros(shape, dtype=dtype, device=device)
        C.recv(t, src=src, group=self.group)
        return t

    def send_backward(self, g, dst=None):
        if dst is None:
            dst = self.hcg.get_p2p_prev_rank()
        C.send(g.contiguous(), dst=dst, group=self.group)

    def recv_backward(self, like, src=None):
        if src is None:
            src = self.hcg.get_p2p_next_rank()
        g = torch.zeros_like(like)
        C.recv(g, src=src, group=self.group)
        return g

    # combined pairs (deadlock-free: post isend+irecv, then wait both --
    # the reference's batched _batched_p2p_ops, p2p_communication.py:327)
    def send_forward_recv_backward(self, out, grad_like):
        nxt = self.hcg.get_p2p_next_rank()
        g = torch.zeros_like(grad_like)
        w1 = C.isend(out.contiguous(), nxt, group=self.group)
        w2 = C.irecv(g, nxt, group=self.group)
        w1.wait()
        w2.wait()
        return g

    def send_backward_recv_forward(self, grad_in, device):
        prv = self.hcg.get_p2p_prev_rank()
        x = torch.zeros(self.recv_shape, dtype=self.recv_dtype, device=device)
        w1 = C.isend(grad_in.contiguous(), prv, group=self.group)
        w2 = C.irecv(x, prv, group=self.group)
        w1.wait()
        w2.wait()
        return x


class PipelineParallel(torch.nn.Module):
    """1F1B schedule (pipeline_parallel.py:575)."""

    def __init__(self, layers: PipelineLayer, hcg, strategy=None):
        super().__init__()
        self._layers = layers
        self._hcg = hcg
        cfg = strategy.pipeline_configs if strategy is not None else {}
        self.accumulate_steps = cfg.get("accumulate_steps", 1)
        self.micro_batch_size = cfg.get("micro_batch_size", 1)
        self.stage_id = hcg.get_pipe_parallel_rank()
        self.num_stages = hcg.get_pipe_parallel_world_size()
        self.p2p = _P2P(hcg)
        self.is_first = hcg.is_first_stage()
        self.is_last = hcg.is_last_stage()

    def forward(self, *a, **kw):
        return self._layers(*a, **kw)

    def _split_micro(self, data):
        x, y = data
        mbs = []
        n = self.accumulate_steps
        xs = x.chunk(n) if x is not None else [None] * n
        ys = y.chunk(n) if y is not None else [None] * n
        return list(zip(xs, ys))

    def train_batch(self, data, optimizer, lr_scheduler=None, scaler=None):
        assert self.num_stages > 1
        import collections
        micro = self._split_micro(data)
        n_micro = len(micro)
        warmup = min(self.num_stages - self.stage_id - 1, n_micro)
        steady = n_micro - warmup
        dev = (torch.device("cuda", torch.cuda.current_device())
               if torch.cuda.is_available() else torch.device("cpu"))

        losses = []
        pending = collections.deque()  # (inp, out) awaiting backward, FIFO
        fwd_i = 0

        def get_input(i):
            if self.is_first:
                return micro[i][0]
            inp = self.p2p.recv_forward(dev)
            inp.requires_grad_(True)
            return inp

        def run_fwd(inp, i):
            out = self._layers(inp)
            if self.is_last:
                loss = self._layers.loss_fn(out, micro[i][1]) / n_micro
                losses.append(loss.detach())
                out = loss
            return out

        def run_bwd(binp, bout, grad_out):
            if self.is_last:
                bout.backward()
            else:
                bout.backward(gradient=grad_out)
            return binp.grad if (binp is not None and not self.is_first) else None

        # warmup forwards (never on the last stage: its warmup == 0)
        for _ in range(warmup):
            inp = get_input(fwd_i)
            out = run_fwd(inp, fwd_i)
            self.p2p.send_forward(out)
            pending.append((None if self.is_first else inp, out))
            fwd_i += 1
        # steady 1F1B
        inp = get_input(fwd_i) if steady > 0 else None
        for k in range(steady):
            out = run_fwd(inp, fwd_i)
            fwd_i += 1
            pending.append((None if self.is_first else inp, out))
            binp, bout = pending.popleft()
            if self.is_last:
                grad_out = None
            else:
                grad_out = self.p2p.send_forward_recv_backward(out, bout)
            grad_in = run_bwd(binp, bout, grad_out)
            last_k = k == steady - 1
            if self.is_first:
                if not last_k:
                    inp = get_input(fwd_i)
            elif last_k:
                self.p2p.send_backward(grad_in)
            else:
                inp = self.p2p.send_backward_recv_forward(grad_in, dev)
                inp.requires_grad_(True)
        # cooldown: drain deferred backwards
        while pending:
            binp, bout = pending.popleft()
            grad_out = None if self.is_last else self.p2p.recv_backward(bout)
            grad_in = run_bwd(binp, bout, grad_out)
            if not self.is_first:
                self.p2p.send_backward(grad_in)

        self._layers.allreduce_shared_weight_gradients()
        if optimizer is not None:
            if scaler is not None:
                scaler.step(optimizer)
                scaler.update()
            else:
                optimizer.step()
            optimizer.clear_grad()
            if lr_scheduler is not None:
                lr_scheduler.step()
        if self.is_last:
            total = torch.stack(losses).sum()
            return total
        return torch.zeros(1, device=dev)

    def eval_batch(self, data, compute_loss=True):
        with torch.no_grad():
            micro = self._split_micro(data)
            dev = (torch.device("cuda", torch.cuda.current_device())
                   if torch.cuda.is_available() else torch.device("cpu"))
            losses = []
            for x_mb, y_mb in micro:
                inp = x_mb if self.is_first else self.p2p.recv_forward(dev)
                out = self._layers(inp)
                if self.is_last:
                    if compute_loss:
                        losses.append(self._layers.loss_fn(out, y_mb))
                else:
                    self.p2p.send_forward(out)
            if self.is_last and losses:
                return torch.stack(losses).mean()
            return torch.zeros(1, device=dev)

    def parameters(self, *a, **kw):
        return self._layers.parameters(*a, **kw)

    def state_dict(self, *a, **kw):
        return self._layers.state_dict(*a, **kw)

    def __getattr__(self, name):
        try:
            return super().__getattr__(name)
        except AttributeError:
            return getattr(self._layers, name)


class VirtualPipelineLayer(torch.nn.Module):
    """Interleaved virtual-pipeline partitioning (reference: pp_layers.py:98
    num_virtual_pipeline_stage -- model split into pp*v chunks; rank r owns
    chunks r, r+pp, r+2*pp, ...).

    Global stage g runs chunk g; its owner is rank g % pp, so every
    activation hop goes to the NEXT pp rank (one xGMI link) and every
    gradient hop to the previous one -- the wrap-around between chunk
    generations lands on the same links, which keeps the relay schedule
    below deadlock-free over blocking p2p.
    """

    def __init__(self, layers, topology=None, loss_fn=None,
                 num_virtual_pipeline_stages=2, recompute_interval=0, hcg=None,
                 placement="interleaved"):
        super().__init__()
        from . import get_hybrid_communicate_group
        self._hcg = hcg or get_hybrid_communicate_group()
        self.pp = self._hcg.get_pipe_parallel_world_size()
        self.rank = self._hcg.get_pipe_parallel_rank()
        self.v = num_virtual_pipeline_stages
        self.loss_fn = loss_fn
        self.descs = list(layers)
        total = self.pp * self.v
        n = len(self.descs)
        per = [n // total + (1 if i < n % total else 0) for i in range(total)]
        starts = [sum(per[:i]) for i in range(total + 1)]
        # stage->rank placement: "interleaved" (g % pp) or "zbv" (ZB-V
        # paper V-shape, v == 2: rank r owns chunks r and 2*pp-1-r, so
        # the last stage's B starts on the rank that also holds stage 0
        # -- the geometry that removes the H1 schedule's tail bubble)
        if placement == "zbv":
            assert self.v == 2, "ZB-V placement is defined for v == 2"
            self.owner = [g if g < self.pp else 2 * self.pp - 1 - g
                          for g in range(total)]
        else:
            self.owner = [g % self.pp for g in range(total)]
        self.placement = placement
        self.my_stages = [g for g in range(total) if self.owner[g] == self.rank]
        self.total_stages = total
        self.chunks = torch.nn.ModuleList()
        self._chunk_funcs = {}
        for g in self.my_stages:
            funcs, mods = [], []
            for i in range(starts[g], starts[g + 1]):
                d = self.descs[i]
                if isinstance(d, LayerDesc):
                    layer = d.build_layer()
                    funcs.append((layer, getattr(d, "forward_func", None)))
                    mods.append(layer)
                else:
                    funcs.append((d, None))
                    if isinstance(d, torch.nn.Module):
                        mods.append(d)
            self.chunks.append(torch.nn.ModuleList(mods))
            self._chunk_funcs[g] = funcs

    def run_chunk(self, g, x):
        for layer, fwd in self._chunk_funcs[g]:
            x = fwd(layer, x) if fwd else layer(x)
        return x

    def sharding_units(self):
        return [m for c in self.chunks for m in c]


class InterleavedPipelineParallel(torch.nn.Module):
    """Virtual-pipeline schedule (reference: pipeline_parallel.py
    forward_backward_pipeline with interleave, :359).

    Schedule here is depth-first relay: microbatch-major, global stages in
    ascending order for forward then descending for backward.  Numerically
    identical to interleaved 1F1B (same chunk partitioning, same grad
    accumulation); activation memory is O(n_micro x v) instead of the
    1F1B working set -- acceptable at 288 GB HBM3E per GPU, and trivially
    deadlock-free.  The overlap-optimised interleave is a follow-up.
    """

    def __init__(self, layers: VirtualPipelineLayer, hcg, strategy=None):
        super().__init__()
        self._layers = layers
        self._hcg = hcg
        cfg = strategy.pipeline_configs if strategy is not None else {}
        self.accumulate_steps = cfg.get("accumulate_steps", 1)
        self.p2p = _P2P(hcg)
        self.pp = layers.pp
        self.rank = layers.rank

    def _split_micro(self, data):
        x, y = data
        n = self.accumulate_steps
        xs = x.chunk(n) if x is not None else [None] * n
        ys = y.chunk(n) if y is not None else [None] * n
        return list(zip(xs, ys))

    def _hcg_pp_ranks(self):
        ranks = getattr(self._hcg, "_pp_group", None)
        if ranks is None:
            ranks = list(range(self.pp))
        return ranks

    def train_batch(self, data, optimizer, lr_scheduler=None, scaler=None):
        micro = self._split_micro(data)
        n_micro = len(micro)
        dev = (torch.device("cuda", torch.cuda.current_device())
               if torch.cuda.is_available() else torch.device("cpu"))
        L = self._layers
        last_g = L.total_stages - 1
        losses = []
        # saved[(mb, g)] = (input, output) for the backward relay
        saved = {}

        owner = getattr(L, "owner", [g % self.pp for g in range(L.total_stages)])
        zbv = getattr(L, "placement", "interleaved") != "interleaved"
        pp_ranks = (self._hcg_pp_ranks() if zbv else None)

        def peer(stage):
            return pp_ranks[owner[stage]] if zbv else None

        local_fwd = {}
        for i in range(n_micro):
            for g in L.my_stages:
                if g == 0:
                    inp = micro[i][0]
                elif owner[g - 1] == self.rank:
                    inp = local_fwd.pop((i, g)).detach()   # V-turn: same rank
                    inp.requires_grad_(True)
                else:
                    inp = self.p2p.recv_forward(dev, src=peer(g - 1))
                    inp.requires_grad_(True)
                out = L.run_chunk(g, inp)
                if g == last_g:
                    loss = L.loss_fn(out, micro[i][1]) / n_micro
                    losses.append(loss.detach())
                    out = loss
                elif owner[g + 1] == self.rank:
                    local_fwd[(i, g + 1)] = out
                else:
                    self.p2p.send_forward(out, dst=peer(g + 1))
                saved[(i, g)] = (inp, out)

        local_bwd = {}
        for i in reversed(range(n_micro)):
            for g in reversed(L.my_stages):
                inp, out = saved.pop((i, g))
                if g == last_g:
                    out.backward()
                elif owner[g + 1] == self.rank:
                    out.backward(gradient=local_bwd.pop((i, g)))
                else:
                    grad_out = self.p2p.recv_backward(out, src=peer(g + 1))
                    out.backward(gradient=grad_out)
                if g != 0:
                    if owner[g - 1] == self.rank:
                        local_bwd[(i, g - 1)] = inp.grad
                    else:
                        self.p2p.send_backward(inp.grad, dst=peer(g - 1))

        if optimizer is not None:
            if scaler is not None:
                scaler.step(optimizer)
                scaler.update()
            else:
                optimizer.step()
            optimizer.clear_grad()
            if lr_scheduler is not None:
                lr_scheduler.step()
        if losses:
            return torch.stack(losses).sum()
        return torch.zeros(1, device=dev)

    def parameters(self, *a, **kw):
        return self._layers.parameters(*a, **kw)

    def state_dict(self, *a, **kw):
        return self._layers.state_dict(*a, **kw)


# ---------------------------------------------------------------------------
# Zero-bubble (ZB-H1) schedule: backward split into B (input-grad, on the
# critical p2p path) and W (weight-grad, deferred into pipeline bubbles).
# Reference: passes/pipeline_scheduler_pass/__init__.py:33 (FThenB/ZBH1
# registry) and the ZB-H1 paper schedule.  The W phase runs through
# fused_linear_param_grad_add (one accumulating GEMM per linear).
# ---------------------------------------------------------------------------
class _DeferredLinear(torch.autograd.Function):
    """Linear whose backward returns dX immediately and queues (x, dy) for
    a later dW accumulation -- the B/W split of zero-bubble schedules."""

    @staticmethod
    def forward(ctx, x, w, b, store):
        ctx.save_for_backward(x, w)
        ctx.store = store
        ctx.bias = b
        ctx.wref = w
        out = x @ w
        if b is not None:
            out = out + b
        return out

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        dx = dy @ w.t()
        ctx.store.append((x, dy, ctx.wref, ctx.bias))   # W work, deferred
        return dx, None, None, None


def _convert_to_zb(module, store):
    """Swap every nn-style Linear under `module` to the deferred-W path."""
    n = 0
    for sub in module.modules():
        if type(sub).__name__ == "Linear" and hasattr(sub, "weight") \
                and sub.weight.dim() == 2:
            def fwd(self, x, _store=store):
                return _DeferredLinear.apply(x, self.weight, self.bias, _store)
            sub.forward = fwd.__get__(sub)
            n += 1
    return n


def _flush_w(store, limit=None):
    """Run deferred weight-gradient GEMMs (the W phase)."""
    from ...ops import functional as hot
    n = 0
    while store and (limit is None or n < limit):
        x, dy, w, b = store.pop()
        x2 = x.reshape(-1, x.shape[-1])
        dy2 = dy.reshape(-1, dy.shape[-1])
        if w.grad is None:
            w.grad = torch.zeros_like(w)
        if b is not None and b.grad is None:
            b.grad = torch.zeros_like(b)
        hot.fused_linear_param_grad_add(x2, dy2, w.grad,
                                        b.grad if b is not None else None,
                                        has_bias=b is not None)
        n += 1
    return n


class _ZBP2P:
    """p2p delegate that queues a slice of deferred W GEMMs before each
    blocking receive -- on GPU the W kernels then execute while the host
    waits on the wire (the ZB bubble fill)."""

    def __init__(self, inner, store, per_recv=4):
        self._inner = inner
        self._store = store
        self._per_recv = per_recv

    def recv_backward(self, like, src=None):
        _flush_w(self._store, limit=self._per_recv)
        return self._inner.recv_backward(like, src=src)

    def send_backward_recv_forward(self, grad_in, device):
        _flush_w(self._store, limit=self._per_recv)
        return self._inner.send_backward_recv_forward(grad_in, device)

    def __getattr__(self, name):
        return getattr(self._inner, name)


class ZeroBubblePipelineParallel(PipelineParallel):
    """ZB-H1: 1F1B order for F and B; each micro-step's W (weight-grad)
    work is deferred and drained into the schedule's bubbles -- queued
    ahead of every blocking p2p receive and fully flushed before the
    tied-weight grad allreduce.  Gradients are bit-identical to 1F1B;
    the p2p-critical backward path only carries dX."""

    def __init__(self, layers: PipelineLayer, hcg, strategy=None):
        super().__init__(layers, hcg, strategy)
        self._w_store = []
        self._n_zb = _convert_to_zb(layers, self._w_store)
        self.p2p = _ZBP2P(self.p2p, self._w_store)
        # W must be complete before tied-weight grads are allreduced
        inner_ar = self._layers.allreduce_shared_weight_gradients

        def ar_with_flush():
            _flush_w(self._w_store)
            return inner_ar()
        self._layers.allreduce_shared_weight_gradients = ar_with_flush


class ZeroBubbleInterleavedPipelineParallel(InterleavedPipelineParallel):
    """ZB-VPP (reference passes/pipeline_scheduler_pass/__init__.py:33-38
    registry): the interleaved virtual-pipeline schedule with backward
    split into B (input-grad, stays on the p2p relay) and W (weight-grad
    GEMMs deferred into bubbles and drained before optimizer.step).
    Gradients are bit-identical to plain interleaved VPP."""

    def __init__(self, layers: VirtualPipelineLayer, hcg, strategy=None):
        super().__init__(layers, hcg, strategy)
        self._w_store = []
        self._n_zb = _convert_to_zb(layers, self._w_store)
        self.p2p = _ZBP2P(self.p2p, self._w_store)

    def train_batch(self, data, optimizer, lr_scheduler=None, scaler=None):
        store = self._w_store

        class _FlushingOpt:
            """Drains the deferred W phase before the real step."""

            def __init__(self, opt):
                self._opt = opt

            def step(self):
                _flush_w(store)
                self._opt.step()

            def __getattr__(self, name):
                return getattr(self._opt, name)

        wrapped = _FlushingOpt(optimizer) if optimizer is not None else None
        out = super().train_batch(data, wrapped, lr_scheduler, scaler)
        _flush_w(store)   # eval/no-opt paths: leave nothing queued
        return out
