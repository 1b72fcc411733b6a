"""ZeRO sharding stages 1/2/3 over RCCL/xGMI -- the north-star config.

Reference semantics: python/paddle/distributed/fleet/meta_parallel/sharding/
group_sharded_stage3.py:85 (hooks/prefetch/release), stage2, and
dygraph_sharding_optimizer.py (stage 1).  SURVEY.md A.2 records the exact
reference hook algorithm; this is a re-design, not a port:

MI355X-first differences (deliberate):
  * flat per-unit bf16 parameter buffers: each "unit" (transformer layer /
    embedding / head) flattens its params into ONE contiguous buffer,
    padded to world_size, gathered with a single all_gather_into_tensor
    per unit -- big collectives suit 153 GB/s point-to-point xGMI links
    (the reference gathers per-param).
  * grads are reduce-scattered (the reference stage-3 does all_reduce +
    local slice -- SURVEY.md A.2 notes this as a known inefficiency).
  * optimizer state is a flat fp32 master/m/v shard per unit driven by
    ONE fused AdamW kernel launch per unit (csrc/kernels/adamw.hip).
  * prefetch: unit i's pre-hook waits on its own pending gather and
    issues the async gather of the next unit in recorded order (forward)
    / previous unit (backward) -- same pipelining as the reference's
    _order_tracer/_wait_layer machinery.
"""
from __future__ import annotations

import contextlib
from typing import Dict, List, Optional

import torch

from ... import framework
from ...ops import functional as hot
from .. import collective as C
from ..parallel import get_rank, get_world_size

_ALIGN = 64  # element alignment inside a flat buffer (128B for bf16)


def _pad(n, m):
    return (n + m - 1) // m * m


class _Unit:
    """One shardable unit: a module subtree whose params live in one
    flat buffer."""

    def __init__(self, idx, module, params, world, rank, device, dtype,
                 group=None, sync_init=True, use_main_grad=False):
        self.idx = idx
        self.module = module
        self.params = params
        self.world = world
        self.rank = rank
        self.dtype = dtype
        self.group = group
        # layout
        self.offsets = []
        self.numels = [p.numel() for p in params]
        off = 0
        for n in self.numels:
            self.offsets.append(off)
            off = _pad(off + n, _ALIGN)
        self.total = _pad(off, _ALIGN * world)
        self.shard_size = self.total // world
        # build shard from current param values
        flat = torch.zeros(self.total, dtype=dtype, device=device)
        with torch.no_grad():
            for p, o, n in zip(params, self.offsets, self.numels):
                flat[o:o + n].copy_(p.data.reshape(-1).to(dtype))
            if sync_init and world > 1:
                # sync model init across ranks BEFORE slicing (each rank must
                # end up with rank0's values for ITS slice)
                C.broadcast(flat, src=(group.ranks[0] if group else 0), group=group)
        self.shard = flat[rank * self.shard_size:(rank + 1) * self.shard_size].clone()
        # fp32 master shard (optimizer state lives here)
        self.master = self.shard.float()
        self.grad_shard_fp32 = torch.zeros_like(self.master)
        # grad_ready: the tensor step()/grad-norm consume this step -- the
        # bf16 flat buffer directly on the fast (no-accumulation) paths, or
        # the fp32 accumulator.  grad_premul folds the 1/world reduce divide.
        self.grad_ready = None
        self.grad_premul = 1.0
        self.defer_flat_zero = False
        # fp32 main-grad accumulation (reference: main_grad hooks in
        # mixed_precision_utils) -- full-unit fp32 buffer, lazily allocated
        self.use_main_grad = use_main_grad
        self.main_grad = None
        del flat
        self.full: Optional[torch.Tensor] = None
        self.gather_work = None
        self.grad_work = None
        self.grad_flat = None
        self.ready_grads = 0
        self.gathered = False
        self.in_backward = False
        self.accum_steps = 0

    # -- param materialization ----------------------------------------------
    def launch_gather(self, async_op=True):
        if self.full is not None:
            return
        with torch.no_grad():
            if self.world == 1:
                # degenerate shard == full: alias, no copy (1-GPU fast path)
                self.full = self.shard
                self.gather_work = None
                return
            self.full = torch.empty(self.total, dtype=self.dtype, device=self.shard.device)
            self.gather_work = C.all_gather_into_tensor(self.full, self.shard,
                                                        group=self.group,
                                                        sync_op=not async_op)

    def wait_gather_and_bind(self):
        if self.full is None:
            self.launch_gather(async_op=False)
        if self.gather_work is not None:
            self.gather_work.wait()
            self.gather_work = None
        if not self.gathered:
            self.owner._storage2unit[self.full.untyped_storage().data_ptr()] = self
            for p, o, n in zip(self.params, self.offsets, self.numels):
                p.data = self.full[o:o + n].view(p._orig_shape)
            self.gathered = True

    def release(self):
        if self.full is not None:
            self.owner._storage2unit.pop(self.full.untyped_storage().data_ptr(), None)
        for p in self.params:
            p.data = p._placeholder
        self.full = None
        self.gathered = False

    # -- grads ---------------------------------------------------------------
    # autograd accumulates DIRECTLY into grad_flat: bind_grad_views() points
    # every p.grad at a slice before the unit's backward runs, so there is no
    # flatten copy at reduce time (the reference's FusedCommBuffer does the
    # same for stage-1-v2; here it covers stage-3 too).
    def bind_grad_views(self):
        if self.grad_flat is None:
            self.grad_flat = torch.zeros(self.total, dtype=self.dtype,
                                         device=self.shard.device)
        for p, o, n in zip(self.params, self.offsets, self.numels):
            p.grad = self.grad_flat[o:o + n].view(p._orig_shape)

    def reduce_grads(self, accumulate):
        flat = self.grad_flat
        for p in self.params:
            p.grad = None
        self.accum_steps += 1
        if self.use_main_grad:
            # fold this micro-step's bf16 grads into the fp32 accumulator
            # (one add pass; bf16 rounding only within a single micro-step)
            if self.main_grad is None:
                self.main_grad = torch.zeros(self.total, dtype=torch.float32,
                                             device=self.shard.device)
            self.main_grad.add_(flat)
            flat.zero_()
            if accumulate:
                return
            if self.world == 1:
                self.grad_ready = self.main_grad[0:self.shard_size]
                self.grad_premul = 1.0
                self.defer_flat_zero = False
            else:
                out = torch.empty(self.shard_size, dtype=torch.float32,
                                  device=self.shard.device)
                work = C.reduce_scatter_tensor(out, self.main_grad,
                                               group=self.group, sync_op=False)
                self.grad_work = (work, out)
            return
        if accumulate:
            # DDP-style no_sync: micro-step grads keep accumulating into the
            # flat buffer locally (bind_grad_views re-binds the same views on
            # the next backward, autograd adds in place); the collective runs
            # only on the final, synchronizing micro-step
            return
        if self.world == 1:
            # fast path: the optimizer consumes the bf16 flat buffer
            # directly (fused grad-scale in the AdamW kernel); zeroing
            # is deferred to clear_grad() after step() has read it.
            self.grad_ready = flat[0:self.shard_size]
            self.grad_premul = 1.0
            self.defer_flat_zero = True
        else:
            # async reduce-scatter overlapped with the remaining backward;
            # the optimizer's step() calls finish_grad_reduce() first.
            # Grads are SUMMED (no div pass over the 2x-total-bytes flat
            # buffer); the 1/world lands in grad_premul for the AdamW
            # kernel's fused grad scale.
            out = torch.empty(self.shard_size, dtype=self.dtype,
                              device=self.shard.device)
            work = C.reduce_scatter_tensor(out, flat, group=self.group,
                                           sync_op=False)
            self.grad_work = (work, out)

    def finish_grad_reduce(self):
        if self.grad_work is not None:
            work, out = self.grad_work
            if work is not None:
                work.wait()
            self.grad_ready = out
            self.grad_premul = 1.0 / self.world
            self.grad_flat.zero_()
            self.grad_work = None


class GroupShardedStage3(torch.nn.Module):
    """Wraps a Layer; params sharded at unit granularity (ZeRO-3)."""

    def __init__(self, layer, optimizer=None, group=None, sync_buffers=False,
                 device=None, segment_size=2 ** 20, pertrain_sync_models=True,
                 offload=False, sync_comm=False, dp_group=None,
                 exclude_layer=None, param_dtype=None, use_main_grad=False):
        super().__init__()
        self._use_main_grad = use_main_grad
        self._layer = layer
        self.group = group
        self.world = get_world_size(group)
        self.rank = get_rank(group) if group is None else group.rank
        if device is None:
            # follow the MODEL, not CUDA availability: a CPU-built layer
            # on a GPU box must shard on CPU (gloo tests run everywhere)
            p = next(iter(layer.parameters()), None)   # paddle .parameters() is a list
            device = p.device if p is not None else (
                torch.device("cuda", torch.cuda.current_device())
                if torch.cuda.is_available() else torch.device("cpu"))
        dev = torch.device(device)
        self.device = dev
        self.sync_comm = sync_comm
        self._units: List[_Unit] = []
        self._p2unit: Dict[int, _Unit] = {}
        self._storage2unit: Dict[int, _Unit] = {}
        self._order: List[int] = []
        self._order_recorded = False
        self._accumulating = False
        self._sync_init = pertrain_sync_models
        self._build_units(param_dtype)
        self._register_hooks()

    # -- unit discovery -------------------------------------------------------
    def _unit_modules(self):
        """Units = modules the model declares via `sharding_units()` or the
        direct children owning parameters (embedding / layers / head)."""
        if hasattr(self._layer, "sharding_units"):
            return list(self._layer.sharding_units())
        units = []

        def walk(m):
            for child in m.children():
                nparam = sum(1 for _ in child.parameters())
                # big children with direct structure get split further
                if nparam == 0:
                    continue
                from ...nn.layer import LayerList, Sequential
                if isinstance(child, (LayerList, Sequential, torch.nn.ModuleList)):
                    walk(child)
                else:
                    units.append(child)

        walk(self._layer)
        # params not covered by any unit (directly on root): one extra unit
        covered = set()
        for u in units:
            for p in u.parameters():
                covered.add(id(p))
        root_extra = [p for p in self._layer.parameters() if id(p) not in covered]
        return units + ([self._layer] if root_extra else [])

    def _build_units(self, param_dtype):
        seen = set()
        for idx, mod in enumerate(self._unit_modules()):
            params = [p for p in mod.parameters()
                      if p.requires_grad and id(p) not in seen]
            for p in params:
                seen.add(id(p))
            if not params:
                continue
            dtype = param_dtype or params[0].dtype
            for p in params:
                p._orig_shape = p.shape
                p._placeholder = torch.empty(0, dtype=p.dtype, device=self.device)
            u = _Unit(len(self._units), mod, params, self.world, self.rank,
                      self.device, dtype, group=self.group,
                      sync_init=self._sync_init,
                      use_main_grad=self._use_main_grad)
            u.owner = self
            self._units.append(u)
            for p in params:
                self._p2unit[id(p)] = u
            # release the original full params now
            u.release()

    # -- hooks ----------------------------------------------------------------
    def _register_hooks(self):
        # the full-backward-pre hook intentionally keys off OUTPUT grads
        # ("backward reached this unit"); torch warns when module inputs
        # don't require grad (the embedding unit's int ids) -- expected
        import warnings
        warnings.filterwarnings(
            "ignore", message="Full backward hook is firing")
        for u in self._units:
            u.module.register_forward_pre_hook(self._make_fwd_pre(u))
            u.module.register_forward_hook(self._make_fwd_post(u))
            u.module.register_full_backward_pre_hook(self._make_bwd_pre(u))
            for p in u.params:
                p.register_post_accumulate_grad_hook(self._on_grad)

    def _make_fwd_pre(self, u):
        def hook(mod, inputs):
            u.wait_gather_and_bind()
            # prefetch next unit in recorded order (pipelined gathers --
            # the reference's _order_tracer/_allgather_buffer machinery)
            if self._order_recorded:
                pos = self._order.index(u.idx)
                if pos + 1 < len(self._order):
                    self._units[self._order[pos + 1]].launch_gather(
                        async_op=not self.sync_comm)
            else:
                self._order.append(u.idx)
            return None
        return hook

    def _make_fwd_post(self, u):
        def hook(mod, inputs, outputs):
            u.release()
            return None
        return hook

    def _make_bwd_pre(self, u):
        def hook(mod, grad_output):
            if not self._order_recorded:
                self._order_recorded = True
            u.in_backward = True
            u.wait_gather_and_bind()
            u.bind_grad_views()  # grads accumulate straight into the flat buffer
            # prefetch previous unit (next to run backward)
            pos = self._order.index(u.idx)
            if pos - 1 >= 0:
                self._units[self._order[pos - 1]].launch_gather(async_op=not self.sync_comm)
            return None
        return hook

    def _on_grad(self, p):
        u = self._p2unit[id(p)]
        u.ready_grads += 1
        if u.ready_grads == len(u.params):
            u.ready_grads = 0
            u.reduce_grads(accumulate=self._accumulating)
            u.release()
            u.in_backward = False

    # -- saved-tensor indirection -------------------------------------------
    # Ops that use unit params save them for backward; saving the raw view
    # would keep the released gather buffer alive in the autograd graph.
    # pack replaces any tensor living in a unit's full-buffer storage with a
    # lightweight marker; unpack re-gathers (the bwd-pre hook normally did
    # already) and returns a fresh view.  This is what makes release()
    # actually free memory -- the ZeRO-3 point.
    def _pack(self, t):
        try:
            key = t.untyped_storage().data_ptr()
        except Exception:
            return t
        u = self._storage2unit.get(key)
        if u is not None:
            return ("__pa_unit__", u.idx, t.storage_offset(), tuple(t.shape),
                    tuple(t.stride()), t.dtype)
        return t

    def _unpack(self, x):
        if isinstance(x, tuple) and len(x) == 6 and x[0] == "__pa_unit__":
            _, idx, off, shape, stride, dtype = x
            u = self._units[idx]
            u.wait_gather_and_bind()
            return torch.as_strided(u.full, shape, stride, off)
        return x

    # -- API ------------------------------------------------------------------
    def forward(self, *args, **kwargs):
        # kick off the first unit's gather up-front
        if self._order_recorded and self._order:
            self._units[self._order[0]].launch_gather(async_op=True)
        with torch.autograd.graph.saved_tensors_hooks(self._pack, self._unpack):
            out = self._layer(*args, **kwargs)
        return out

    def no_sync(self):
        @contextlib.contextmanager
        def ctx():
            self._accumulating = True
            try:
                yield
            finally:
                self._accumulating = False
        return ctx()

    def get_all_parameters(self, convert2cpu=False):
        """materialize full params (reference :695) -- for save/eval."""
        for u in self._units:
            u.wait_gather_and_bind()
        if convert2cpu:
            for u in self._units:
                for p in u.params:
                    p.data = p.data.cpu()
        return list(self._layer.parameters())

    def state_dict(self, *a, **kw):
        self.get_all_parameters()
        sd = self._layer.state_dict(*a, **kw)
        for u in self._units:
            u.release()
        return sd

    def set_state_dict(self, sd, **kw):
        for u in self._units:
            u.wait_gather_and_bind()
        self._layer.set_state_dict(sd, **kw) if hasattr(self._layer, "set_state_dict") \
            else self._layer.load_state_dict(sd)
        with torch.no_grad():
            for u in self._units:
                flat = torch.zeros(u.total, dtype=u.dtype, device=self.device)
                for p, o, n in zip(u.params, u.offsets, u.numels):
                    flat[o:o + n].copy_(p.data.reshape(-1).to(u.dtype))
                u.shard.copy_(flat[self.rank * u.shard_size:(self.rank + 1) * u.shard_size])
                u.master.copy_(u.shard.float())
                u.release()

    def parameters(self, *a, **kw):
        return self._layer.parameters(*a, **kw)

    def named_parameters(self, *a, **kw):
        return self._layer.named_parameters(*a, **kw)

    def __getattr__(self, name):
        try:
            return super().__getattr__(name)
        except AttributeError:
            return getattr(self._layer, name)


class ShardedAdamW:
    """Flat-shard AdamW for GroupShardedStage3 -- one fused kernel launch
    per unit (fp32 master/m/v shards; bf16 shard write-back)."""

    def __init__(self, sharded_model: GroupShardedStage3, learning_rate=1e-3,
                 beta1=0.9, beta2=0.999, epsilon=1e-8, weight_decay=0.01,
                 grad_clip=None, clip_groups=None):
        self.model = sharded_model
        self._lr = learning_rate
        self.beta1, self.beta2, self.eps = beta1, beta2, epsilon
        self.weight_decay = weight_decay
        self.grad_clip = grad_clip
        self.step_count = 0
        self._m = [torch.zeros_like(u.master) for u in sharded_model._units]
        self._v = [torch.zeros_like(u.master) for u in sharded_model._units]

    def get_lr(self):
        from ...optimizer.lr import LRScheduler
        return self._lr() if isinstance(self._lr, LRScheduler) else self._lr

    def set_lr(self, lr):
        self._lr = lr

    def _global_grad_norm(self):
        total = None
        for u in self.model._units:
            u.finish_grad_reduce()
        for u in self.model._units:
            g = u.grad_ready if u.grad_ready is not None else u.grad_shard_fp32
            sq = hot.l2_norm_squared(g)
            if u.grad_premul != 1.0:
                sq = sq * (u.grad_premul ** 2)
            total = sq if total is None else total + sq
        if total is None:
            return None
        if self.model.world > 1:
            C.all_reduce(total, group=self.model.group)
        return total.sqrt()

    @torch.no_grad()
    def step(self):
        self.step_count += 1
        lr = self.get_lr()
        for u in self.model._units:
            u.finish_grad_reduce()
        clip_coeff = None
        if self.grad_clip is not None:
            clip_norm = getattr(self.grad_clip, "clip_norm", self.grad_clip)
            gnorm = self._global_grad_norm()
            if gnorm is not None:
                clip_coeff = float(clip_norm) / float(max(float(gnorm), float(clip_norm)))
        for u, m, v in zip(self.model._units, self._m, self._v):
            g = u.grad_ready if u.grad_ready is not None else u.grad_shard_fp32
            scale = u.grad_premul
            if clip_coeff is not None and clip_coeff < 1.0:
                scale = scale * clip_coeff
            hot.fused_adamw_step(u.master, u.shard, g, m, v, lr, self.beta1,
                                 self.beta2, self.eps, self.weight_decay,
                                 self.step_count, grad_scale=scale)
            u.accum_steps = 0
            u.grad_ready = None

    def clear_grad(self, set_to_zero=True):
        for u in self.model._units:
            u.finish_grad_reduce()  # consume any pending async reduce
            if u.defer_flat_zero and u.grad_flat is not None:
                u.grad_flat.zero_()
                u.defer_flat_zero = False
            if u.use_main_grad and u.main_grad is not None:
                u.main_grad.zero_()
            u.grad_ready = None
            u.accum_steps = 0

    clear_gradients = clear_grad

    def state_dict(self):
        sd = {"step": self.step_count}
        for i, u in enumerate(self.model._units):
            sd[f"unit{i}_master"] = u.master
            sd[f"unit{i}_m"] = self._m[i]
            sd[f"unit{i}_v"] = self._v[i]
        return sd

    # -- distcp resharding hooks ---------------------------------------------
    # each unit's master/m/v is the flat slice [rank*shard_size, +shard_size)
    # of the unit's padded global buffer -- exactly the shard_info model of
    # distributed.checkpoint (reference: checkpoint/save_state_dict.py)
    def sharded_state_dict(self):
        sd = {"step": self.step_count}
        info = {}
        for i, u in enumerate(self.model._units):
            off = u.rank * u.shard_size
            for nm, t in (("master", u.master), ("m", self._m[i]),
                          ("v", self._v[i])):
                k = f"unit{i}_{nm}"
                sd[k] = t
                info[k] = {"global_numel": u.total, "offset": off}
        return sd, info

    def save_sharded(self, path):
        from ...distributed import checkpoint as dcp
        sd, info = self.sharded_state_dict()
        dcp.save_state_dict(sd, path, shard_info=info)

    def load_sharded(self, path):
        from ...distributed import checkpoint as dcp
        sd, info = self.sharded_state_dict()
        dcp.load_state_dict(sd, path, shard_info=info)
        self.step_count = int(sd["step"])
        # masters changed: refresh the bf16 model shards
        for u in self.model._units:
            with torch.no_grad():
                u.shard.copy_(u.master.to(u.shard.dtype))

    def set_state_dict(self, sd):
        self.step_count = int(sd.get("step", 0))
        for i, u in enumerate(self.model._units):
            if f"unit{i}_master" in sd:
                u.master.copy_(torch.as_tensor(sd[f"unit{i}_master"]).to(u.master.device))
                u.shard.copy_(u.master.to(u.dtype))
                self._m[i].copy_(torch.as_tensor(sd[f"unit{i}_m"]).to(u.master.device))
                self._v[i].copy_(torch.as_tensor(sd[f"unit{i}_v"]).to(u.master.device))


# ---------------------------------------------------------------------------
# stages 1/2: optimizer-level sharding at parameter granularity
# (dygraph_sharding_optimizer.py:54 / group_sharded_optimizer_stage2)
# ---------------------------------------------------------------------------
class DygraphShardingOptimizer:
    """Stage 1: each rank owns a param partition (greedy by size); grads
    all-reduced (by DataParallel or sync_gradients); owner updates; params
    broadcast back."""

    def __init__(self, optimizer, hcg=None, group=None):
        self._inner = optimizer
        self.group = group or (hcg.get_sharding_parallel_group() if hcg else None)
        self.world = get_world_size(self.group)
        self.rank = get_rank(self.group) if self.group is None else self.group.rank
        self._partition()

    def _partition(self):
        params = sorted(self._inner._params, key=lambda p: -p.numel())
        sizes = [0] * self.world
        owner = {}
        for p in params:
            r = sizes.index(min(sizes))
            sizes[r] += p.numel()
            owner[id(p)] = r
        self._owner = owner
        self._local = [p for p in self._inner._params if owner[id(p)] == self.rank]
        self._all_params = list(self._inner._params)
        self._inner._params = self._local

    def step(self):
        self._inner.step()
        if self.world > 1:
            with torch.no_grad():
                for p in self._all_params:
                    src_local = self._owner[id(p)]
                    src = self.group.ranks[src_local] if self.group else src_local
                    C.broadcast(p.data, src=src, group=self.group)

    def clear_grad(self, set_to_zero=True):
        for p in self._all_params:
            if p.grad is not None:
                if set_to_zero:
                    p.grad.zero_()
                else:
                    p.grad = None

    clear_gradients = clear_grad

    def __getattr__(self, name):
        return getattr(self._inner, name)


def group_sharded_parallel(model, optimizer, level, scaler=None, group=None,
                           offload=False, sync_buffers=False, buffer_max_size=2 ** 23,
                           segment_size=2 ** 20, sync_comm=False, dp_group=None,
                           exclude_layer=None):
    """python/paddle/distributed/sharding/group_sharded.py parity.
    level: 'os' (stage1) | 'os_g' (stage2) | 'p_g_os' (stage3)."""
    if level == "os":
        from ..parallel import DataParallel
        model = DataParallel(model, group=group)
        opt = DygraphShardingOptimizer(optimizer, group=group)
        return model, opt, scaler
    if level == "os_g":
        model = GroupShardedStage2(model, optimizer, group=group)
        return model, model.optimizer, scaler
    if level == "p_g_os":
        wrapped = GroupShardedStage3(model, optimizer, group=group,
                                     sync_comm=sync_comm)
        opt = ShardedAdamW(wrapped,
                           learning_rate=optimizer._lr,
                           beta1=getattr(optimizer, "_beta1", 0.9),
                           beta2=getattr(optimizer, "_beta2", 0.999),
                           epsilon=getattr(optimizer, "_eps", 1e-8),
                           weight_decay=getattr(optimizer, "_weight_decay", 0.0),
                           grad_clip=getattr(optimizer, "_grad_clip", None))
        return wrapped, opt, scaler
    raise ValueError(f"unknown sharding level {level!r}")


class GroupShardedStage2(torch.nn.Module):
    """ZeRO-2: grads reduce-scattered at parameter granularity to the
    owning rank; optimizer state sharded; updated params broadcast."""

    def __init__(self, layer, optimizer, group=None, sync_buffers=False,
                 buffer_max_size=2 ** 23, device=None):
        super().__init__()
        self._layer = layer
        self.group = group
        self.world = get_world_size(group)
        self.rank = get_rank(group) if group is None else group.rank
        self._inner_opt = optimizer
        self._shard_opt = DygraphShardingOptimizer(optimizer, group=group)
        self._hooks = []
        if self.world > 1:
            for p in layer.parameters():
                if p.requires_grad:
                    self._hooks.append(p.register_post_accumulate_grad_hook(self._on_grad))

    def _on_grad(self, p):
        # reduce to owner only (stage-2 grad sharding): reduce op to owner rank
        owner_local = self._shard_opt._owner[id(p)]
        dst = self.group.ranks[owner_local] if self.group else owner_local
        p.grad.div_(self.world)
        C.reduce(p.grad, dst=dst, group=self.group)
        if owner_local != self.rank:
            p.grad = None  # free non-owned grads

    @property
    def optimizer(self):
        return self._shard_opt

    def forward(self, *a, **kw):
        return self._layer(*a, **kw)

    def state_dict(self, *a, **kw):
        return self._layer.state_dict(*a, **kw)

    def parameters(self, *a, **kw):
        return self._layer.parameters(*a, **kw)

    def __getattr__(self, name):
        try:
            return super().__getattr__(name)
        except AttributeError:
            return getattr(self._layer, name)
