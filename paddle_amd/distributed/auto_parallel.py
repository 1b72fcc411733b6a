"""Semi-auto parallel (DistTensor) API subset.

Reference: python/paddle/distributed/auto_parallel/api.py (shard_tensor:
206, reshard:705, shard_layer:806, shard_optimizer:1591) and
phi/core/distributed/auto_parallel/ (placements, reshard functions).

Round-1 scope (SURVEY.md §2.3 marks DistTensor as phase-2): a working
manual-mesh subset -- ProcessMesh + Shard/Replicate/Partial placements,
shard_tensor producing rank-local shards with an attached dist_attr,
and reshard implementing the common placement conversions (s->r, r->s,
p->r) over the comm layer.  SPMD propagation rules are future work.
"""
from __future__ import annotations

from typing import List, Optional, Sequence

import numpy as np
import torch

from . import collective as C
from .parallel import get_rank, get_world_size


class Placement:
    pass


class Replicate(Placement):
    def __repr__(self):
        return "Replicate()"

    def __eq__(self, o):
        return isinstance(o, Replicate)


class Shard(Placement):
    def __init__(self, dim):
        self.dim = dim

    def __repr__(self):
        return f"Shard(dim={self.dim})"

    def __eq__(self, o):
        return isinstance(o, Shard) and o.dim == self.dim


class Partial(Placement):
    def __init__(self, reduce_type="sum"):
        self.reduce_type = reduce_type

    def __repr__(self):
        return "Partial()"

    def __eq__(self, o):
        return isinstance(o, Partial)


class ProcessMesh:
    """reference: python/paddle/distributed/auto_parallel/process_mesh.py:85"""

    def __init__(self, mesh, dim_names=None, shape=None, process_ids=None):
        arr = np.asarray(mesh)
        self._mesh = arr
        self._dim_names = list(dim_names) if dim_names else \
            [f"d{i}" for i in range(arr.ndim)]

    @property
    def shape(self):
        return list(self._mesh.shape)

    @property
    def ndim(self):
        return self._mesh.ndim

    @property
    def process_ids(self):
        return self._mesh.flatten().tolist()

    @property
    def dim_names(self):
        return self._dim_names

    def get_dim_size(self, name):
        return self._mesh.shape[self._dim_names.index(name)]

    def get_rank_by_dim_and_process_id(self, dim, pid):
        idx = np.argwhere(self._mesh == pid)
        return int(idx[0][self._dim_names.index(dim) if isinstance(dim, str) else dim])

    def my_coord(self):
        me = get_rank()
        loc = np.argwhere(self._mesh == me)
        return tuple(loc[0]) if len(loc) else None

    def _group_along(self, mesh_dim):
        """ranks varying along mesh_dim with my other coords fixed."""
        coord = self.my_coord()
        if coord is None:
            return None, []
        sl = list(coord)
        sl[mesh_dim] = slice(None)
        ranks = self._mesh[tuple(sl)].flatten().tolist()
        return ranks

    def __eq__(self, o):
        return isinstance(o, ProcessMesh) and np.array_equal(self._mesh, o._mesh)

    def __repr__(self):
        return f"ProcessMesh({self._mesh.tolist()}, dim_names={self._dim_names})"


class DistAttr:
    def __init__(self, mesh: ProcessMesh, placements: Sequence[Placement]):
        self.process_mesh = mesh
        self.placements = list(placements)

    def __repr__(self):
        return f"DistAttr(mesh={self.process_mesh}, placements={self.placements})"


_group_cache = {}


def _mesh_group(mesh: ProcessMesh, dim: int):
    key = (tuple(mesh.process_ids), tuple(mesh.shape), dim)
    if key not in _group_cache:
        # must be created collectively on all ranks
        groups = {}
        arr = mesh._mesh
        other = [i for i in range(arr.ndim) if i != dim]
        import itertools
        me = get_rank()
        my_group = None
        for combo in itertools.product(*(range(arr.shape[i]) for i in other)):
            sl = list(combo)
            sl.insert(dim, slice(None))
            ranks = arr[tuple(sl)].flatten().tolist()
            g = C.new_group(ranks)
            if me in ranks:
                my_group = g
        _group_cache[key] = my_group
    return _group_cache[key]


def shard_tensor(data, mesh: ProcessMesh, placements, dtype=None, place=None,
                 stop_gradient=None):
    """Return this rank's local shard with dist metadata attached."""
    t = data if isinstance(data, torch.Tensor) else torch.as_tensor(data)
    local = t
    for mdim, p in enumerate(placements):
        if isinstance(p, Shard):
            n = mesh.shape[mdim]
            coord = mesh.my_coord()
            idx = coord[mdim] if coord is not None else 0
            size = local.shape[p.dim] // n
            local = local.narrow(p.dim, idx * size, size).contiguous()
    local = local.clone()
    local.dist_attr = DistAttr(mesh, placements)
    local.process_mesh = mesh
    local.placements = list(placements)
    if stop_gradient is not None:
        local.requires_grad_(not stop_gradient and local.is_floating_point())
    return local


def dtensor_from_fn(fn, mesh, placements, *args, **kwargs):
    return shard_tensor(fn(*args, **kwargs), mesh, placements)


def reshard(dist_tensor, mesh: ProcessMesh, placements):
    """Common conversions: Shard->Replicate (all_gather), Partial->
    Replicate (all_reduce), Replicate->Shard (slice)."""
    src = getattr(dist_tensor, "placements", [Replicate()] * mesh.ndim)
    out = dist_tensor
    for mdim, (sp, dp) in enumerate(zip(src, placements)):
        if sp == dp:
            continue
        g = _mesh_group(mesh, mdim)
        if isinstance(sp, Shard) and isinstance(dp, Replicate):
            n = mesh.shape[mdim]
            if g is not None and n > 1:
                parts = [torch.empty_like(out) for _ in range(n)]
                C.all_gather(parts, out.contiguous(), group=g)
                out = torch.cat(parts, dim=sp.dim)
        elif isinstance(sp, Partial) and isinstance(dp, Replicate):
            if g is not None:
                out = out.clone()
                C.all_reduce(out, group=g)
        elif isinstance(sp, Replicate) and isinstance(dp, Shard):
            n = mesh.shape[mdim]
            coord = mesh.my_coord()
            idx = coord[mdim] if coord is not None else 0
            size = out.shape[dp.dim] // n
            out = out.narrow(dp.dim, idx * size, size).contiguous()
        elif isinstance(sp, Partial) and isinstance(dp, Shard):
            n = mesh.shape[mdim]
            if g is not None and n > 1:
                size = out.shape[dp.dim] // n
                shards = list(out.split(size, dim=dp.dim))
                dst = torch.empty_like(shards[0])
                C.reduce_scatter(dst, [s.contiguous() for s in shards], group=g)
                out = dst
        elif isinstance(sp, Shard) and isinstance(dp, Shard) and sp.dim != dp.dim:
            # s_to_s (reference reshard/s_to_s_reshard_function.cc): re-shard
            # from dim a to dim b with one all-to-all -- no gather round-trip
            n = mesh.shape[mdim]
            if g is not None and n > 1:
                pieces = [p.contiguous() for p in out.chunk(n, dim=dp.dim)]
                recv = [torch.empty_like(pieces[0]) for _ in range(n)]
                C.alltoall(pieces, recv, group=g)
                out = torch.cat(recv, dim=sp.dim)
        elif isinstance(sp, Replicate) and isinstance(dp, Partial):
            # r_to_p (reference r_to_p_reshard_function.cc): rank 0 keeps the
            # value, others zero -- the sum over the mesh dim reproduces it
            coord = mesh.my_coord()
            idx = coord[mdim] if coord is not None else 0
            out = out if idx == 0 else torch.zeros_like(out)
        else:
            raise NotImplementedError(f"reshard {sp} -> {dp}")
    out = out.clone() if out is dist_tensor else out
    out.dist_attr = DistAttr(mesh, placements)
    out.process_mesh = mesh
    out.placements = list(placements)
    return out


def shard_layer(layer, process_mesh, shard_fn=None, input_fn=None, output_fn=None):
    """api.py:806 parity: apply shard_fn(name, layer, mesh) to sublayers."""
    if shard_fn is not None:
        for name, sub in layer.named_sublayers() if hasattr(layer, "named_sublayers") \
                else layer.named_modules():
            shard_fn(name, sub, process_mesh)
    if input_fn is not None:
        layer.register_forward_pre_hook(
            lambda mod, inp: input_fn(inp, process_mesh))
    if output_fn is not None:
        layer.register_forward_hook(
            lambda mod, inp, out: output_fn(out, process_mesh))
    return layer


def shard_optimizer(optimizer, shard_fn=None):
    """api.py:1591 parity hook point; manual-parallel optimizers are already
    shard-aware in this build (fleet.sharding)."""
    return optimizer


def get_mesh():
    return _default_mesh


_default_mesh: Optional[ProcessMesh] = None


def set_mesh(mesh):
    global _default_mesh
    _default_mesh = mesh


# ---------------------------------------------------------------------------
# SPMD rules: einsum-notation placement propagation
# (reference: paddle/phi/infermeta/spmd_rules/*.cc -- matmul.cc,
#  elementwise.cc, reduction.cc use the same dim-char formulation)
# ---------------------------------------------------------------------------
def infer_spmd(notation: str, placements_list, mesh: ProcessMesh):
    """Given an einsum-style notation ("mk,kn->mn") and the input
    placements, return the output placements.

    Rules (matching the reference's dim-mapping semantics):
      - a mesh dim sharding an input dim whose char appears in the output
        -> output Shard(out_index)
      - a mesh dim sharding only contracted/reduced chars -> output Partial
      - conflicts (one mesh dim sharding different chars) -> Replicate
        (caller reshards an input first)
    """
    ins, out = notation.replace(" ", "").split("->")
    in_specs = ins.split(",")
    assert len(in_specs) == len(placements_list)
    by_mesh_dim = {}
    for spec, pls in zip(in_specs, placements_list):
        for mdim, p in enumerate(pls):
            if isinstance(p, Shard):
                if p.dim >= len(spec):
                    continue
                by_mesh_dim.setdefault(mdim, set()).add(spec[p.dim])
            elif isinstance(p, Partial):
                by_mesh_dim.setdefault(mdim, set()).add("+")
    out_pl = []
    for mdim in range(mesh.ndim):
        chars = by_mesh_dim.get(mdim, set())
        if len(chars) != 1:
            out_pl.append(Replicate())
            continue
        c = next(iter(chars))
        if c == "+":
            out_pl.append(Partial())
        elif c in out:
            out_pl.append(Shard(out.index(c)))
        else:
            out_pl.append(Partial())      # contracted sharded dim
    return out_pl


def _placements_of(t, mesh):
    return getattr(t, "placements", [Replicate()] * mesh.ndim)


def _local_op(fn, out_notation_placements, mesh, *tensors):
    out = fn(*tensors)
    out.dist_attr = DistAttr(mesh, out_notation_placements)
    out.process_mesh = mesh
    out.placements = list(out_notation_placements)
    return out


def dist_matmul(x, y):
    """SPMD matmul: local matmul + inferred placements.  A contraction-dim
    conflict (k sharded in only one operand) reshards that operand to
    Replicate first -- the reference's matmul.cc does the same fallback."""
    mesh = getattr(x, "process_mesh", None) or getattr(y, "process_mesh", None)
    assert mesh is not None, "dist_matmul needs dist tensors"
    px, py = _placements_of(x, mesh), _placements_of(y, mesh)
    # k = last dim of x, first of y
    for mdim in range(mesh.ndim):
        sx, sy = px[mdim], py[mdim]
        x_shards_k = isinstance(sx, Shard) and sx.dim == x.dim() - 1
        y_shards_k = isinstance(sy, Shard) and sy.dim == 0
        if x_shards_k != y_shards_k:        # one-sided contraction shard
            if x_shards_k:
                x = reshard(x, mesh, [Replicate() if i == mdim else p
                                      for i, p in enumerate(px)])
                px = x.placements
            else:
                y = reshard(y, mesh, [Replicate() if i == mdim else p
                                      for i, p in enumerate(py)])
                py = y.placements
    if x.dim() == 2 and y.dim() == 2:
        notation = "mk,kn->mn"
    elif x.dim() == 3 and y.dim() == 2:
        notation = "bmk,kn->bmn"
    else:
        notation = "mk,kn->mn"
    out_pl = infer_spmd(notation, [px, py], mesh)
    return _local_op(torch.matmul, out_pl, mesh, x, y)


def dist_elementwise(fn, *tensors):
    mesh = next(t.process_mesh for t in tensors if hasattr(t, "process_mesh"))
    specs = []
    pls = []
    chars = "abcdefgh"
    nd = max(t.dim() for t in tensors if isinstance(t, torch.Tensor))
    for t in tensors:
        specs.append(chars[nd - t.dim():nd])
        pls.append(_placements_of(t, mesh))
    out_pl = infer_spmd(",".join(specs) + "->" + chars[:nd], pls, mesh)
    return _local_op(fn, out_pl, mesh, *tensors)


def dist_reduce(x, axis, fn=torch.sum, keepdim=False):
    mesh = x.process_mesh
    nd = x.dim()
    chars = "abcdefgh"[:nd]
    out_chars = "".join(c for i, c in enumerate(chars) if i != axis % nd)
    out_pl = infer_spmd(chars + "->" + out_chars, [_placements_of(x, mesh)], mesh)
    return _local_op(lambda t: fn(t, dim=axis, keepdim=keepdim), out_pl, mesh, x)


class Strategy:
    """auto-parallel strategy knobs (reference: auto_parallel/strategy.py).
    Dataclass-lite: attribute bags per feature."""

    def __init__(self):
        self.auto_mode = "semi"
        self.sharding = type("S", (), {"enable": False, "degree": 1,
                                       "stage": 1})()
        self.recompute = type("R", (), {"enable": False})()
        self.pipeline = type("P", (), {"enable": False,
                                       "schedule_mode": "1F1B"})()
        self.amp = type("A", (), {"enable": False, "dtype": "bfloat16"})()


class Engine:
    """Semi-auto training engine (reference: distributed/auto_parallel/
    static/engine.py:Engine -- fit/evaluate/predict over a DistTensor
    program).  This build executes the same API on the dygraph path:
    shard_fn placements are applied via shard_layer, sharding/recompute
    strategy knobs map onto the Fleet implementations."""

    def __init__(self, model, loss=None, optimizer=None, metrics=None,
                 strategy=None):
        self.model = model
        self.loss = loss
        self.optimizer = optimizer
        self.metrics = metrics or []
        self.strategy = strategy or Strategy()
        if self.strategy.sharding.enable and self.strategy.sharding.stage == 3:
            from .fleet.sharding import GroupShardedStage3
            self.model = GroupShardedStage3(model)
        if self.strategy.recompute.enable and hasattr(model, "enable_recompute"):
            model.enable_recompute()

    def _step(self, batch, train=True):
        import contextlib

        import torch
        x, y = batch if isinstance(batch, (list, tuple)) else (batch, None)
        amp = self.strategy.amp
        ctx = (torch.autocast("cuda", dtype=torch.bfloat16)
               if amp.enable and torch.cuda.is_available()
               else contextlib.nullcontext())
        with ctx:
            out = self.model(x)
            loss = self.loss(out, y) if self.loss is not None else out
        if train:
            loss.backward()
            if self.optimizer is not None:
                self.optimizer.step()
                self.optimizer.clear_grad()
        return loss, out

    def fit(self, train_data, epochs=1, batch_size=1, steps_per_epoch=None,
            log_freq=10, verbose=0):
        from .. import io as pio
        loader = train_data if hasattr(train_data, "__iter__") else \
            pio.DataLoader(train_data, batch_size=batch_size, shuffle=True)
        history = []
        for ep in range(epochs):
            for step, batch in enumerate(loader):
                loss, _ = self._step(batch, train=True)
                if steps_per_epoch and step + 1 >= steps_per_epoch:
                    break
            history.append(float(loss.detach().float()))
        return history

    def evaluate(self, eval_data, batch_size=1, steps=None):
        import torch
        from .. import io as pio
        loader = eval_data if hasattr(eval_data, "__iter__") else \
            pio.DataLoader(eval_data, batch_size=batch_size)
        tot, n = 0.0, 0
        with torch.no_grad():
            for i, batch in enumerate(loader):
                loss, _ = self._step(batch, train=False)
                tot += float(loss.detach().float())
                n += 1
                if steps and i + 1 >= steps:
                    break
        return {"loss": tot / max(n, 1)}

    def predict(self, test_data, batch_size=1, steps=None):
        import torch
        from .. import io as pio
        loader = test_data if hasattr(test_data, "__iter__") else \
            pio.DataLoader(test_data, batch_size=batch_size)
        outs = []
        with torch.no_grad():
            for i, batch in enumerate(loader):
                x = batch[0] if isinstance(batch, (list, tuple)) else batch
                outs.append(self.model(x))
                if steps and i + 1 >= steps:
                    break
        return outs


def dist_embedding(ids, weight):
    """Vocab-sharded embedding (reference c_embedding_op + spmd rule
    phi/infermeta/spmd_rules/c_embedding.cc): each rank looks up its vocab
    slice (OOV rows -> 0) and the output is Partial over the sharding
    mesh dim (reshard/allreduce completes it)."""
    mesh = getattr(weight, "process_mesh", None)
    assert mesh is not None
    pw = _placements_of(weight, mesh)
    out_pl = []
    local = None
    for mdim, p in enumerate(pw):
        if isinstance(p, Shard) and p.dim == 0:
            n = mesh.shape[mdim]
            coord = mesh.my_coord()
            idx = coord[mdim] if coord is not None else 0
            per = weight.shape[0]
            start = idx * per
            shifted = ids - start
            ok = (shifted >= 0) & (shifted < per)
            local = torch.nn.functional.embedding(shifted.clamp(0, per - 1),
                                                  weight)
            local = local * ok.unsqueeze(-1).to(local.dtype)
            out_pl.append(Partial())
        elif isinstance(p, Shard):          # hidden-dim sharded
            out_pl.append(Shard(ids.dim()))
        else:
            out_pl.append(Replicate())
    if local is None:
        local = torch.nn.functional.embedding(ids, weight)
    local.dist_attr = DistAttr(mesh, out_pl)
    local.process_mesh = mesh
    local.placements = out_pl
    return local


def dist_cross_entropy(logits, labels):
    """Vocab-parallel softmax cross entropy (reference
    c_softmax_with_cross_entropy_op.cu + its spmd rule): logits sharded on
    the class dim stay sharded; the loss comes out Replicated via the
    max/sum/logit allreduce triplet."""
    mesh = getattr(logits, "process_mesh", None)
    assert mesh is not None
    pl = _placements_of(logits, mesh)
    vdim = logits.dim() - 1
    shard_mdim = next((m for m, p in enumerate(pl)
                       if isinstance(p, Shard) and p.dim == vdim), None)
    if shard_mdim is None:
        loss = torch.nn.functional.cross_entropy(
            logits.reshape(-1, logits.shape[-1]).float(), labels.reshape(-1),
            reduction="none").reshape(labels.shape)
        loss.dist_attr = DistAttr(mesh, [Replicate()] * mesh.ndim)
        loss.process_mesh = mesh
        loss.placements = [Replicate()] * mesh.ndim
        return loss
    g = _mesh_group(mesh, shard_mdim)
    n = mesh.shape[shard_mdim]
    coord = mesh.my_coord()
    idx = coord[shard_mdim] if coord is not None else 0
    per = logits.shape[-1]
    lf = logits.float().reshape(-1, per)
    lab = labels.reshape(-1)
    mx = lf.max(-1).values
    if g is not None:
        C.all_reduce(mx, op=C.ReduceOp.MAX, group=g)
    ex = torch.exp(lf - mx.unsqueeze(1))
    den = ex.sum(-1)
    if g is not None:
        C.all_reduce(den, group=g)
    shifted = lab - idx * per
    ok = (shifted >= 0) & (shifted < per)
    picked = lf.gather(1, shifted.clamp(0, per - 1).unsqueeze(1)).squeeze(1)
    picked = torch.where(ok, picked, torch.zeros_like(picked))
    if g is not None:
        C.all_reduce(picked, group=g)
    loss = (torch.log(den) + mx - picked).reshape(labels.shape)
    out_pl = [Replicate()] * mesh.ndim
    loss.dist_attr = DistAttr(mesh, out_pl)
    loss.process_mesh = mesh
    loss.placements = out_pl
    return loss


def dist_flash_attention(q, k, v, causal=True, scale=None):
    """Head-sharded flash attention SPMD rule (reference
    spmd_rules/flash_attention.cc): Shard on the head dim of q/k/v passes
    straight through (attention is head-wise independent); any seq/batch
    Partial or contraction conflicts fall back to Replicate inputs."""
    from ..ops import functional as hot
    mesh = getattr(q, "process_mesh", None)
    assert mesh is not None
    pq = _placements_of(q, mesh)
    o, _ = hot.flash_attention(q, k, v, causal=causal, scale=scale)
    o.dist_attr = DistAttr(mesh, list(pq))
    o.process_mesh = mesh
    o.placements = list(pq)
    return o
