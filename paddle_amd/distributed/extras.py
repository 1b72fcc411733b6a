"""paddle.distributed long-tail API (reference distributed/__init__.py
exports not covered by collective/parallel/auto_parallel):
split (auto-TP functional, fleet/layers/mpu/mp_ops.py), gloo_* helpers
(parallel_with_gloo.py), scatter_object_list, ParallelMode
(fleet/base/topology.py:42), ReduceType (phi/common/reduce_type.h),
shard_scaler / unshard_dtensor / shard_dataloader / DistModel / to_static
(auto_parallel/api.py), ShardingStage1-3 strategy markers."""
from __future__ import annotations

import torch
import torch.distributed as dist

from . import collective as C


class ParallelMode:
    """reference fleet/base/topology.py:42."""
    DATA_PARALLEL = 0
    TENSOR_PARALLEL = 1
    PIPELINE_PARALLEL = 2
    SHARDING_PARALLEL = 3


class ReduceType:
    """reference paddle/phi/common/reduce_type.h."""
    kRedSum = 0
    kRedMax = 1
    kRedMin = 2
    kRedProd = 3
    kRedAvg = 4
    kRedAny = 5
    kRedAll = 6


def is_available():
    return dist.is_available()


def get_backend(group=None):
    g = group.pg if group is not None and hasattr(group, "pg") else None
    return dist.get_backend(g)


def scatter_object_list(out_object_list, in_object_list=None, src=0, group=None):
    g = group.pg if group is not None and hasattr(group, "pg") else None
    return dist.scatter_object_list(out_object_list, in_object_list, src=src,
                                    group=g)


# -- gloo helpers (reference parallel_with_gloo.py) --------------------------
def gloo_init_parallel_env(rank_id, rank_num, server_endpoint):
    import os
    host, port = server_endpoint.rsplit(":", 1)
    os.environ.setdefault("MASTER_ADDR", host)
    os.environ.setdefault("MASTER_PORT", port)
    if not dist.is_initialized():
        dist.init_process_group("gloo", rank=rank_id, world_size=rank_num)


def gloo_barrier():
    dist.barrier()


def gloo_release():
    if dist.is_initialized():
        dist.destroy_process_group()


# -- split: functional auto-TP (reference fleet/layers/mpu/mp_ops.py) --------
def split(x, size, operation, axis=0, num_partitions=1, gather_out=True,
          weight_attr=None, bias_attr=None, name=None):
    """Build-and-apply a tensor-parallel linear/embedding over the model-
    parallel group.  operation='linear': axis=0 splits the in dim (row
    parallel), axis=1 the out dim (column parallel); 'embedding' splits
    the vocab.  Like the reference dygraph path, parameters are created
    at call time -- layer APIs (fleet.meta_parallel.*) are preferred for
    training loops."""
    from .fleet.mpu import (ColumnParallelLinear, RowParallelLinear,
                            VocabParallelEmbedding)
    if operation == "embedding":
        layer = VocabParallelEmbedding(size[0], size[1], weight_attr=weight_attr)
        return layer(x)
    if operation != "linear":
        raise ValueError(f"split: unsupported operation {operation!r}")
    if axis == 1:
        layer = ColumnParallelLinear(size[0], size[1], weight_attr=weight_attr,
                                     has_bias=bias_attr is not False,
                                     gather_output=gather_out)
    elif axis == 0:
        layer = RowParallelLinear(size[0], size[1], weight_attr=weight_attr,
                                  has_bias=bias_attr is not False,
                                  input_is_parallel=False)
    else:
        raise ValueError("split: axis must be 0 or 1 for linear")
    return layer(x)


# -- semi-auto helpers (reference auto_parallel/api.py) ----------------------
def shard_scaler(scaler):
    """api.py:1642 parity: the returned scaler synchronizes found_inf
    across ranks -- our amp.GradScaler already all-reduces found_inf
    (MAX) whenever torch.distributed is initialized, so this is the
    identity with the contract documented."""
    return scaler


def unshard_dtensor(dist_tensor):
    """api.py:2854: gather a dist tensor to a fully-replicated dense
    tensor."""
    from .auto_parallel import Replicate, reshard
    mesh = getattr(dist_tensor, "process_mesh", None)
    if mesh is None:
        return dist_tensor
    out = reshard(dist_tensor, mesh, [Replicate()] * mesh.ndim)
    for attr in ("dist_attr", "process_mesh", "placements"):
        if hasattr(out, attr):
            try:
                delattr(out, attr)
            except AttributeError:
                pass
    return out


class ShardDataloader:
    """api.py:3208 parity: iterate the wrapped loader, marking each batch
    tensor as a dist tensor on `meshes` with `shard_dims` placements."""

    def __init__(self, dataloader, meshes, input_keys=None, shard_dims=None,
                 is_dataset_splitted=False):
        self._loader = dataloader
        self._meshes = meshes if isinstance(meshes, (list, tuple)) else [meshes]
        self._shard_dims = shard_dims

    def __len__(self):
        return len(self._loader)

    def __iter__(self):
        from .auto_parallel import Replicate, Shard, shard_tensor
        mesh = self._meshes[0]
        for batch in self._loader:
            items = batch if isinstance(batch, (list, tuple)) else [batch]
            out = []
            for it in items:
                if isinstance(it, torch.Tensor):
                    pl = [Shard(0) if self._shard_dims is not None else Replicate()]
                    pl = pl * mesh.ndim
                    out.append(shard_tensor(it, mesh, pl))
                else:
                    out.append(it)
            yield out if isinstance(batch, (list, tuple)) else out[0]


def shard_dataloader(dataloader, meshes, input_keys=None, shard_dims=None,
                     is_dataset_splitted=False):
    return ShardDataloader(dataloader, meshes, input_keys, shard_dims,
                           is_dataset_splitted)


class ShardingStage1:
    """Strategy marker for shard_optimizer (reference api.py ShardingStage1)."""
    level = 1

    def __init__(self, mesh=None, axis=None):
        self.mesh, self.axis = mesh, axis


class ShardingStage2(ShardingStage1):
    level = 2


class ShardingStage3(ShardingStage1):
    level = 3


class DistModel:
    """Minimal reference DistModel (auto_parallel/api.py): wraps a layer
    with loss/optimizer for train/eval/predict stepping under the
    dygraph semi-auto engine."""

    def __init__(self, layer, loader=None, loss=None, optimizer=None,
                 strategy=None):
        self.network = layer
        self._loss = loss
        self._opt = optimizer
        self._mode = "train"

    def train(self):
        self._mode = "train"
        self.network.train()

    def eval(self):
        self._mode = "eval"
        self.network.eval()

    def predict(self):
        self._mode = "predict"
        self.network.eval()

    def __call__(self, *args):
        if self._mode == "predict" or self._loss is None:
            return self.network(*args)
        *inputs, labels = args
        out = self.network(*inputs)
        loss = self._loss(out, labels)
        if self._mode == "train" and loss.requires_grad:
            loss.backward()
            if self._opt is not None:
                self._opt.step()
                self._opt.clear_grad()
        return loss

    def state_dict(self, *a, **k):
        return self.network.state_dict(*a, **k)


def to_static(layer, loader=None, loss=None, optimizer=None, strategy=None):
    """reference dist.to_static: returns a DistModel driving one
    train/eval/predict step per call."""
    return DistModel(layer, loader, loss, optimizer, strategy)


# -- parameter-server data pipeline (reference entry_attr.py, fleet
#    InMemoryDataset/QueueDataset): PS mode is out of scope per SURVEY
#    §2.3 -- the names exist and raise with an explicit reason ----------
def _ps_gate(name):
    class _Gated:
        def __init__(self, *a, **k):
            raise NotImplementedError(
                f"paddle.distributed.{name} belongs to parameter-server "
                "mode, which is out of scope for this MI355X collective-"
                "mode build (SURVEY.md §2.3); use collective data "
                "loading (paddle.io.DataLoader + DistributedBatchSampler).")
    _Gated.__name__ = name
    return _Gated


QueueDataset = _ps_gate("QueueDataset")
InMemoryDataset = _ps_gate("InMemoryDataset")
CountFilterEntry = _ps_gate("CountFilterEntry")
ShowClickEntry = _ps_gate("ShowClickEntry")
ProbabilityEntry = _ps_gate("ProbabilityEntry")
MultiSlotDataGenerator = _ps_gate("MultiSlotDataGenerator")
MultiSlotStringDataGenerator = _ps_gate("MultiSlotStringDataGenerator")
