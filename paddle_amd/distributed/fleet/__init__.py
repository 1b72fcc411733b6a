"""Fleet facade (reference: python/paddle/distributed/fleet/fleet.py:151).

fleet.init(is_collective=True, strategy) -> builds HybridCommunicateGroup
from strategy.hybrid_configs; distributed_model / distributed_optimizer
wrap for the active parallelism mix (DP / sharding 1-3 / TP / PP).
"""
from __future__ import annotations

from typing import Optional

import torch

from .. import collective as C
from ..parallel import DataParallel, get_rank, get_world_size, init_parallel_env
from . import sharding as sharding_mod
from .random import RNGStatesTracker, get_rng_state_tracker, model_parallel_random_seed
from .sharding import (DygraphShardingOptimizer, GroupShardedStage2,
                       GroupShardedStage3, ShardedAdamW, group_sharded_parallel)
from .topology import CommunicateTopology, HybridCommunicateGroup


class DistributedStrategy:
    """reference: fleet/base/distributed_strategy.py (proto-backed; here a
    plain config object with the same field names)."""

    def __init__(self):
        self.hybrid_configs = {
            "dp_degree": 1,
            "mp_degree": 1,
            "pp_degree": 1,
            "sharding_degree": 1,
            "sep_degree": 1,
        }
        self.sharding_configs = {
            "stage": 1,
            "degree": 1,
            "offload": False,
            "comm_overlap": True,
        }
        self.pipeline_configs = {
            "accumulate_steps": 1,
            "micro_batch_size": 1,
        }
        self.amp = False
        self.amp_configs = {}
        self.recompute = False
        self.recompute_configs = {}
        self.gradient_merge = False
        self.gradient_merge_configs = {}
        self.find_unused_parameters = False

    def __repr__(self):
        return f"DistributedStrategy(hybrid={self.hybrid_configs})"


class _FleetState:
    def __init__(self):
        self.initialized = False
        self.strategy: Optional[DistributedStrategy] = None
        self.hcg: Optional[HybridCommunicateGroup] = None
        self.is_collective = True


_state = _FleetState()


def init(role_maker=None, is_collective=True, strategy=None, log_level="INFO"):
    if strategy is None:
        strategy = DistributedStrategy()
    _state.strategy = strategy
    _state.is_collective = is_collective
    world = get_world_size()
    if world > 1 or torch.distributed.is_available():
        try:
            init_parallel_env()
        except Exception:
            if world > 1:
                raise
    hc = strategy.hybrid_configs
    dp = hc.get("dp_degree", 1)
    mp = hc.get("mp_degree", 1)
    pp = hc.get("pp_degree", 1)
    sh = hc.get("sharding_degree", 1)
    sep = hc.get("sep_degree", 1)
    # auto-fill dp from world size (fleet behavior)
    prod = mp * pp * sh * sep
    if dp * prod != world and world % prod == 0:
        dp = world // prod
        hc["dp_degree"] = dp
    topo = CommunicateTopology(("data", "pipe", "sharding", "sep", "model"),
                               (dp, pp, sh, sep, mp))
    _state.hcg = HybridCommunicateGroup(topo)
    _state.initialized = True
    return _state


def get_hybrid_communicate_group() -> HybridCommunicateGroup:
    return _state.hcg


def distributed_model(model):
    """fleet/model.py:134 parity: wrap for the strategy's parallelism."""
    assert _state.initialized, "call fleet.init first"
    hcg = _state.hcg
    strategy = _state.strategy
    sh_deg = hcg.get_sharding_parallel_world_size()
    dp_deg = hcg.get_data_parallel_world_size()
    pp_deg = hcg.get_pipe_parallel_world_size()
    if pp_deg > 1:
        from .pipeline import PipelineParallel
        model = PipelineParallel(model, hcg, strategy)
        return model
    stage = _state.strategy.sharding_configs.get("stage", 1) if sh_deg > 1 else 0
    if sh_deg > 1 and stage == 3:
        model = GroupShardedStage3(model, group=hcg.get_sharding_parallel_group())
    elif sh_deg > 1 and stage == 2:
        # stage2 wrapping happens in distributed_optimizer (needs the opt)
        pass
    if dp_deg > 1:
        model = DataParallel(model, group=hcg.get_data_parallel_group())
    return model


def distributed_optimizer(optimizer, strategy=None):
    """fleet/fleet.py:1427 parity."""
    assert _state.initialized
    hcg = _state.hcg
    sh_deg = hcg.get_sharding_parallel_world_size()
    stage = _state.strategy.sharding_configs.get("stage", 1)
    if sh_deg > 1 and stage == 1:
        return DygraphShardingOptimizer(optimizer, hcg=hcg)
    return HybridParallelOptimizer(optimizer, hcg, _state.strategy)


class HybridParallelOptimizer:
    """reference: dygraph_optimizer/hybrid_parallel_optimizer.py:266 --
    grad sync across mp/pp for shared params + inner step."""

    def __init__(self, optimizer, hcg, strategy):
        self._inner = optimizer
        self._hcg = hcg

    def step(self):
        hcg = self._hcg
        # sync grads across mp group for params marked is_distributed=False
        mp_group = hcg.get_model_parallel_group()
        if mp_group is not None and hcg.get_model_parallel_world_size() > 1:
            for p in self._inner._params:
                if p.grad is not None and not getattr(p, "is_distributed", False):
                    C.all_reduce(p.grad, group=mp_group)
                    p.grad.div_(hcg.get_model_parallel_world_size())
        self._inner.step()

    def clear_grad(self, set_to_zero=True):
        self._inner.clear_grad(set_to_zero)

    clear_gradients = clear_grad

    def __getattr__(self, name):
        return getattr(self._inner, name)


def worker_index():
    return get_rank()


def worker_num():
    return get_world_size()


def is_first_worker():
    return get_rank() == 0


def barrier_worker():
    if C.is_initialized():
        C.barrier()


class UtilBase:
    def all_reduce(self, input, mode="sum"):
        import numpy as np
        t = torch.as_tensor(np.asarray(input))
        C.all_reduce(t)
        return t.numpy()


util = UtilBase()

# meta_parallel namespace parity
from . import mpu  # noqa: E402,F401
from . import utils  # noqa: F401
from .pipeline import PipelineLayer, PipelineParallel, LayerDesc, SharedLayerDesc  # noqa: E402,F401
from .recompute import recompute  # noqa: E402,F401

from . import meta_parallel  # noqa: E402,F401
from . import metrics  # noqa: E402,F401


# -- role makers + Fleet class (reference fleet/base/role_maker.py:40, 548
#    and fleet/fleet.py; the PS-specific SERVER/HETER roles are plumbing
#    only -- this build is collective-mode) --------------------------------
class Role:
    WORKER = 1
    SERVER = 2
    HETER_WORKER = 3
    ALL = 4
    COORDINATOR = 5


class RoleMakerBase:
    def __init__(self, is_collective=True, **kwargs):
        import os
        self._is_collective = is_collective
        self._rank = int(os.environ.get("RANK",
                         os.environ.get("PADDLE_TRAINER_ID", "0")))
        self._world = int(os.environ.get("WORLD_SIZE",
                          os.environ.get("PADDLE_TRAINERS_NUM", "1")))

    def _role(self):
        return Role.WORKER

    def worker_index(self):
        return self._rank

    def worker_num(self):
        return self._world

    def is_worker(self):
        return True

    def is_first_worker(self):
        return self._rank == 0

    def is_server(self):
        return False


class PaddleCloudRoleMaker(RoleMakerBase):
    """Env-var driven role discovery (reference role_maker.py:548);
    collective mode reads RANK/WORLD_SIZE (or PADDLE_TRAINER_*)."""


class UserDefinedRoleMaker(RoleMakerBase):
    """Explicit role assignment (reference role_maker.py UserDefined)."""

    def __init__(self, is_collective=True, current_id=0, role=Role.WORKER,
                 worker_num=1, **kwargs):
        super().__init__(is_collective, **kwargs)
        self._rank = current_id
        self._world = worker_num
        self._role_v = role

    def _role(self):
        return self._role_v


class Fleet:
    """The class behind the module-level fleet facade (reference
    fleet/fleet.py Fleet); init() delegates to the module functions so
    `fleet.Fleet().init(...)` and `fleet.init(...)` behave alike."""

    def __init__(self):
        self._role_maker = None

    def init(self, role_maker=None, is_collective=True, strategy=None, log_level="INFO"):
        self._role_maker = role_maker or PaddleCloudRoleMaker(is_collective)
        init(is_collective=is_collective, strategy=strategy)
        return self

    def worker_index(self):
        return (self._role_maker or PaddleCloudRoleMaker()).worker_index()

    def worker_num(self):
        return (self._role_maker or PaddleCloudRoleMaker()).worker_num()

    def is_first_worker(self):
        return self.worker_index() == 0

    def __getattr__(self, name):
        import sys
        mod = sys.modules[__name__]
        if hasattr(mod, name):
            return getattr(mod, name)
        raise AttributeError(name)


from ..extras import (  # noqa: E402,F401  (PS-mode gates)
    MultiSlotDataGenerator,
    MultiSlotStringDataGenerator,
)
