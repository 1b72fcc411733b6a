"""CommunicateTopology / HybridCommunicateGroup.

Reference: python/paddle/distributed/fleet/base/topology.py:70,189.
Axis order [data, pipe, sharding, sep, model] preserved exactly --
rank placement and group membership must match the reference so
checkpoints and launch scripts interchange.
"""
from __future__ import annotations

import itertools
from functools import reduce

import numpy as np

from .. import collective as C
from ..parallel import get_rank, get_world_size


class CommunicateTopology:
    def __init__(self, hybrid_group_names=("data", "pipe", "sharding", "sep", "model"),
                 dims=(1, 1, 1, 1, 1)):
        self._parallel_names = list(hybrid_group_names)
        self._dims = list(dims)
        self.coordinate = list(itertools.product(*(range(d) for d in dims)))
        self._word_size = reduce(lambda a, b: a * b, dims)
        self._rank2coord = {self._coord_to_rank(c): c for c in self.coordinate}
        self._coord2rank = {c: self._coord_to_rank(c) for c in self.coordinate}

    def _coord_to_rank(self, coord):
        r = 0
        for i, c in enumerate(coord):
            r = r * self._dims[i] + c
        return r

    def get_hybrid_group_names(self):
        return self._parallel_names

    def get_dim(self, axis_name):
        return self._dims[self._parallel_names.index(axis_name)]

    get_dim_size = get_dim

    def world_size(self):
        return self._word_size

    def get_rank(self, **kwargs):
        coord = tuple(kwargs[n] for n in self._parallel_names)
        return self._coord2rank[coord]

    def get_coord(self, rank):
        return self._rank2coord[rank]

    def get_axis_list(self, axis_name, index):
        """all ranks whose coord on axis == index"""
        axis = self._parallel_names.index(axis_name)
        return [r for c, r in self._coord2rank.items() if c[axis] == index]

    def get_comm_list(self, axis_name):
        """list of rank-lists: each varies only along axis_name"""
        axis = self._parallel_names.index(axis_name)
        other = [i for i in range(len(self._dims)) if i != axis]
        groups = []
        for combo in itertools.product(*(range(self._dims[i]) for i in other)):
            ranks = []
            for v in range(self._dims[axis]):
                coord = list(combo)
                coord.insert(axis, v)
                ranks.append(self._coord2rank[tuple(coord)])
            groups.append(ranks)
        return groups

    def get_rank_from_stage(self, global_rank, **kwargs):
        coord = list(self.get_coord(global_rank))
        for k, v in kwargs.items():
            coord[self._parallel_names.index(k)] = v
        return self._coord2rank[tuple(coord)]


class HybridCommunicateGroup:
    def __init__(self, topology: CommunicateTopology):
        self._topo = topology
        self.global_rank = get_rank()
        self.nranks = get_world_size()
        assert self.nranks == topology.world_size(), (
            f"world size {self.nranks} != topology {topology.world_size()}")
        self._dp_degree = topology.get_dim("data")
        self._pp_degree = topology.get_dim("pipe")
        self._sharding_degree = topology.get_dim("sharding")
        self._sep_degree = topology.get_dim("sep") if "sep" in topology.get_hybrid_group_names() else 1
        self._mp_degree = topology.get_dim("model")

        self._dp_group, self._dp_comm_group = self._build("data")
        self._pp_group, self._pp_comm_group = self._build("pipe")
        self._sharding_group, self._sharding_comm_group = self._build("sharding")
        if "sep" in topology.get_hybrid_group_names():
            self._sep_group, self._sep_comm_group = self._build("sep")
        else:
            self._sep_group, self._sep_comm_group = None, None
        self._mp_group, self._mp_comm_group = self._build("model")
        # p2p neighbors for pipeline
        self._p2p_next, self._p2p_prev = self._build_p2p()

    def _build(self, axis):
        if not C.is_initialized() or self.nranks == 1:
            return [self.global_rank], None
        comm_lists = self._topo.get_comm_list(axis)
        my_group = None
        my_ranks = None
        for ranks in comm_lists:
            g = C.new_group(ranks)
            if self.global_rank in ranks:
                my_group = g
                my_ranks = ranks
        return my_ranks, my_group

    def _build_p2p(self):
        if self._pp_degree <= 1:
            return None, None
        stage = self.stage_id
        ranks = self._pp_group
        i = ranks.index(self.global_rank)
        nxt = ranks[(i + 1) % len(ranks)]
        prv = ranks[(i - 1) % len(ranks)]
        return nxt, prv

    # -- info ----------------------------------------------------------------
    @property
    def stage_id(self):
        return self._topo.get_coord(self.global_rank)[
            self._topo.get_hybrid_group_names().index("pipe")]

    def topology(self):
        return self._topo

    def get_global_rank(self):
        return self.global_rank

    # data parallel
    def get_data_parallel_rank(self):
        return self._topo.get_coord(self.global_rank)[0]

    def get_data_parallel_world_size(self):
        return self._dp_degree

    def get_data_parallel_group(self):
        return self._dp_comm_group

    def get_data_parallel_group_src_rank(self):
        return self._dp_group[0]

    # model parallel
    def get_model_parallel_rank(self):
        names = self._topo.get_hybrid_group_names()
        return self._topo.get_coord(self.global_rank)[names.index("model")]

    def get_model_parallel_world_size(self):
        return self._mp_degree

    def get_model_parallel_group(self):
        return self._mp_comm_group

    def get_model_parallel_group_src_rank(self):
        return self._mp_group[0]

    # pipe
    def get_stage_id(self):
        return self.stage_id

    def get_pipe_parallel_rank(self):
        return self.stage_id

    def get_pipe_parallel_world_size(self):
        return self._pp_degree

    def get_pipe_parallel_group(self):
        return self._pp_comm_group

    def get_p2p_next_rank(self):
        return self._p2p_next

    def get_p2p_prev_rank(self):
        return self._p2p_prev

    def is_first_stage(self):
        return self.stage_id == 0

    def is_last_stage(self):
        return self.stage_id == self._pp_degree - 1

    # sharding
    def get_sharding_parallel_rank(self):
        names = self._topo.get_hybrid_group_names()
        return self._topo.get_coord(self.global_rank)[names.index("sharding")]

    def get_sharding_parallel_world_size(self):
        return self._sharding_degree

    def get_sharding_parallel_group(self):
        return self._sharding_comm_group

    def get_sharding_parallel_group_src_rank(self):
        return self._sharding_group[0]

    # sep
    def get_sep_parallel_rank(self):
        if self._sep_group is None:
            return 0
        names = self._topo.get_hybrid_group_names()
        return self._topo.get_coord(self.global_rank)[names.index("sep")]

    def get_sep_parallel_world_size(self):
        return self._sep_degree

    def get_sep_parallel_group(self):
        return self._sep_comm_group
