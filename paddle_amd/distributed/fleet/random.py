"""TP RNG state tracker (reference: fleet/layers/mpu/random.py:34
RNGStatesTracker, :103 model_parallel_random_seed).

Seed derivation replicated exactly: global_seed = seed,
local_seed = seed + 1 + mp_rank * pp_size + pp_rank.
"""
from __future__ import annotations

import contextlib

import torch

MODEL_PARALLEL_RNG = "model_parallel_rng"


class RNGStatesTracker:
    def __init__(self):
        self.states_ = {}
        self.seeds_ = set()

    def reset(self):
        self.states_ = {}
        self.seeds_ = set()

    def add(self, name, seed):
        if seed in self.seeds_:
            raise ValueError(f"seed {seed} already exists")
        self.seeds_.add(seed)
        if name in self.states_:
            raise ValueError(f"state {name} already exists")
        if torch.cuda.is_available():
            orig = torch.cuda.get_rng_state()
            torch.cuda.manual_seed(seed)
            self.states_[name] = torch.cuda.get_rng_state()
            torch.cuda.set_rng_state(orig)
        else:
            orig = torch.get_rng_state()
            torch.manual_seed(seed)
            self.states_[name] = torch.get_rng_state()
            torch.set_rng_state(orig)

    @contextlib.contextmanager
    def rng_state(self, name=MODEL_PARALLEL_RNG):
        if name not in self.states_:
            yield
            return
        cuda = torch.cuda.is_available()
        get_s = torch.cuda.get_rng_state if cuda else torch.get_rng_state
        set_s = torch.cuda.set_rng_state if cuda else torch.set_rng_state
        orig = get_s()
        set_s(self.states_[name])
        try:
            yield
        finally:
            self.states_[name] = get_s()
            set_s(orig)

    def get_states_tracker(self):
        return dict(self.states_)

    def set_states_tracker(self, states):
        self.states_ = states


_RNG_STATE_TRACKER = RNGStatesTracker()


def get_rng_state_tracker():
    return _RNG_STATE_TRACKER


def model_parallel_random_seed(seed=None):
    import random
    from . import get_hybrid_communicate_group
    hcg = get_hybrid_communicate_group()
    rank = hcg.get_model_parallel_rank() if hcg else 0
    pp_rank = hcg.get_pipe_parallel_rank() if hcg else 0
    pp_size = hcg.get_pipe_parallel_world_size() if hcg else 1
    if seed is None:
        seed = random.randint(0, 2 ** 31 - 1)
    global_seed = seed
    local_seed = seed + 1 + rank * pp_size + pp_rank  # mpu/random.py:103
    _RNG_STATE_TRACKER.reset()
    torch.manual_seed(global_seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed(global_seed)
    _RNG_STATE_TRACKER.add(MODEL_PARALLEL_RNG, local_seed)


def dropout(x, p=0.5, training=True, mode="upscale_in_train", rng_name=None, name=None):
    """mpu/random.py:127 parity: dropout under a named rng state."""
    if rng_name is None or not training or p == 0:
        import torch.nn.functional as TF
        return TF.dropout(x, p, training) if training else x
    with _RNG_STATE_TRACKER.rng_state(rng_name):
        import torch.nn.functional as TF
        return TF.dropout(x, p, training)
