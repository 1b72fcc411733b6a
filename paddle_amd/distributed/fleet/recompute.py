"""Activation recomputation (reference: fleet/recompute/recompute.py:124
RecomputeFunction + :319 non-reentrant variant).

Implemented over torch.utils.checkpoint (non-reentrant, deterministic RNG
replay) with the TP RNGStatesTracker state captured/restored so dropout
inside TP regions replays identically (recompute_hybrid.py semantics).
"""
from __future__ import annotations

import torch

from .random import get_rng_state_tracker


def recompute(function, *args, use_reentrant=False, preserve_rng_state=True, **kwargs):
    tracker = get_rng_state_tracker()
    states = tracker.get_states_tracker()

    def wrapped(*inner):
        tracker.set_states_tracker(dict(states))
        return function(*inner, **kwargs)

    return torch.utils.checkpoint.checkpoint(
        wrapped, *args, use_reentrant=use_reentrant,
        preserve_rng_state=preserve_rng_state)


def recompute_sequential(ctx, functions, *args):
    for fn in functions:
        args = (recompute(fn, *args),)
    return args[0]
