"""Kernel autotuning config (reference: python/paddle/incubate/autotune.py).

paddle.incubate.autotune.set_config({"kernel": {...}, "layout": {...},
"dataloader": {...}}) toggles runtime tuning.  On the MI355X stack the
"kernel" knob maps to PyTorch TunableOp (hipBLASLt algorithm search) and
MIOpen benchmark mode; "layout" maps to channels-last autotuning for
conv workloads; "dataloader" tunes worker count.
"""
from __future__ import annotations

import json

import torch

_config = {
    "kernel": {"enable": False, "tuning_range": [1, 10]},
    "layout": {"enable": False},
    "dataloader": {"enable": False, "tuning_steps": 500},
}


def set_config(config=None):
    """Enable/disable autotuning subsystems.

    config may be a dict or a path to a JSON file (paddle parity).
    """
    global _config
    if config is None:
        for k in _config:
            _config[k]["enable"] = True
    else:
        if isinstance(config, str):
            with open(config) as f:
                config = json.load(f)
        for k, v in config.items():
            _config.setdefault(k, {}).update(v)

    kern = _config.get("kernel", {})
    if kern.get("enable"):
        try:
            import torch.cuda.tunable as tunable
            tunable.enable(True)
            tunable.tuning_enable(True)
            rng = kern.get("tuning_range", [1, 10])
            tunable.set_max_tuning_iterations(int(rng[-1]))
        except Exception:
            pass
        torch.backends.cudnn.benchmark = True  # MIOpen find-mode
    return dict(_config)


def get_config():
    return {k: dict(v) for k, v in _config.items()}
