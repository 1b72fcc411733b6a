"""Parallel-config auto-tuner (reference: python/paddle/distributed/
auto_tuner/{tuner,search,prune,recorder}.py).

Grid-searches (dp, mp, pp, sharding-stage, micro-batch) combinations,
prunes infeasible ones by divisibility/memory heuristics, launches
trials, records tokens/s, and reports the best config.
"""
from __future__ import annotations

import itertools
import json
import os
import subprocess
import sys
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional


@dataclass
class TunerConfig:
    world_size: int = 8
    model_params_b: float = 6.7        # billions
    hbm_gb: float = 288.0
    candidates: Dict = field(default_factory=lambda: {
        "dp_degree": [1, 2, 4, 8],
        "mp_degree": [1, 2, 4],
        "pp_degree": [1, 2],
        "sharding_stage": [1, 2, 3],
        "micro_batch": [1, 2, 4, 8],
    })


def prune(cfg: TunerConfig, trial: Dict) -> Optional[str]:
    """Return a reason string if the trial is infeasible (reference
    prune.py rules: divisibility + memory)."""
    dp, mp, pp = trial["dp_degree"], trial["mp_degree"], trial["pp_degree"]
    if dp * mp * pp != cfg.world_size:
        return "dp*mp*pp != world_size"
    if trial["sharding_stage"] > 1 and dp == 1:
        return "sharding needs dp>1"
    # memory heuristic: params(bf16)+grads+opt(fp32 m,v,master) per GPU
    p = cfg.model_params_b * 1e9 / (mp * pp)
    shard = dp if trial["sharding_stage"] >= 1 else 1
    mem = (2 * p  # weights
           + 2 * p / (shard if trial["sharding_stage"] >= 2 else 1)  # grads
           + 12 * p / shard)  # master+m+v
    mem_gb = mem / 2 ** 30
    if mem_gb > cfg.hbm_gb * 0.85:
        return f"est mem {mem_gb:.0f} GB > budget"
    return None


def search_space(cfg: TunerConfig) -> List[Dict]:
    keys = list(cfg.candidates.keys())
    out = []
    for combo in itertools.product(*(cfg.candidates[k] for k in keys)):
        trial = dict(zip(keys, combo))
        if prune(cfg, trial) is None:
            out.append(trial)
    return out


class Recorder:
    def __init__(self, path="autotuner_history.jsonl"):
        self.path = path
        self.records = []

    def add(self, trial, metric):
        rec = {"trial": trial, "tokens_per_s": metric, "ts": time.time()}
        self.records.append(rec)
        with open(self.path, "a") as f:
            f.write(json.dumps(rec) + "\n")

    def best(self):
        if not self.records:
            return None
        return max(self.records, key=lambda r: r["tokens_per_s"] or 0)


class AutoTuner:
    def __init__(self, cfg: TunerConfig, launch_cmd_fn=None, recorder=None):
        self.cfg = cfg
        self.launch_cmd_fn = launch_cmd_fn or self._default_cmd
        self.recorder = recorder or Recorder()

    def _default_cmd(self, trial):
        return [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
                f"--nproc-per-node={self.cfg.world_size}",
                "--master-addr", "127.0.0.1", "bench.py",
                "--batch", str(trial["micro_batch"]),
                "--sharding-stage", str(trial["sharding_stage"])]

    def run_trial(self, trial, timeout=600):
        cmd = self.launch_cmd_fn(trial)
        try:
            r = subprocess.run(cmd, capture_output=True, text=True, timeout=timeout)
            for line in reversed(r.stdout.splitlines()):
                if line.startswith("{"):
                    return json.loads(line).get("value")
        except Exception:
            return None
        return None

    def tune(self, max_trials=None):
        trials = search_space(self.cfg)
        if max_trials:
            trials = trials[:max_trials]
        for t in trials:
            metric = self.run_trial(t)
            self.recorder.add(t, metric)
        return self.recorder.best()
