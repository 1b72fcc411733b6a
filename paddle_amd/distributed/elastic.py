"""Elastic training manager (reference: fleet/elastic/manager.py:125
ElasticManager -- etcd-registered scale in/out, fault watch, restart).

MI355X round-1 scope: file/TCPStore-based membership (etcd optional and
absent in this environment), heartbeat lease, watch loop that triggers
restart callbacks on membership change -- the same three recovery tiers
as the reference (watchdog abort -> launcher restart -> checkpoint
resume)."""
from __future__ import annotations

import json
import os
import threading
import time
from typing import Callable, List, Optional


class ElasticLevel:
    NONE = 0
    FAULT_TOLERANCE = 1
    ELASTIC = 2


class ElasticManager:
    def __init__(self, job_id="default", np=1, host=None, scale=0,
                 force=False, backend="store", store_dir=None,
                 heartbeat_interval=5.0, lease_ttl=15.0):
        self.job_id = job_id
        self.np = np
        self.host = host or f"127.0.0.1:{os.getpid()}"
        self.store_dir = store_dir or os.path.join("/tmp", f"elastic_{job_id}")
        os.makedirs(self.store_dir, exist_ok=True)
        self.hb_interval = heartbeat_interval
        self.lease_ttl = lease_ttl
        self._stop = threading.Event()
        self._threads: List[threading.Thread] = []
        self._on_change: Optional[Callable] = None
        self.enabled = True

    def _my_key(self):
        return os.path.join(self.store_dir, self.host.replace(":", "_") + ".node")

    def register(self):
        self._heartbeat()
        t = threading.Thread(target=self._hb_loop, daemon=True)
        t.start()
        self._threads.append(t)

    def _heartbeat(self):
        # atomic lease refresh: open("w") truncates in place, so a
        # concurrent hosts() read can see a partial file and silently
        # drop the host (observed as membership flaps under load)
        key = self._my_key()
        tmp = key + f".tmp{os.getpid()}"
        with open(tmp, "w") as f:
            json.dump({"host": self.host, "ts": time.time()}, f)
        os.replace(tmp, key)

    def _hb_loop(self):
        while not self._stop.wait(self.hb_interval):
            self._heartbeat()

    def hosts(self) -> List[str]:
        now = time.time()
        out = []
        for fn in os.listdir(self.store_dir):
            if not fn.endswith(".node"):
                continue
            try:
                with open(os.path.join(self.store_dir, fn)) as f:
                    rec = json.load(f)
                if now - rec["ts"] <= self.lease_ttl:
                    out.append(rec["host"])
            except Exception:
                pass
        return sorted(out)

    def watch(self, on_change: Callable[[List[str]], None]):
        self._on_change = on_change
        t = threading.Thread(target=self._watch_loop, daemon=True)
        t.start()
        self._threads.append(t)

    def _watch_loop(self):
        prev = self.hosts()
        while not self._stop.wait(self.hb_interval):
            cur = self.hosts()
            if cur != prev:
                if self._on_change:
                    self._on_change(cur)
                prev = cur

    def exit(self):
        self._stop.set()
        try:
            os.remove(self._my_key())
        except OSError:
            pass


# ---------------------------------------------------------------------------
# scale in/out (reference fleet/elastic/manager.py:484 _update_elastic_scale_out
# / :507 _update_elastic_scale_in): membership changes within [min_np, max_np]
# produce a new world plan; the agent relaunches workers with the new size.
# ---------------------------------------------------------------------------
def parse_np_range(np_spec):
    """'2' -> (2, 2); '2:8' -> (2, 8) (reference --np min:max form)."""
    if isinstance(np_spec, int):
        return np_spec, np_spec
    s = str(np_spec)
    if ":" in s:
        lo, hi = s.split(":")
        return int(lo), int(hi)
    return int(s), int(s)


class ScalePlan:
    def __init__(self, action, world, hosts):
        self.action = action      # "none" | "scale_out" | "scale_in" | "wait"
        self.world = world
        self.hosts = hosts

    def __repr__(self):
        return f"ScalePlan({self.action}, world={self.world}, hosts={self.hosts})"


class ElasticScaler:
    """Turns membership snapshots into scale decisions with a debounce
    window so transient heartbeat blips don't thrash restarts."""

    def __init__(self, manager: "ElasticManager", np_spec, debounce=2.0):
        self.mgr = manager
        self.min_np, self.max_np = parse_np_range(np_spec)
        self.debounce = debounce
        self.current_world = None
        self._pending = None
        self._pending_since = 0.0

    def plan(self, hosts=None, now=None) -> ScalePlan:
        hosts = self.mgr.hosts() if hosts is None else sorted(hosts)
        now = time.time() if now is None else now
        n = len(hosts)
        if self.current_world is None:
            world = max(self.min_np, min(n, self.max_np))
            return ScalePlan("none", world, hosts[:world])
        if n == self.current_world:
            self._pending = None
            return ScalePlan("none", self.current_world, hosts)
        # debounce: a change must persist for `debounce` seconds
        if self._pending != n:
            self._pending = n
            self._pending_since = now
            return ScalePlan("wait", self.current_world, hosts)
        if now - self._pending_since < self.debounce:
            return ScalePlan("wait", self.current_world, hosts)
        self._pending = None
        if n > self.current_world and self.current_world < self.max_np:
            world = min(n, self.max_np)
            return ScalePlan("scale_out", world, hosts[:world])
        if n < self.current_world:
            if n >= self.min_np:
                return ScalePlan("scale_in", n, hosts)
            return ScalePlan("wait", self.current_world, hosts)  # below min: hold
        return ScalePlan("none", self.current_world, hosts)

    def commit(self, plan: ScalePlan):
        self.current_world = plan.world


class LocalElasticAgent:
    """Launcher-tier elastic agent: runs `nproc` local workers, watches
    membership, and on a committed scale decision relaunches the worker
    set at the new world size (restart-with-checkpoint-resume semantics,
    the reference's recovery tier 2)."""

    def __init__(self, manager, np_spec, entry_args, debounce=2.0,
                 python=None):
        import sys as _sys
        self.mgr = manager
        self.scaler = ElasticScaler(manager, np_spec, debounce)
        self.entry_args = entry_args        # argv after `python`
        self.python = python or _sys.executable
        self.procs = []
        self.restarts = 0

    def _launch(self, world):
        import subprocess
        self._stop_workers()
        env_base = dict(os.environ)
        env_base["PADDLE_ELASTIC_RESTART"] = str(self.restarts)
        for r in range(world):
            env = dict(env_base)
            env.update({"RANK": str(r), "WORLD_SIZE": str(world),
                        "MASTER_ADDR": "127.0.0.1",
                        "PADDLE_TRAINER_ID": str(r),
                        "PADDLE_TRAINERS_NUM": str(world)})
            self.procs.append(__import__("subprocess").Popen(
                [self.python] + self.entry_args, env=env))

    def _stop_workers(self):
        for p in self.procs:
            if p.poll() is None:
                p.terminate()
        for p in self.procs:
            try:
                p.wait(timeout=10)
            except Exception:
                p.kill()
        self.procs = []

    def check_failures(self):
        """Recovery tier 2 (reference elastic/manager.py fault watch): a
        worker that exited nonzero poisons the incarnation -- stop the
        rest and relaunch the full set at the same world size (workers
        resume from their own checkpoints)."""
        if not self.procs:
            return False
        failed = any(p.poll() is not None and p.poll() != 0 for p in self.procs)
        if failed:
            world = len(self.procs)
            self.restarts += 1
            self._launch(world)
        return failed

    def step(self, hosts=None, now=None):
        """One watch iteration; returns the plan taken."""
        if self.check_failures():
            return ScalePlan("restart", len(self.procs),
                             self.scaler.last_hosts
                             if hasattr(self.scaler, "last_hosts") else None)
        plan = self.scaler.plan(hosts, now)
        if plan.action in ("scale_out", "scale_in") or \
                (plan.action == "none" and self.scaler.current_world is None):
            self.scaler.commit(plan)
            self.restarts += (self.restarts >= 0 and plan.action != "none")
            self._launch(plan.world)
        return plan

    def shutdown(self):
        self._stop_workers()
        self.mgr.exit()
