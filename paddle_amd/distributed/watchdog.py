"""Collective watchdog (reference: phi/core/distributed/comm_task_manager.cc
+ nccl_comm_task.cc: per-collective events, timeout detection, abort).

MI355X implementation: a background thread tracks registered collective
tasks (hipEvent recorded at enqueue); a task that has not completed
within `timeout` triggers the configured action (log / abort the process
group / kill).  Enabled via FLAGS_enable_async_trace like the reference.
"""
from __future__ import annotations

import threading
import time
from dataclasses import dataclass, field
from typing import Callable, List, Optional

import torch

from .. import framework


@dataclass
class _Task:
    name: str
    start_time: float
    event: Optional[object] = None  # torch.cuda.Event recorded after enqueue
    work: Optional[object] = None
    done: bool = False


class CommTaskManager:
    def __init__(self, timeout_s: float = 30 * 60, poll_interval: float = 5.0,
                 on_timeout: Optional[Callable] = None):
        self.timeout_s = timeout_s
        self.poll = poll_interval
        self.on_timeout = on_timeout or self._default_timeout
        self._tasks: List[_Task] = []
        self._lock = threading.Lock()
        self._thread: Optional[threading.Thread] = None
        self._stop = threading.Event()

    def start(self):
        if self._thread is None:
            self._thread = threading.Thread(target=self._loop, daemon=True)
            self._thread.start()

    def shutdown(self):
        self._stop.set()

    def register(self, name: str, work=None) -> _Task:
        ev = None
        if torch.cuda.is_available():
            ev = torch.cuda.Event()
            ev.record()
        t = _Task(name=name, start_time=time.time(), event=ev, work=work)
        with self._lock:
            self._tasks.append(t)
        self.start()
        return t

    def _default_timeout(self, task: _Task):
        import sys
        print(f"[watchdog] collective '{task.name}' exceeded "
              f"{self.timeout_s}s -- possible hang; aborting process group",
              file=sys.stderr, flush=True)
        try:
            import torch.distributed as dist
            if dist.is_initialized():
                dist.destroy_process_group()
        finally:
            import os
            os._exit(66)

    def _loop(self):
        while not self._stop.wait(self.poll):
            now = time.time()
            with self._lock:
                alive = []
                for t in self._tasks:
                    completed = False
                    if t.work is not None and hasattr(t.work, "is_completed"):
                        completed = t.work.is_completed()
                    elif t.event is not None:
                        completed = t.event.query()
                    else:
                        completed = True
                    if completed:
                        continue
                    if now - t.start_time > self.timeout_s:
                        self.on_timeout(t)
                    alive.append(t)
                self._tasks = alive


_manager: Optional[CommTaskManager] = None


def get_comm_task_manager() -> CommTaskManager:
    global _manager
    if _manager is None:
        _manager = CommTaskManager()
    return _manager


def watch(name: str, work=None):
    """Register a collective for watchdog tracking (no-op unless
    FLAGS_enable_async_trace is set, matching the reference)."""
    if not framework.get_flag("FLAGS_enable_async_trace"):
        return None
    return get_comm_task_manager().register(name, work)
