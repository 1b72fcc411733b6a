"""paddle.profiler parity (reference: python/paddle/profiler/profiler.py:358).

Device tracing on MI355X runs through torch.profiler's kineto/roctracer
backend; export_chrome_tracing and the summary table mirror the
reference's chrome-trace + stat-table outputs (SURVEY.md §5 tracing).
"""
from __future__ import annotations

import enum
import os
from typing import Callable, Iterable, Optional

import torch


class ProfilerTarget(enum.Enum):
    CPU = 0
    GPU = 1


class ProfilerState(enum.Enum):
    CLOSED = 0
    READY = 1
    RECORD = 2
    RECORD_AND_RETURN = 3


class SortedKeys(enum.Enum):
    CPUTotal = 0
    CPUAvg = 1
    GPUTotal = 2
    GPUAvg = 3


def make_scheduler(*, closed: int, ready: int, record: int, repeat: int = 0,
                   skip_first: int = 0):
    return torch.profiler.schedule(wait=closed, warmup=ready, active=record,
                                   repeat=repeat, skip_first=skip_first)


def export_chrome_tracing(dir_name: str, worker_name: Optional[str] = None):
    os.makedirs(dir_name, exist_ok=True)

    def handler(prof):
        import time
        name = worker_name or f"worker_{os.getpid()}"
        prof.export_chrome_trace(os.path.join(dir_name, f"{name}_{int(time.time())}.json"))

    return handler


def export_protobuf(dir_name: str, worker_name: Optional[str] = None):
    return export_chrome_tracing(dir_name, worker_name)


class Profiler:
    def __init__(self, *, targets: Optional[Iterable] = None, scheduler=None,
                 on_trace_ready: Optional[Callable] = None, record_shapes=False,
                 profile_memory=False, timer_only=False, with_flops=False):
        activities = [torch.profiler.ProfilerActivity.CPU]
        if targets is None or any(t == ProfilerTarget.GPU for t in (targets or [])):
            if torch.cuda.is_available():
                activities.append(torch.profiler.ProfilerActivity.CUDA)
        sched = None
        if isinstance(scheduler, tuple):
            sched = torch.profiler.schedule(wait=0, warmup=scheduler[0],
                                            active=scheduler[1] - scheduler[0])
        elif scheduler is not None:
            sched = scheduler
        self._prof = torch.profiler.profile(
            activities=activities, schedule=sched,
            on_trace_ready=on_trace_ready, record_shapes=record_shapes,
            profile_memory=profile_memory, with_flops=with_flops)
        self._started = False

    def start(self):
        self._prof.__enter__()
        self._started = True

    def stop(self):
        if self._started:
            self._prof.__exit__(None, None, None)
            self._started = False

    def step(self, num_samples=None):
        self._prof.step()

    def __enter__(self):
        self.start()
        return self

    def __exit__(self, *a):
        self.stop()

    def export(self, path, format="json"):
        self._prof.export_chrome_trace(path)

    def summary(self, sorted_by=SortedKeys.CPUTotal, op_detail=True,
                thread_sep=False, time_unit="ms"):
        key = "self_cuda_time_total" if (sorted_by in (SortedKeys.GPUTotal, SortedKeys.GPUAvg)
                                         and torch.cuda.is_available()) else "cpu_time_total"
        print(self._prof.key_averages().table(sort_by=key, row_limit=30))


class RecordEvent:
    """paddle.profiler.RecordEvent -> torch.profiler.record_function
    (maps to rocTX ranges on ROCm)."""

    def __init__(self, name: str, event_type=None):
        self._rf = torch.profiler.record_function(name)

    def begin(self):
        self._rf.__enter__()

    def end(self):
        self._rf.__exit__(None, None, None)

    def __enter__(self):
        self.begin()
        return self

    def __exit__(self, *a):
        self.end()


def load_profiler_result(path):
    import json
    with open(path) as f:
        return json.load(f)


class SummaryView:
    """reference profiler/profiler.py:55 SummaryView enum."""
    DeviceView = 0
    OverView = 1
    ModelView = 2
    DistributedView = 3
    KernelView = 4
    OperatorView = 5
    MemoryView = 6
    MemoryManipulationView = 7
    UDFView = 8
