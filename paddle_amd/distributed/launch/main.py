"""python -m paddle_amd.distributed.launch -- collective launcher CLI.

Reference: python/paddle/distributed/launch/main.py:23 + controllers/
collective.py (CollectiveController.build_pod): one subprocess per
device with both PADDLE_TRAINER_* and torch-style RANK/WORLD_SIZE env,
rendezvous on MASTER_ADDR/PORT, per-rank logs under --log_dir, failure
watch (any rank dies -> kill pod, non-zero exit).
"""
from __future__ import annotations

import argparse
import os
import signal
import subprocess
import sys
import time


def _device_count():
    try:
        import torch
        if torch.cuda.is_available():
            return torch.cuda.device_count()
    except Exception:
        pass
    return 1


def parse_args(argv=None):
    ap = argparse.ArgumentParser(prog="paddle_amd.distributed.launch")
    ap.add_argument("--devices", "--gpus", type=str, default=None,
                    help="e.g. 0,1,2,3 (default: all visible)")
    ap.add_argument("--nnodes", type=str, default="1")
    ap.add_argument("--nproc_per_node", type=int, default=None)
    ap.add_argument("--master", type=str, default=None, help="host:port")
    ap.add_argument("--master_addr", type=str, default="127.0.0.1")
    ap.add_argument("--master_port", type=int, default=29765)
    ap.add_argument("--rank", type=int, default=0, help="node rank")
    ap.add_argument("--log_dir", type=str, default="log")
    ap.add_argument("--gpu_watch_interval", type=float, default=0.0,
                    help="seconds between GPU-utilization log lines "
                         "(reference launch/controllers/watcher.py; 0 = off)")
    ap.add_argument("--job_id", type=str, default="default")
    ap.add_argument("training_script", type=str)
    ap.add_argument("training_script_args", nargs=argparse.REMAINDER)
    return ap.parse_args(argv)


def launch(argv=None):
    args = parse_args(argv)
    if args.devices:
        devices = [d for d in args.devices.split(",") if d != ""]
    else:
        devices = [str(i) for i in range(args.nproc_per_node or _device_count())]
    nproc = args.nproc_per_node or len(devices)
    nnodes = int(str(args.nnodes).split(":")[0])
    if args.master:
        addr, port = args.master.rsplit(":", 1)
    else:
        addr, port = args.master_addr, str(args.master_port)
    world = nproc * nnodes

    os.makedirs(args.log_dir, exist_ok=True)
    watcher = _GPUWatcher(args.log_dir, args.gpu_watch_interval)
    watcher.start()
    procs = []
    logs = []
    for local_rank in range(nproc):
        rank = args.rank * nproc + local_rank
        env = dict(os.environ)
        env.update({
            "MASTER_ADDR": addr,
            "MASTER_PORT": str(port),
            "RANK": str(rank),
            "LOCAL_RANK": str(local_rank),
            "WORLD_SIZE": str(world),
            "LOCAL_WORLD_SIZE": str(nproc),
            # paddle-style env for scripts that read it
            "PADDLE_TRAINER_ID": str(rank),
            "PADDLE_TRAINERS_NUM": str(world),
            "PADDLE_RANK_IN_NODE": str(local_rank),
            "PADDLE_MASTER": f"{addr}:{port}",
            "PADDLE_LOCAL_SIZE": str(nproc),
            "FLAGS_selected_gpus": devices[local_rank % len(devices)],
            "HIP_VISIBLE_DEVICES": env.get("HIP_VISIBLE_DEVICES",
                                           ",".join(devices)),
        })
        logf = open(os.path.join(args.log_dir, f"workerlog.{local_rank}"), "w")
        logs.append(logf)
        cmd = [sys.executable, "-u", args.training_script] + args.training_script_args
        p = subprocess.Popen(cmd, env=env, stdout=logf if local_rank != 0 else None,
                             stderr=subprocess.STDOUT if local_rank != 0 else None)
        procs.append(p)

    code = 0
    try:
        while True:
            alive = 0
            for p in procs:
                rc = p.poll()
                if rc is None:
                    alive += 1
                elif rc != 0:
                    code = rc
            if code != 0:
                for p in procs:
                    if p.poll() is None:
                        p.send_signal(signal.SIGTERM)
                time.sleep(3)
                for p in procs:
                    if p.poll() is None:
                        p.kill()
                break
            if alive == 0:
                break
            time.sleep(0.5)
    except KeyboardInterrupt:
        for p in procs:
            if p.poll() is None:
                p.terminate()
        code = 130
    finally:
        watcher.stop()
        for f in logs:
            f.close()
    return code


class _GPUWatcher:
    """Periodic GPU utilization/VRAM logger (reference
    launch/controllers/watcher.py) -- rocm-smi when present, else
    torch.cuda memory counters; writes log_dir/gpulog."""

    def __init__(self, log_dir, interval):
        self.interval = interval
        self.path = os.path.join(log_dir, "gpulog")
        self._stop = None
        self._thread = None

    def start(self):
        if not self.interval or self.interval <= 0:
            return
        import threading
        self._stop = threading.Event()
        self._thread = threading.Thread(target=self._run, daemon=True)
        self._thread.start()

    def _sample(self):
        import shutil
        if shutil.which("rocm-smi"):
            try:
                out = subprocess.run(
                    ["rocm-smi", "--showuse", "--showmemuse", "--csv"],
                    capture_output=True, text=True, timeout=10).stdout
                return out.strip()
            except Exception:
                pass
        try:
            import torch
            if torch.cuda.is_available():
                return " ".join(
                    f"gpu{i}:alloc={torch.cuda.memory_allocated(i)>>20}MiB"
                    for i in range(torch.cuda.device_count()))
        except Exception:
            pass
        return "n/a"

    def _run(self):
        with open(self.path, "a") as f:
            while not self._stop.wait(self.interval):
                f.write(f"{time.time():.1f} {self._sample()}\n")
                f.flush()

    def stop(self):
        if self._stop is not None:
            self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=5)


if __name__ == "__main__":
    sys.exit(launch())
