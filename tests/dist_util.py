"""Subprocess harness for multi-process CPU (gloo) distributed tests.

Mirrors the reference's real-subprocess test style
(test/collective/test_communication_api_base.py:28): spawn world_size
python processes with env rendezvous on 127.0.0.1, assert exit codes.
"""
from __future__ import annotations

import os
import socket
import subprocess
import sys
import textwrap

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


_RETRYABLE = ("Address already in use", "EADDRINUSE", "Connection refused",
              "Connection reset", "terminate called without an active exception")


def run_dist(script_body: str, world_size: int = 2, timeout: int = 240,
             _attempt: int = 0):
    """Run `script_body` (python source) in world_size processes over gloo.
    The body can `import paddle_amd as paddle` and use torch.distributed.
    Retries once on rendezvous races (port TIME_WAIT etc.)."""
    port = free_port()
    script = textwrap.dedent(script_body)
    # clean teardown: gloo subgroup destructors race at interpreter exit
    # ("terminate called without an active exception") without this
    script += (
        "\n\nimport torch.distributed as _dist\n"
        "if _dist.is_available() and _dist.is_initialized():\n"
        "    _dist.barrier()\n"
        "    _dist.destroy_process_group()\n")
    procs = []
    for rank in range(world_size):
        env = dict(os.environ)
        env.update({
            "RANK": str(rank),
            "LOCAL_RANK": str(rank),
            "WORLD_SIZE": str(world_size),
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port),
            "PYTHONPATH": REPO + os.pathsep + env.get("PYTHONPATH", ""),
            "CUDA_VISIBLE_DEVICES": "",
            # ROCm ignores CUDA_VISIBLE_DEVICES -- hide GPUs properly so the
            # gloo path is taken even on a GPU box
            "HIP_VISIBLE_DEVICES": "",
            "ROCR_VISIBLE_DEVICES": "",
        })
        procs.append(subprocess.Popen([sys.executable, "-c", script], env=env,
                                      stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
    outs = []
    for p in procs:
        try:
            out, _ = p.communicate(timeout=timeout)
        except subprocess.TimeoutExpired:
            for q in procs:
                q.kill()
            raise
        outs.append(out.decode())
    failed = [(rank, out) for rank, (p, out) in enumerate(zip(procs, outs))
              if p.returncode != 0]
    if failed and _attempt == 0 and any(m in out for _, out in failed
                                        for m in _RETRYABLE):
        return run_dist(script_body, world_size, timeout, _attempt=1)
    for rank, out in failed:
        raise AssertionError(f"rank {rank} failed:\n{out}")
    return outs
