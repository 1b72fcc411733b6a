#!/usr/bin/env python3
"""xGMI collective microbench: busbw for all_reduce / reduce_scatter /
all_gather / all_to_all at the framework's actual bucket sizes.

SURVEY §5: the 8-GPU xGMI node is a full mesh (7 x ~153 GB/s links per
GPU).  Ring collectives are per-link bound, so RCCL needs >= 7 channels
to use all links; this harness measures achieved busbw per algorithm and
size so the ring-vs-direct (one-shot) decision and the reducer bucket
size (distributed/parallel.py, 128 MB default) are data, not guesses.

Launch (driver or by hand):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \\
      --master-addr 127.0.0.1 bench_collectives.py [--sizes-mb 4,32,128]
Env knobs probed: NCCL_MIN_NCHANNELS / NCCL_MAX_NCHANNELS (ring count).

busbw conventions (nccl-tests): AR 2(n-1)/n x size/t; RS/AG (n-1)/n x
size/t; A2A (n-1)/n x size/t.
"""
import argparse
import json
import os
import time

import torch
import torch.distributed as dist


def bench_op(fn, iters, warmup):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    dist.barrier()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--sizes-mb", default="1,4,16,64,128,256")
    ap.add_argument("--iters", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--out", default="gpurun_out/collectives.json")
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local = int(os.environ.get("LOCAL_RANK", "0"))
    torch.cuda.set_device(local % torch.cuda.device_count())
    dist.init_process_group("nccl")
    n = world
    results = {"world": n, "device": torch.cuda.get_device_name(0),
               "nccl_env": {k: v for k, v in os.environ.items()
                            if k.startswith("NCCL_") or k.startswith("RCCL_")},
               "rows": []}
    for mb in [float(s) for s in args.sizes_mb.split(",")]:
        numel = int(mb * 1e6 / 2)  # bf16
        x = torch.randn(numel, device="cuda", dtype=torch.bfloat16)
        out_full = torch.empty(numel * n, device="cuda", dtype=torch.bfloat16)
        shard = torch.empty(numel // n * n, device="cuda", dtype=torch.bfloat16)
        size_b = numel * 2

        t_ar = bench_op(lambda: dist.all_reduce(x), args.iters, args.warmup)
        t_ag = bench_op(lambda: dist.all_gather_into_tensor(out_full, x),
                        args.iters, args.warmup)
        rs_out = torch.empty(numel // n, device="cuda", dtype=torch.bfloat16)
        t_rs = bench_op(lambda: dist.reduce_scatter_tensor(rs_out, shard),
                        args.iters, args.warmup)
        a2a_out = torch.empty_like(shard)
        t_a2a = bench_op(lambda: dist.all_to_all_single(a2a_out, shard),
                         args.iters, args.warmup)
        row = {
            "size_mb": mb,
            "allreduce_busbw_gbs": 2 * (n - 1) / n * size_b / t_ar / 1e9,
            "allgather_busbw_gbs": (n - 1) / n * size_b * n / t_ag / 1e9,
            "reducescatter_busbw_gbs": (n - 1) / n * size_b / t_rs / 1e9,
            "alltoall_busbw_gbs": (n - 1) / n * size_b / t_a2a / 1e9,
            "allreduce_ms": t_ar * 1e3,
        }
        results["rows"].append(row)
        if rank == 0:
            print(f"{mb:7.1f} MB  AR {row['allreduce_busbw_gbs']:7.1f}  "
                  f"AG {row['allgather_busbw_gbs']:7.1f}  "
                  f"RS {row['reducescatter_busbw_gbs']:7.1f}  "
                  f"A2A {row['alltoall_busbw_gbs']:7.1f}  GB/s busbw")
    if rank == 0:
        os.makedirs(os.path.dirname(args.out) or ".", exist_ok=True)
        with open(args.out, "w") as f:
            json.dump(results, f, indent=1)
        # decision hint: ring is per-link bound at ~153 GB/s/link;
        # full-mesh one-shot RS/AG is worth hand-writing if measured busbw
        # at 64-256 MB stays well under 7*153*(n-1)/n
        if n > 1:
            peak = max(r["allreduce_busbw_gbs"] for r in results["rows"])
            target = 153.0 * 7 * (n - 1) / n
            print(f"\npeak AR busbw {peak:.0f} GB/s vs full-mesh bound "
                  f"~{target:.0f} GB/s -> "
                  + ("rings saturate the mesh; keep RCCL rings"
                     if peak > 0.6 * target else
                     "rings leave links idle; raise NCCL_MIN_NCHANNELS>=7 "
                     "or use the one-shot direct RS/AG"))
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
