"""RCCL/xGMI all-reduce bandwidth sweep (SURVEY §2.4 deliverable b).

Launch with torchrun, one rank per GPU:

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 --standalone benchmarks/bench_allreduce.py

Reports algorithmic bandwidth (2*(N-1)/N * bytes / t) per message size —
the number to compare against the xGMI per-link bound (~153 GB/s for a
ring) — plus a DDP-style bucket sweep on the ResNet-18 gradient volume.
"""
import json
import os
import sys
import time

import torch
import torch.distributed as dist

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from mi355x_scale.parallel.comm import init_distributed, destroy  # noqa: E402

SIZES_MB = [0.25, 1, 4, 16, 45, 64, 128, 256]
ITERS = 20


def bench_size(nbytes: int, device, world: int) -> float:
    x = torch.ones(nbytes // 4, dtype=torch.float32, device=device)
    for _ in range(3):
        dist.all_reduce(x)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(ITERS):
        dist.all_reduce(x)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / ITERS
    busbw = 2 * (world - 1) / world * nbytes / dt
    return busbw / 1e9


def main():
    ctx = init_distributed()
    if ctx.world_size < 2:
        print(json.dumps({"error": "needs >= 2 ranks (torchrun)"}))
        return
    device = ctx.device
    results = {}
    for mb in SIZES_MB:
        results[f"{mb}MB"] = round(
            bench_size(int(mb * 1e6) // 4 * 4, device, ctx.world_size), 2)
    # bucketed sweep over the ResNet-18 gradient volume (~45 MB):
    total = int(45e6)
    bucket_results = {}
    for bucket_mb in [4, 8, 16, 25, 45]:
        b = int(bucket_mb * 1e6)
        chunks = [min(b, total - i) for i in range(0, total, b)]
        tensors = [torch.ones(c // 4, dtype=torch.float32, device=device)
                   for c in chunks]
        for t in tensors:
            dist.all_reduce(t)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(ITERS):
            for t in tensors:
                dist.all_reduce(t)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / ITERS
        bucket_results[f"{bucket_mb}MB"] = round(dt * 1e3, 3)  # ms/step
    if ctx.rank == 0:
        print(json.dumps({
            "metric": "allreduce busbw GB/s by size",
            "world_size": ctx.world_size,
            "busbw_GBps": results,
            "resnet18_grads_ms_by_bucket": bucket_results,
        }))
    destroy()


if __name__ == "__main__":
    main()
