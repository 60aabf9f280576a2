"""BASELINE config 4: fine-grained demand forecasting — 100k store-SKU
groups batch-fit on MI355X. Metric: groups/sec (whole job: candidate
evaluation over the (p,d,q) grid + best-per-group final fit).

    python benchmarks/bench_groupfit.py --groups 100000 --candidates 10

Multi-GPU (groups sharded across ranks, one rank per GPU):

    python -m torch.distributed.run --nproc-per-node 8 --nnodes 1 \
        --master-addr 127.0.0.1 --standalone \
        benchmarks/bench_groupfit.py --groups 100000
"""
import argparse
import json
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from mi355x_scale.data.generator import demand_factors, _week_dates  # noqa: E402
from mi355x_scale.forecast.batched import batched_fit_gpu, batched_fit_reference  # noqa: E402

CANDIDATES = [(0, 1, 0), (1, 0, 0), (1, 1, 0), (0, 1, 1), (1, 1, 1),
              (2, 1, 0), (2, 0, 1), (0, 2, 1), (2, 1, 2), (4, 1, 2)]


def synth_groups(G: int, T: int, seed: int = 123):
    """Vectorized synthetic demand panel (AR(1)+trend+seasonal factors),
    seed-123 like the reference generator but sized for 100k groups."""
    rng = np.random.default_rng(seed)
    dates = _week_dates(T)
    fac = demand_factors(dates)
    base = rng.uniform(400, 1200, size=(G, 1))
    trend = rng.uniform(2, 8, size=(G, 1))
    phi = rng.uniform(0.3, 0.8, size=(G, 1))
    e = rng.standard_normal((G, T)) * (base * 0.05)
    u = np.zeros((G, T))
    for t in range(1, T):
        u[:, t] = phi[:, 0] * u[:, t - 1] + e[:, t]
    t_ax = np.arange(T)[None, :]
    y = base + trend * np.sqrt(t_ax) + u
    y *= (1 - 0.5 * fac["covid"].to_numpy()[None, :])
    y *= (1 + 0.35 * fac["christmas"].to_numpy()[None, :])
    y = np.clip(np.round(y), 0, None)
    exog = fac[["covid", "christmas", "new_year"]].to_numpy()
    return y.astype(np.float32), exog


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--groups", type=int, default=100_000)
    ap.add_argument("--weeks", type=int, default=157)
    ap.add_argument("--train-len", type=int, default=117)
    ap.add_argument("--candidates", type=int, default=10)
    ap.add_argument("--repeats", type=int, default=3)
    ap.add_argument("--cpu-oracle-groups", type=int, default=0,
                    help="also run the numpy oracle on this many groups "
                         "for a throughput comparison")
    ap.add_argument("--end-to-end", action="store_true",
                    help="also time the whole DataFrame->forecast-frame "
                         "job (C++ gather + GPU fit + frame rebuild)")
    args = ap.parse_args()
    orders = CANDIDATES[:args.candidates]

    # multi-rank: shard the group axis, one rank per GPU (config 4's
    # "100k store-SKU groups batch-fit sharded across 8 GPUs")
    import torch.distributed as dist
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    if world > 1:
        from mi355x_scale.parallel.comm import init_distributed
        ctx = init_distributed()
        world, rank = ctx.world_size, ctx.rank

    y, exog = synth_groups(args.groups, args.weeks)
    if world > 1:
        per = (args.groups + world - 1) // world
        y = y[rank * per:(rank + 1) * per]
    if not torch.cuda.is_available():
        t0 = time.perf_counter()
        batched_fit_reference(y[:64], exog, orders, args.train_len)
        dt = time.perf_counter() - t0
        if rank == 0:
            print(json.dumps({"error": "no GPU; oracle only"}))
            print(json.dumps({"metric": "groups/sec", "value": 64 / dt,
                              "n_gpus": 0, "path": "numpy-oracle"}))
        return

    # warmup (includes H2D and design prep)
    out = batched_fit_gpu(y[:1024], exog, orders, args.train_len)
    torch.cuda.synchronize()

    times = []
    for _ in range(args.repeats):
        if world > 1:
            dist.barrier()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        out = batched_fit_gpu(y, exog, orders, args.train_len)
        torch.cuda.synchronize()
        if world > 1:
            dist.barrier()
        times.append(time.perf_counter() - t0)
    dt = min(times)
    if world > 1:
        t = torch.tensor([dt], dtype=torch.float64,
                         device="cuda" if dist.get_backend() == "nccl"
                         else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        dt = float(t.item())
    ok = float(out["status"].float().mean().item())
    if world > 1 and rank != 0:
        return

    result = {
        "metric": "groups/sec",
        "value": args.groups / dt,
        "unit": "groups/s",
        "n_gpus": world,
        "steps": args.repeats,
        "warmup": 1,
        "ms_per_step": dt * 1000.0,
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "fp32",
        "data": "synthetic",
        "config": {
            "model": f"ARIMAX grid {len(orders)} candidates + final fit",
            "groups": args.groups, "weeks": args.weeks,
            "train_len": args.train_len, "ok_fraction": ok,
        },
    }
    print(json.dumps(result))

    if args.cpu_oracle_groups > 0:
        g = args.cpu_oracle_groups
        t0 = time.perf_counter()
        batched_fit_reference(y[:g], exog, orders, args.train_len)
        dt_cpu = time.perf_counter() - t0
        print(json.dumps({"metric": "groups/sec", "value": g / dt_cpu,
                          "path": "numpy-oracle-1core", "groups": g}))

    if args.end_to_end:
        # whole W1 job: long DataFrame (arrow-backed keys, the parquet
        # shape) -> C++ gather -> GPU fit -> long forecast frame
        import pandas as pd
        from mi355x_scale.forecast.pipeline import (
            run_fine_grained_forecast_gpu)
        G, T = args.groups, args.weeks
        dates = _week_dates(T)
        df = pd.DataFrame({
            "Product": pd.array(
                np.repeat([f"P{i:04d}" for i in range(G)], T),
                dtype="string[pyarrow]"),
            "SKU": pd.array(
                np.repeat([f"SKU{i:06d}" for i in range(G)], T),
                dtype="string[pyarrow]"),
            "Date": np.tile(dates.to_numpy(), G),
            "Demand": y.reshape(-1).astype(np.float64),
        })
        t0 = time.perf_counter()
        res = run_fine_grained_forecast_gpu(df, orders,
                                            horizon=T - args.train_len)
        dt_e2e = time.perf_counter() - t0
        assert len(res) == G * T
        print(json.dumps({
            "metric": "groups/sec (end-to-end DataFrame->forecast frame)",
            "value": G / dt_e2e, "unit": "groups/s", "n_gpus": 1,
            "ms_per_step": dt_e2e * 1000.0, "higher_is_better": True,
            "config": {"groups": G, "weeks": T, "rows": G * T},
        }))


if __name__ == "__main__":
    main()
