"""BASELINE config 1: single store-SKU demand forecast on CPU via the
pandas path (plumbing check, no GPU). Metric: wall-clock seconds.

    python benchmarks/bench_single_sku.py
"""
import json
import sys
import time

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from mi355x_scale.data.generator import generate_demand_data  # noqa: E402
from mi355x_scale.forecast import build_tune_and_score_model  # noqa: E402


def main():
    df = generate_demand_data(n_products=1, skus_per_product=1, n_weeks=157)
    t0 = time.perf_counter()
    out = build_tune_and_score_model(df, max_evals=10, seed=123)
    dt = time.perf_counter() - t0
    import numpy as np
    d = out["Demand"].to_numpy()
    f = out["Demand_Fitted"].to_numpy()
    mse = float(np.mean((d[10:] - f[10:]) ** 2))
    print(json.dumps({
        "metric": "wall-clock seconds (single SKU, 10 TPE evals + final fit)",
        "value": dt,
        "unit": "s",
        "n_gpus": 0,
        "steps": 1,
        "warmup": 0,
        "ms_per_step": dt * 1000.0,
        "higher_is_better": False,
        "scaling": "strong",
        "vs_baseline": None,
        "dtype": "fp64",
        "data": "synthetic (seed 123)",
        "config": {"model": "SARIMAX-lite (p,d,q) TPE", "weeks": 157,
                   "fitted_mse": mse},
    }))


if __name__ == "__main__":
    main()
