"""W1 walkthrough — the `group_apply/01_Introduction_And_Setup.py` +
`02_Fine_Grained_Demand_Forecasting.py` notebooks as a script.

Steps (mirroring the reference cells):
  1. generate the seed-123 demand panel (reference `_resources/00-setup` +
     `01-data-generator`),
  2. single-series modeling walkthrough: 4 ExponentialSmoothing variants
     + 2 SARIMAX variants (reference :69-245),
  3. notebook-level TPE tuning of one SKU (reference :250-324),
  4. per-SKU fan-out over the group engine (reference :340-556) — and,
     with a GPU, the batched-kernel path.
"""
import sys

import numpy as np

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from mi355x_scale import track
from mi355x_scale.data.generator import generate_demand_data
from mi355x_scale.forecast import (SARIMAX, ExponentialSmoothing,
                                   add_exo_variables,
                                   build_tune_and_score_model,
                                   evaluate_model, run_fine_grained_forecast,
                                   split_train_score_data, SEARCH_SPACE)
from mi355x_scale.tune import Trials, fmin, tpe


def main():
    track.set_experiment("fine_grained_forecasting")

    # 1. data (5 products x 10 SKUs x 157 weeks, seed 123)
    df = generate_demand_data()
    print(f"demand panel: {df['SKU'].nunique()} SKUs, {len(df)} rows")

    # 2. single-series walkthrough
    one = df[df["SKU"] == df["SKU"].iloc[0]].sort_values("Date")
    y = one["Demand"].to_numpy()
    variants = {
        "simple": ExponentialSmoothing(y).fit(),
        "trend": ExponentialSmoothing(y, trend="add").fit(),
        "damped": ExponentialSmoothing(y, trend="add",
                                       damped_trend=True).fit(),
        "seasonal": ExponentialSmoothing(y, trend="add", seasonal="add",
                                         seasonal_periods=52).fit(),
    }
    for name, fit in variants.items():
        print(f"  HW {name:9s} sse/n = {fit.sse / len(y):9.1f}")
    exo = one[["covid", "christmas", "new_year"]].to_numpy()
    for order in [(1, 0, 0), (1, 1, 1)]:
        res = SARIMAX(y, exog=exo, order=order).fit()
        print(f"  SARIMAX{order} mse = {res.mse:9.1f}")

    # 3. TPE tuning of that SKU (10 evals, seed 123 — ref :304-315)
    train, score = split_train_score_data(one)
    with track.start_run("single-sku-tpe"):
        trials = Trials()
        best = fmin(lambda p: evaluate_model(p, train, score),
                    SEARCH_SPACE, algo=tpe.suggest, max_evals=10,
                    trials=trials, rstate=np.random.default_rng(123))
    print(f"  best (p,d,q) = {best}")

    # 4. fan-out over every SKU
    out = run_fine_grained_forecast(df, max_evals=10)
    mse = float(np.mean((out["Demand"] - out["Demand_Fitted"]) ** 2))
    print(f"fan-out: {out['SKU'].nunique()} SKUs fitted, panel MSE {mse:.1f}")

    import torch
    if torch.cuda.is_available():
        from mi355x_scale.forecast import run_fine_grained_forecast_gpu
        out_gpu = run_fine_grained_forecast_gpu(df)
        mse_gpu = float(np.mean(
            (out_gpu["Demand"] - out_gpu["Demand_Fitted"]) ** 2))
        print(f"GPU batched path: panel MSE {mse_gpu:.1f}")

    # 5. restartable persisted output (the reference's Delta write,
    # group_apply/02_...py:544-552): per-group-shard Parquet + atomic
    # rename — a killed job resumes without refitting finished shards
    import tempfile
    from mi355x_scale.forecast import (read_forecast_shards,
                                       run_fine_grained_forecast_sharded)
    with tempfile.TemporaryDirectory() as d:
        run_fine_grained_forecast_sharded(df, d, num_shards=4,
                                          max_evals=4)
        persisted = read_forecast_shards(d)
        print(f"sharded output: {len(persisted)} rows across 4 shards")
    return out


if __name__ == "__main__":
    main()
