"""W1 tests: SARIMAX-lite numerics, groupby engine semantics, and the
end-to-end per-SKU pipeline on seed-123 data."""
import numpy as np
import pandas as pd
import pytest

from mi355x_scale.data.generator import generate_demand_data
from mi355x_scale.forecast import (SARIMAX, add_exo_variables,
                                   build_tune_and_score_model,
                                   run_fine_grained_forecast,
                                   split_train_score_data)
from mi355x_scale.groupby import LocalFrame, apply_in_pandas


def _simulate_arma(T, phi, theta, rng, burn=200):
    e = rng.standard_normal(T + burn)
    u = np.zeros(T + burn)
    for t in range(1, T + burn):
        u[t] = phi * u[t - 1] + e[t] + theta * e[t - 1]
    return u[burn:], e[burn:]


def test_sarimax_parameter_recovery():
    rng = np.random.default_rng(123)
    T = 400
    u, e = _simulate_arma(T, 0.6, 0.3, rng)
    x = rng.standard_normal((T, 2))
    y = 10 + x @ np.array([2.0, -1.0]) + u
    res = SARIMAX(y, exog=x, order=(1, 0, 1)).fit()
    assert abs(res.const - 10) < 0.5
    assert np.allclose(res.beta, [2.0, -1.0], atol=0.3)
    assert abs(res.phi[0] - 0.6) < 0.15
    assert abs(res.theta[0] - 0.3) < 0.2
    # one-step-ahead residual variance ≈ innovation variance
    assert res.mse < 1.5 * e.var()


def test_sarimax_d1_forecast_tracks_trend():
    rng = np.random.default_rng(7)
    t = np.arange(200.0)
    y = 5.0 * t + rng.standard_normal(200)
    res = SARIMAX(y, order=(1, 1, 0)).fit()
    fc = res.forecast(10)
    expect = 5.0 * (t[-1] + np.arange(1, 11))
    assert np.abs(fc - expect).max() < 10.0


def test_sarimax_predict_range_matches_reference_call():
    """predict(0, T+h-1, exog=...) — the reference's full-range call
    (group_apply/02_...py:484-488)."""
    rng = np.random.default_rng(0)
    y = rng.standard_normal(100).cumsum() + 50
    x = rng.standard_normal((100, 1))
    res = SARIMAX(y, exog=x, order=(1, 1, 0)).fit()
    pred = res.predict(0, 109, exog=np.zeros((10, 1)))
    assert len(pred) == 110
    assert np.isfinite(pred).all()


def test_sarimax_pure_ar_zero_order():
    y = np.arange(50.0) + 1
    res = SARIMAX(y, order=(0, 1, 0)).fit()
    fc = res.forecast(5)
    assert np.allclose(fc, y[-1] + 1 + np.arange(5), atol=1e-6)


def test_groupby_matches_pandas_reference():
    """Bit-match vs a plain pandas groupby-apply."""
    df = pd.DataFrame({
        "k": ["a"] * 5 + ["b"] * 7 + ["c"] * 3,
        "v": np.arange(15.0),
    })

    def fn(g):
        return pd.DataFrame({"k": [g["k"].iloc[0]],
                             "mean_v": [g["v"].mean()]})

    out = apply_in_pandas(df, ["k"], fn, "k string, mean_v double",
                          num_workers=1)
    ref = df.groupby("k")["v"].mean()
    got = out.set_index("k")["mean_v"]
    for k in ref.index:
        assert got[k] == ref[k]


def test_groupby_parallel_equals_serial():
    df = pd.DataFrame({"k": np.repeat(np.arange(12), 4),
                       "v": np.arange(48.0)})

    def fn(g):
        return pd.DataFrame({"k": [int(g["k"].iloc[0])],
                             "s": [g["v"].sum()]})

    a = apply_in_pandas(df, ["k"], fn, "k long, s double", num_workers=1)
    b = apply_in_pandas(df, ["k"], fn, "k long, s double", num_workers=3)
    a = a.sort_values("k").reset_index(drop=True)
    b = b.sort_values("k").reset_index(drop=True)
    pd.testing.assert_frame_equal(a, b)


def test_groupby_failure_isolation():
    """One crashing group must not kill the job."""
    df = pd.DataFrame({"k": ["a"] * 3 + ["bad"] * 3 + ["c"] * 3,
                       "v": np.arange(9.0)})

    def fn(g):
        if g["k"].iloc[0] == "bad":
            raise RuntimeError("group exploded")
        return pd.DataFrame({"k": [g["k"].iloc[0]], "n": [len(g)]})

    gf = LocalFrame(df, num_workers=1).groupBy("k")
    out = gf.applyInPandas(fn, "k string, n long")
    assert set(out["k"]) == {"a", "c"}
    assert len(gf.failures) == 1
    assert gf.failures[0][0] == ("bad",)
    assert "group exploded" in gf.failures[0][1]


def test_build_tune_and_score_single_sku():
    """Config 1 of BASELINE.json: single store-SKU forecast, CPU pandas
    path — fitted series must beat the naive mean on seed-123 data."""
    df = generate_demand_data(n_products=1, skus_per_product=1, n_weeks=157)
    out = build_tune_and_score_model(df, max_evals=6)
    assert list(out.columns) == ["Product", "SKU", "Date", "Demand",
                                 "Demand_Fitted"]
    assert len(out) == 157
    d, f = out["Demand"].to_numpy(), out["Demand_Fitted"].to_numpy()
    tail = slice(10, None)  # skip startup transient
    mse_model = np.mean((d[tail] - f[tail]) ** 2)
    mse_naive = np.var(d[tail])
    assert mse_model < 0.6 * mse_naive


def test_run_fine_grained_forecast_small():
    """W1 end-to-end: 4 SKUs through the group engine."""
    df = generate_demand_data(n_products=2, skus_per_product=2, n_weeks=120)
    out = run_fine_grained_forecast(df, num_workers=2, max_evals=2,
                                    horizon=20)
    assert out["SKU"].nunique() == 4
    assert len(out) == 4 * 120
    assert np.isfinite(out["Demand_Fitted"]).all()


def test_holt_winters_variants():
    """The four walkthrough variants (ref group_apply/02_...py:143-188):
    simple, trend, damped trend, trend+seasonal."""
    from mi355x_scale.forecast import ExponentialSmoothing
    rng = np.random.default_rng(3)
    t = np.arange(208.0)
    season = 10 * np.sin(2 * np.pi * t / 52)
    y = 100 + 0.5 * t + season + rng.standard_normal(208)

    simple = ExponentialSmoothing(y).fit()
    trend = ExponentialSmoothing(y, trend="add").fit()
    damped = ExponentialSmoothing(y, trend="add", damped_trend=True).fit()
    hw = ExponentialSmoothing(y, trend="add", seasonal="add",
                              seasonal_periods=52).fit()
    # one-step fit: seasonal+trend at least matches simple smoothing
    assert hw.sse < simple.sse
    assert np.isfinite(trend.sse) and np.isfinite(damped.sse)
    # the long-horizon forecast is where seasonality pays: the 52-step
    # dynamic forecast must track trend+sine closely (simple smoothing
    # would be off by up to ~23 = trend drift + amplitude)
    fc = hw.forecast(52)
    assert len(fc) == 52
    expect = 100 + 0.5 * (t[-1] + np.arange(1, 53)) + \
        10 * np.sin(2 * np.pi * (t[-1] + np.arange(1, 53)) / 52)
    assert np.abs(fc - expect).mean() < 3.0


def test_holt_winters_multiplicative():
    from mi355x_scale.forecast import ExponentialSmoothing
    t = np.arange(156.0)
    y = (50 + t) * (1 + 0.2 * np.sin(2 * np.pi * t / 52))
    fit = ExponentialSmoothing(y, trend="add", seasonal="mul",
                               seasonal_periods=52).fit()
    assert fit.sse / len(y) < np.var(y) * 0.1


def test_panel_gather_roundtrip():
    """Vectorized scatter-gather matches the pandas pivot reference."""
    from mi355x_scale.groupby.gather import panel_from_long, long_from_panel
    df = generate_demand_data(n_products=2, skus_per_product=3, n_weeks=30)
    df_shuffled = df.sample(frac=1.0, random_state=0)  # any row order
    panel, gindex, tvals = panel_from_long(
        df_shuffled, ["Product", "SKU"], "Date", "Demand")
    piv = df.pivot_table(index=["Product", "SKU"], columns="Date",
                         values="Demand", sort=True)
    assert panel.shape == piv.shape
    assert np.allclose(panel, piv.to_numpy(), atol=1e-4)
    back = long_from_panel(panel, gindex, tvals, ["Product", "SKU"],
                           "Date", [("Demand", panel)])
    assert len(back) == panel.size
    merged = back.merge(df, on=["Product", "SKU", "Date"],
                        suffixes=("_got", "_want"))
    assert np.allclose(merged["Demand_got"], merged["Demand_want"],
                       atol=1e-4)
