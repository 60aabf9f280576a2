"""Batched group-fit: numpy oracle sanity (CPU) + HIP-kernel-vs-oracle
numerics (GPU) — SURVEY §4's kernel-vs-CPU-reference rule."""
import numpy as np
import pytest

from mi355x_scale.data.generator import demand_factors, generate_demand_data
from mi355x_scale.forecast.batched import (batched_fit_reference,
                                           fitted_values_reference,
                                           make_exog_designs)
from mi355x_scale.forecast.pipeline import EXO_COLS

ORDERS = [(0, 1, 0), (1, 0, 0), (1, 1, 1), (2, 1, 0), (0, 1, 1), (2, 0, 2)]


def _dataset(n_products=2, skus=4, weeks=157):
    df = generate_demand_data(n_products, skus, weeks)
    piv = df.pivot_table(index=["Product", "SKU"], columns="Date",
                         values="Demand")
    y = piv.to_numpy()
    exog = df[df["SKU"] == df["SKU"].iloc[0]][EXO_COLS].to_numpy()
    return y, exog


def test_reference_selects_reasonable_models():
    y, exog = _dataset()
    S = 117
    mse = batched_fit_reference(y, exog, ORDERS, S)
    assert mse.shape == (8, len(ORDERS))
    assert np.isfinite(mse).all()
    # chosen model must beat the worst candidate clearly on average
    best = mse.min(axis=1)
    worst = mse.max(axis=1)
    assert (best < worst).all()
    # and beat a naive flat forecast (predict train tail mean)
    naive = np.mean((y[:, S:] - y[:, S - 1:S]) ** 2, axis=1)
    assert (best < naive).mean() >= 0.75


def test_reference_agrees_with_sarimax_lite():
    """Same algorithm family: per-group validation MSEs should be within
    a small factor of the sarimax.py path for shared orders."""
    from mi355x_scale.forecast import SARIMAX
    y, exog = _dataset(1, 2)
    S = 117
    H = y.shape[1] - S
    mse = batched_fit_reference(y, exog, [(1, 1, 1)], S)
    for g in range(y.shape[0]):
        res = SARIMAX(y[g, :S], exog=exog[:S], order=(1, 1, 1)).fit()
        fc = res.forecast(H, exog=exog[S:])
        ref_mse = np.mean((y[g, S:] - fc) ** 2)
        assert mse[g, 0] < 4.0 * ref_mse + 1e-6
        assert ref_mse < 4.0 * mse[g, 0] + 1e-6


def test_fitted_values_reference_tracks_series():
    y, exog = _dataset(1, 1)
    fit = fitted_values_reference(y[0], exog, (1, 1, 1), len(y[0]))
    resid = (y[0] - fit)[8:]
    assert np.mean(resid ** 2) < np.var(y[0][8:])


@pytest.mark.gpu
def test_gpu_eval_matches_numpy_oracle():
    import torch
    from mi355x_scale.forecast.batched import batched_eval_gpu
    y, exog = _dataset(2, 8)
    S = 117
    ref = batched_fit_reference(y, exog, ORDERS, S)
    mse, status = batched_eval_gpu(y, exog, ORDERS, S)
    mse = mse.cpu().numpy()
    assert status.cpu().numpy().all()
    # f32 kernel vs f64 oracle: relative tolerance on the MSE surface
    rel = np.abs(mse - ref) / (np.abs(ref) + 1e-3)
    assert rel.max() < 5e-2, f"max rel err {rel.max()}"
    # candidate RANKING must agree for the argmin in nearly all groups
    agree = (mse.argmin(1) == ref.argmin(1)).mean()
    assert agree >= 0.8


@pytest.mark.gpu
def test_gpu_final_fit_matches_numpy_oracle():
    import torch
    from mi355x_scale.forecast.batched import batched_fit_gpu
    y, exog = _dataset(2, 4)
    S = 117
    out = batched_fit_gpu(y, exog, ORDERS, S)
    assert out["status"].cpu().numpy().all()
    fitted = out["fitted"].cpu().numpy()
    bo = out["best_order"].cpu().numpy()
    for g in range(y.shape[0]):
        ref_fit = fitted_values_reference(y[g], exog, tuple(bo[g]),
                                          len(y[g]))
        err = np.abs(fitted[g, 8:] - ref_fit[8:])
        scale = np.abs(y[g]).mean()
        assert err.max() < 2e-2 * scale, f"group {g}: {err.max()}"


@pytest.mark.gpu
def test_gpu_forecast_pipeline_end_to_end():
    """W1 GPU path end-to-end: DataFrame in, forecast frame out, and the
    fitted series beats the naive mean (statistical parity target)."""
    from mi355x_scale.forecast import run_fine_grained_forecast_gpu
    df = generate_demand_data(n_products=2, skus_per_product=5,
                              n_weeks=157)
    out = run_fine_grained_forecast_gpu(df)
    assert list(out.columns) == ["Product", "SKU", "Date", "Demand",
                                 "Demand_Fitted"]
    assert out["SKU"].nunique() == 10
    d = out["Demand"].to_numpy()
    f = out["Demand_Fitted"].to_numpy()
    assert np.isfinite(f).all()
    mse = np.mean((d[10:] - f[10:]) ** 2)
    assert mse < 0.6 * np.var(d[10:])


@pytest.mark.gpu
def test_mfma_projection_matches_matmul():
    """The design-matrix GEMM on f32 MFMA vs torch matmul (exact-f32
    MFMA: bitwise-class agreement expected, tolerance covers ordering)."""
    import torch
    from mi355x_scale.ops import _C
    rng = np.random.default_rng(0)
    n, G, KX = 117, 1000, 3
    P = torch.tensor(rng.standard_normal((KX, n)), dtype=torch.float32,
                     device="cuda").contiguous()
    wc = torch.tensor(rng.standard_normal((n, G)), dtype=torch.float32,
                      device="cuda").contiguous()
    beta = torch.empty((KX, G), dtype=torch.float32, device="cuda")
    _C.exog_project_mfma(P, wc, beta)
    ref = P @ wc
    assert torch.allclose(beta, ref, atol=1e-4, rtol=1e-5), \
        (beta - ref).abs().max().item()


@pytest.mark.gpu
def test_mfma_path_equals_lane_path():
    """Eval MSEs with MFMA-precomputed stage 1 vs the in-kernel matvec."""
    from mi355x_scale.forecast.batched import batched_eval_gpu
    y, exog = _dataset(2, 8)
    m1, s1 = batched_eval_gpu(y, exog, ORDERS, 117, use_mfma=True)
    m2, s2 = batched_eval_gpu(y, exog, ORDERS, 117, use_mfma=False)
    import torch
    rel = ((m1 - m2).abs() / (m2.abs() + 1e-3)).max().item()
    assert rel < 1e-3, rel


@pytest.mark.gpu
def test_gpu_pipeline_dataframe_to_forecast_frame():
    """Whole W1 job on GPU: long DataFrame (arrow-backed keys) ->
    C++ group gather -> batched CDNA4 fit -> long forecast frame
    (run_fine_grained_forecast_gpu; loads groupby._gather + ops._C)."""
    import pandas as pd

    from mi355x_scale.forecast.pipeline import run_fine_grained_forecast_gpu
    from mi355x_scale.groupby.gather import HAVE_GATHER_EXT

    assert HAVE_GATHER_EXT, "C++ gather engine must be built on GPU boxes"
    rng = np.random.default_rng(123)
    G, T = 500, 157
    dates = pd.date_range("2020-06-29", periods=T, freq="W-MON")
    base = rng.uniform(400, 1200, size=(G, 1))
    y = base + rng.normal(0, base * 0.05, size=(G, T))
    df = pd.DataFrame({
        "Product": pd.array(np.repeat([f"P{i % 7}" for i in range(G)], T),
                            dtype="string[pyarrow]"),
        "SKU": pd.array(np.repeat([f"S{i:05d}" for i in range(G)], T),
                        dtype="string[pyarrow]"),
        "Date": np.tile(dates.to_numpy(), G),
        "Demand": np.clip(y, 0, None).reshape(-1),
    })
    res = run_fine_grained_forecast_gpu(df)
    assert len(res) == G * T
    assert list(res.columns) == ["Product", "SKU", "Date", "Demand",
                                 "Demand_Fitted"]
    assert res["Demand_Fitted"].notna().mean() > 0.95
    # fitted values must track the actual demand scale per group
    one = res[res["SKU"] == "S00007"]
    ratio = one["Demand_Fitted"].mean() / max(one["Demand"].mean(), 1.0)
    assert 0.5 < ratio < 1.5, ratio
