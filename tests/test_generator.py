import pytest
import numpy as np

from mi355x_scale.data.generator import (arma_generate_sample,
                                         generate_bom,
                                         generate_demand_data)


def test_demand_data_shape_and_determinism():
    df = generate_demand_data(n_products=2, skus_per_product=3, n_weeks=52)
    assert len(df) == 2 * 3 * 52
    assert list(df.columns) == ["Product", "SKU", "Date", "Demand",
                                "covid", "christmas", "new_year"]
    assert df["Demand"].min() >= 0
    df2 = generate_demand_data(n_products=2, skus_per_product=3, n_weeks=52)
    assert (df["Demand"].to_numpy() == df2["Demand"].to_numpy()).all()


def test_demand_reference_scale():
    """The reference's default config: 5 products × 10 SKUs × 157 weeks
    (01-data-generator.py:57,135-145)."""
    df = generate_demand_data()
    assert df["SKU"].nunique() == 50
    assert len(df) == 50 * 157


def test_arma_sample_stationary():
    rng = np.random.default_rng(123)
    x = arma_generate_sample([1.0, -0.5], [1.0, 0.2], 500, rng=rng)
    assert len(x) == 500
    assert abs(np.mean(x)) < 1.0  # zero-mean stationary process
    # lag-1 autocorrelation should be near theoretical for AR(1)+MA(1)
    r1 = np.corrcoef(x[:-1], x[1:])[0, 1]
    assert 0.2 < r1 < 0.9


def test_bom_dag():
    bom, mapper = generate_bom(["SKU1", "SKU2"], levels=2)
    assert len(mapper) == 2
    assert set(bom.columns) == {"material_in", "material_out", "qty"}
    assert (bom["qty"] >= 1).all()


@pytest.mark.gpu
def test_arma_batch_gpu_matches_lfilter():
    """Batched GPU ARMA generator vs scipy lfilter, per group (the N7
    kernel mirrors statsmodels' lfilter(ma, ar, eps) convention)."""
    import torch
    from scipy.signal import lfilter

    from mi355x_scale.data.generator import arma_generate_sample_batch_gpu

    rng = np.random.default_rng(7)
    G, T, burn = 64, 157, 100
    orders = [(rng.integers(0, 5), rng.integers(0, 5)) for _ in range(G)]
    na = 5
    ar = np.zeros((G, na), dtype=np.float64)
    ma = np.zeros((G, na), dtype=np.float64)
    ar[:, 0] = 1.0
    ma[:, 0] = 1.0
    for g, (p, q) in enumerate(orders):
        if p:
            ar[g, 1:1 + p] = rng.uniform(-0.4, 0.4, p) / max(p, 1)
        if q:
            ma[g, 1:1 + q] = rng.uniform(-0.5, 0.5, q)

    # shared noise so GPU and oracle see identical eps
    noise_rng = np.random.default_rng(99)
    out = arma_generate_sample_batch_gpu(ar, ma, T, scale=2.5, burnin=burn,
                                         rng=np.random.default_rng(99))
    eps = 2.5 * noise_rng.standard_normal((T + burn, G)).astype(np.float32)
    got = out.cpu().numpy()
    for g in range(0, G, 7):
        ref = lfilter(ma[g], ar[g], eps[:, g].astype(np.float64))[burn:]
        np.testing.assert_allclose(got[:, g], ref, rtol=2e-3, atol=2e-3)
