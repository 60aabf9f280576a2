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
