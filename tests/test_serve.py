"""Serving layer: health / classify / forecast endpoints (CPU)."""
import numpy as np
import pandas as pd
import pytest

fastapi = pytest.importorskip("fastapi")
from fastapi.testclient import TestClient  # noqa: E402

from mi355x_scale.serve import create_app  # noqa: E402


@pytest.fixture(scope="module")
def client():
    app = create_app(model_name="resnet18", num_classes=10, device="cpu")
    with TestClient(app) as c:
        yield c


def test_health(client):
    r = client.get("/health")
    assert r.status_code == 200 and r.json()["status"] == "ok"


def test_classify(client):
    rng = np.random.default_rng(0)
    img = rng.integers(0, 256, (64, 64, 3), dtype=np.uint8)
    r = client.post("/classify", json={"image": img.tolist(), "top_k": 3})
    assert r.status_code == 200
    body = r.json()
    assert len(body["classes"]) == 3
    assert abs(sum(body["probs"])) <= 1.0 + 1e-3


def test_classify_rejects_bad_shape(client):
    r = client.post("/classify", json={"image": [[1, 2], [3, 4]]})
    assert r.status_code == 400


def test_forecast(client):
    rng = np.random.default_rng(1)
    dates = pd.date_range("2021-01-04", periods=60, freq="W-MON")
    records = []
    for sku in ("A", "B"):
        demand = 100 + rng.normal(0, 5, 60).cumsum()
        for d, v in zip(dates, demand):
            records.append({"Product": "P1", "SKU": sku,
                            "Date": str(d.date()), "Demand": float(v)})
    r = client.post("/forecast", json={"records": records, "horizon": 10})
    assert r.status_code == 200
    rows = r.json()["rows"]
    assert len(rows) == 120
    assert {"Product", "SKU", "Date", "Demand", "Demand_Fitted"} <= \
        set(rows[0])


@pytest.mark.gpu
def test_classify_on_gpu_fused_path():
    import torch
    assert torch.cuda.is_available()
    app = create_app(model_name="resnet18", num_classes=10, device="cuda:0")
    with TestClient(app) as c:
        img = np.random.default_rng(2).integers(
            0, 256, (224, 224, 3), dtype=np.uint8)
        r = c.post("/classify", json={"image": img.tolist(), "top_k": 4})
        assert r.status_code == 200
        assert len(r.json()["classes"]) == 4
