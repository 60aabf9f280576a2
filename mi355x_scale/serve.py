"""Minimal serving layer for the two model families (deployment
counterpart of the training stack; the reference itself never serves —
this rounds out the "production deployment and serving" story).

    from mi355x_scale.serve import create_app
    app = create_app(classifier=my_image_classifier)
    # uvicorn mi355x_scale.serve:app ... or TestClient(app)

Endpoints:
  GET  /health                 -> {"status": "ok", "device": ...}
  POST /classify               -> top-k classes for a uint8 HWC image
                                  (json list) — runs the same fused
                                  normalize + bf16 path as training
  POST /forecast               -> per-group demand forecast: long-format
                                  records in, fitted values out (the W1
                                  GPU pipeline under the hood)
"""
from typing import Optional

import numpy as np
import torch

try:
    from pydantic import BaseModel
except ImportError:  # pragma: no cover - pydantic ships with fastapi
    BaseModel = object


class ClassifyRequest(BaseModel):
    image: list  # HWC uint8 nested list
    top_k: int = 5


class ForecastRequest(BaseModel):
    records: list  # [{Product, SKU, Date, Demand}]
    horizon: int = 40


def create_app(classifier=None, model_name: str = "resnet18",
               num_classes: int = 1000, device: Optional[str] = None):
    from fastapi import FastAPI, HTTPException

    dev = torch.device(device or
                       ("cuda:0" if torch.cuda.is_available() else "cpu"))
    if classifier is None:
        from .train import ImageClassifier
        classifier = ImageClassifier(model_name, num_classes=num_classes)
    classifier = classifier.to(dev).eval()
    if dev.type == "cuda":
        classifier.to(memory_format=torch.channels_last)

    app = FastAPI(title="mi355x_scale serving")

    @app.get("/health")
    def health():
        return {"status": "ok", "device": str(dev)}

    @app.post("/classify")
    def classify(req: ClassifyRequest):
        arr = np.asarray(req.image, dtype=np.uint8)
        if arr.ndim != 3 or arr.shape[-1] != 3:
            raise HTTPException(400, "image must be HWC uint8 with C=3")
        x = torch.from_numpy(arr).unsqueeze(0).to(dev)
        with torch.no_grad(), torch.autocast(
                device_type="cuda", dtype=torch.bfloat16,
                enabled=dev.type == "cuda"):
            from .ops import normalize_images
            xin = normalize_images(x)
            if dev.type == "cpu":
                xin = xin.float()
            logits = classifier(xin).float().softmax(-1)[0]
        k = min(req.top_k, logits.numel())
        probs, idx = torch.topk(logits, k)
        return {"classes": idx.cpu().tolist(),
                "probs": [round(float(p), 6) for p in probs.cpu()]}

    @app.post("/forecast")
    def forecast(req: ForecastRequest):
        import pandas as pd
        df = pd.DataFrame(req.records)
        need = {"Product", "SKU", "Date", "Demand"}
        if not need <= set(df.columns):
            raise HTTPException(400, f"records need columns {sorted(need)}")
        if dev.type == "cuda":
            from .forecast.pipeline import run_fine_grained_forecast_gpu
            out = run_fine_grained_forecast_gpu(df, horizon=req.horizon)
        else:
            from .forecast.pipeline import run_fine_grained_forecast
            out = run_fine_grained_forecast(df, horizon=req.horizon,
                                            max_evals=5)
        out = out.copy()
        out["Date"] = out["Date"].astype(str)
        for c in ("Product", "SKU"):
            out[c] = out[c].astype(str)
        return {"rows": out.to_dict(orient="records")}

    return app
