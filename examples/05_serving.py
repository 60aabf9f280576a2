"""Serving walkthrough: stand up the inference/forecast API.

    python examples/05_serving.py          # self-test with TestClient
    uvicorn "examples.05_serving:app"      # real server

The classify endpoint runs the same fused normalize + bf16 path as
training; forecast runs the W1 pipeline (GPU batched kernel when
available, process-pool pandas otherwise).
"""
import sys

import numpy as np

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from mi355x_scale.serve import create_app  # noqa: E402

app = create_app(model_name="resnet18", num_classes=1000)


def main():
    from fastapi.testclient import TestClient
    with TestClient(app) as client:
        print(client.get("/health").json())
        img = np.random.default_rng(0).integers(
            0, 256, (64, 64, 3), dtype=np.uint8)
        out = client.post("/classify",
                          json={"image": img.tolist(), "top_k": 3}).json()
        print("classify top-3:", out)


if __name__ == "__main__":
    main()
