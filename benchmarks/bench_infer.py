"""Serving-path benchmark: hipGraph-captured bf16 inference throughput
and latency for the flagship classifier (the deployment counterpart of
bench.py's training step).

    python benchmarks/bench_infer.py --batch-size 212 --steps 200
    python benchmarks/bench_infer.py --batch-size 1 --steps 500   # latency
"""
import argparse
import json
import sys
import time

import numpy as np
import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from mi355x_scale.train import ImageClassifier  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch-size", type=int, default=212)
    ap.add_argument("--steps", type=int, default=200)
    ap.add_argument("--warmup", type=int, default=30)
    ap.add_argument("--model", type=str, default="resnet18")
    args = ap.parse_args()
    assert torch.cuda.is_available(), "inference bench needs the GPU"
    dev = torch.device("cuda:0")
    torch.backends.cudnn.benchmark = True

    model = ImageClassifier(args.model, num_classes=1000).to(dev)
    model.to(memory_format=torch.channels_last)
    model.eval()

    x = torch.randint(0, 256, (args.batch_size, 224, 224, 3),
                      dtype=torch.uint8, device=dev)

    def fwd():
        with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
            from mi355x_scale.ops import normalize_images
            return model(normalize_images(x))

    with torch.no_grad():
        for _ in range(args.warmup):
            out = fwd()
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g, capture_error_mode="thread_local"):
            static_out = fwd()

        lat = []
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.steps):
            t1 = time.perf_counter()
            g.replay()
            torch.cuda.synchronize()
            lat.append(time.perf_counter() - t1)
        elapsed = time.perf_counter() - t0

    lat_ms = np.array(lat) * 1e3
    print(json.dumps({
        "metric": "inference samples/sec",
        "value": args.batch_size * args.steps / elapsed,
        "unit": "samples/s",
        "n_gpus": 1,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": elapsed / args.steps * 1000.0,
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "bf16",
        "data": "synthetic",
        "config": {"model": args.model, "global_batch": args.batch_size,
                   "mode": "eval+hipgraph",
                   "latency_ms": {"p50": float(np.median(lat_ms)),
                                  "p99": float(np.percentile(lat_ms, 99))}},
    }))
    assert torch.isfinite(static_out.float()).all()


if __name__ == "__main__":
    main()
