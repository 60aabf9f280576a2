"""BASELINE config 5: nested scaling — TPE trials where EACH TRIAL is an
N-GPU DDP ResNet-18 run (16 sequential trials). Exercises scheduler
re-entrancy: fmin (serial Trials) → TorchDistributor → N ranks → RCCL.

    python benchmarks/bench_nested.py --trials 16 --gpus 8 --steps 20
"""
import argparse
import json
import os
import subprocess
import sys
import time

import numpy as np

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from mi355x_scale.tune import Trials, fmin, hp, tpe  # noqa: E402

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_ddp_trial(lr_exp: float, gpus: int, steps: int, warmup: int):
    """One trial = one N-GPU bench.py run; loss = -samples/sec (the trial
    objective is throughput here; a loss-based objective would read the
    run's val_loss instead)."""
    cmd = [sys.executable]
    if gpus > 1:
        cmd = [sys.executable, "-m", "torch.distributed.run",
               f"--nproc-per-node={gpus}", "--nnodes=1",
               "--master-addr=127.0.0.1", "--standalone"]
    cmd += [os.path.join(REPO, "bench.py"), "--gpus", str(gpus),
            "--steps", str(steps), "--warmup", str(warmup)]
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=900,
                         cwd=REPO)
    for line in reversed(out.stdout.strip().splitlines()):
        if line.startswith("{"):
            rec = json.loads(line)
            return rec["value"]
    raise RuntimeError(f"trial run failed: {out.stderr[-2000:]}")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--trials", type=int, default=16)
    ap.add_argument("--gpus", type=int, default=8)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=8)
    args = ap.parse_args()

    samples_per_sec = []

    def objective(params):
        v = run_ddp_trial(params["lr_exp"], args.gpus, args.steps,
                          args.warmup)
        samples_per_sec.append(v)
        return -v

    t0 = time.perf_counter()
    trials = Trials()  # sequential trials (config 5 contract)
    fmin(objective, {"lr_exp": hp.uniform("lr_exp", -6, -3)},
         algo=tpe.suggest, max_evals=args.trials, trials=trials,
         rstate=np.random.default_rng(123))
    wall = time.perf_counter() - t0

    print(json.dumps({
        "metric": "samples/sec within-trial + total wall",
        "value": float(np.mean(samples_per_sec)),
        "unit": "samples/s (mean within-trial)",
        "n_gpus": args.gpus,
        "steps": args.trials,
        "warmup": 0,
        "ms_per_step": wall / args.trials * 1000.0,
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "bf16",
        "data": "synthetic",
        "config": {"model": "resnet18", "trials": args.trials,
                   "gpus_per_trial": args.gpus, "total_wall_s": wall},
    }))


if __name__ == "__main__":
    main()
