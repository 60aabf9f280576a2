"""BASELINE config 2: Hyperopt-style TPE, 64 trials, sklearn GBT on
synthetic tabular data, one trial per GPU worker. Metric: trials/min.

    python benchmarks/bench_tpe.py --max-evals 64 --parallelism 8
"""
import argparse
import json
import sys
import time

import numpy as np

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from mi355x_scale.tune import GPUTrials, Trials, fmin, hp, scope, tpe  # noqa: E402


def objective(params):
    from sklearn.datasets import make_classification
    from sklearn.ensemble import GradientBoostingClassifier
    from sklearn.model_selection import cross_val_score
    X, y = make_classification(n_samples=2000, n_features=20,
                               n_informative=10, random_state=0)
    clf = GradientBoostingClassifier(
        n_estimators=int(params["n_estimators"]),
        max_depth=int(params["max_depth"]),
        learning_rate=params["learning_rate"],
        subsample=params["subsample"],
        random_state=0,
    )
    return -cross_val_score(clf, X, y, cv=3).mean()


SPACE = {
    "n_estimators": scope.int(hp.quniform("n_estimators", 20, 150, 10)),
    "max_depth": scope.int(hp.quniform("max_depth", 2, 8, 1)),
    "learning_rate": hp.lognormal("learning_rate", np.log(0.1), 0.5),
    "subsample": hp.uniform("subsample", 0.5, 1.0),
}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--max-evals", type=int, default=64)
    ap.add_argument("--parallelism", type=int, default=8)
    args = ap.parse_args()

    trials = GPUTrials(parallelism=args.parallelism)
    t0 = time.perf_counter()
    best = fmin(objective, SPACE, algo=tpe.suggest,
                max_evals=args.max_evals, trials=trials,
                rstate=np.random.default_rng(123))
    dt = time.perf_counter() - t0
    losses = [l for l in trials.losses() if l is not None]
    print(json.dumps({
        "metric": "trials/min",
        "value": args.max_evals / dt * 60.0,
        "unit": "trials/min",
        "n_gpus": trials.parallelism,
        "steps": args.max_evals,
        "warmup": 0,
        "ms_per_step": dt / args.max_evals * 1000.0,
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "fp64",
        "data": "synthetic",
        "config": {"model": "sklearn GBT", "parallelism": args.parallelism,
                   "best_loss": min(losses), "best": {k: float(v) for k, v
                                                      in best.items()}},
    }))


if __name__ == "__main__":
    main()
