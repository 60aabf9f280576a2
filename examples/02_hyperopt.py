"""W2 walkthrough — `hyperopt/1. hyperopt.py` as a script: SVC on iris,
single-machine fmin then distributed GPUTrials (SparkTrials equivalent),
runs logged in the mlruns layout."""
import sys

import numpy as np

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from mi355x_scale import track
from mi355x_scale.tune import SparkTrials, Trials, fmin, hp, tpe


def objective(C):
    # ref hyperopt/1...py:54-62
    from sklearn import datasets
    from sklearn.model_selection import cross_val_score
    from sklearn.svm import SVC
    iris = datasets.load_iris()
    clf = SVC(C=C)
    return -cross_val_score(clf, iris.data, iris.target).mean()


def main():
    track.set_experiment("hyperopt_svc")
    space = hp.lognormal("C", 0, 1.0)           # ref :72

    # single-machine (ref :94-98)
    with track.start_run("serial"):
        best = fmin(objective, space, algo=tpe.suggest, max_evals=20,
                    rstate=np.random.default_rng(123))
    print("serial best C:", best["C"])

    # distributed, one trial per worker (ref :128-136)
    with track.start_run("distributed"):
        trials = SparkTrials(parallelism=4, use_gpu=False)
        best = fmin(objective, space, algo=tpe.suggest, max_evals=20,
                    trials=trials, rstate=np.random.default_rng(123))
    losses = [l for l in trials.losses() if l is not None]
    print(f"distributed best C: {best['C']}, best acc: {-min(losses):.4f}")


if __name__ == "__main__":
    main()
