"""W2 data-shipping strategies — `hyperopt/2. hyperopt on diff sizes of
data.py` as a script: LASSO-alpha tuning with the three strategies the
reference demonstrates for small/medium/large datasets:

  1. closure capture (small data rides in the pickled objective),
  2. shared-memory broadcast (`parallel.broadcast` — the sc.broadcast
     equivalent, ref :92-99),
  3. save-to-shared-filesystem + reload in the worker (ref :114-152).
"""
import os
import sys
import tempfile

import numpy as np

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from mi355x_scale.parallel import broadcast
from mi355x_scale.tune import SparkTrials, fmin, hp, tpe


def gen_data(mb: float):
    # ref :25-33 — regression data sized by bytes
    from sklearn.datasets import make_regression
    n = max(100, int(mb * 1e6 / (8 * 101)))
    X, y = make_regression(n_samples=n, n_features=100, noise=0.5,
                           random_state=0)
    return X, y


def train_and_eval(alpha, X, y):
    # ref :35-43
    from sklearn.linear_model import Lasso
    from sklearn.model_selection import cross_val_score
    return -cross_val_score(Lasso(alpha=alpha), X, y, cv=3).mean()


def tune_alpha(objective, label):
    # ref :45-56 — 4 evals @ parallelism 2
    trials = SparkTrials(parallelism=2, use_gpu=False)
    best = fmin(objective, hp.uniform("alpha", 0.0, 10.0),
                algo=tpe.suggest, max_evals=4, trials=trials,
                rstate=np.random.default_rng(123))
    print(f"{label}: best alpha {best['alpha']:.3f}")
    return best


def main():
    # strategy 1: closure capture (small, ref :73-77)
    Xs, ys = gen_data(1)
    tune_alpha(lambda a: train_and_eval(a, Xs, ys), "closure/1MB")

    # strategy 2: shm broadcast (medium, ref :90-101)
    Xm, ym = gen_data(10)
    bc = broadcast((Xm, ym))
    tune_alpha(lambda a: train_and_eval(a, *bc.value), "broadcast/10MB")
    bc.unpersist()

    # strategy 3: save to shared FS, reload per worker (large, ref :114-152)
    Xl, yl = gen_data(50)
    path = os.path.join(tempfile.gettempdir(), "mi355x_large_data.npz")
    np.savez(path, X=Xl, y=yl)

    def obj(a, _path=path):
        data = np.load(_path)
        return train_and_eval(a, data["X"], data["y"])

    tune_alpha(obj, "shared-fs/50MB")
    os.unlink(path)


if __name__ == "__main__":
    main()
