"""W2 tests: space DSL, TPE convergence, fmin serial + parallel, failure
propagation — mirrors the reference's demos (hyperopt/1. hyperopt.py,
hyperopt/2. hyperopt on diff sizes of data.py)."""
import numpy as np
import pytest

from mi355x_scale.tune import (GPUTrials, STATUS_FAIL, STATUS_OK, Trials,
                               fmin, hp, rand, scope, tpe)


def test_space_sampling_and_internal_roundtrip():
    rng = np.random.default_rng(123)
    u = hp.uniform("u", -1, 1)
    q = hp.quniform("q", 0, 4, 1)
    ln = hp.lognormal("ln", 0, 1)
    i = scope.int(hp.quniform("i", 0, 4, 1))
    for _ in range(100):
        assert -1 <= u.sample(rng) <= 1
        vq = q.sample(rng)
        assert vq in (0.0, 1.0, 2.0, 3.0, 4.0)
        assert ln.sample(rng) > 0
        vi = i.sample(rng)
        assert isinstance(vi, int) and 0 <= vi <= 4
    # internal roundtrip
    assert ln.from_internal(ln.to_internal(2.5)) == pytest.approx(2.5)
    c = hp.choice("c", ["a", "b", "c"])
    assert c.from_internal(c.to_internal("b")) == "b"


def test_fmin_serial_quadratic():
    """TPE must find the quadratic minimum much better than the prior
    mean — the statistical-fidelity check (SURVEY §7 hard-part 4)."""
    space = {"x": hp.uniform("x", -10, 10)}
    best = fmin(lambda p: (p["x"] - 3.0) ** 2, space, algo=tpe.suggest,
                max_evals=60, rstate=np.random.default_rng(123))
    assert abs(best["x"] - 3.0) < 1.0


def test_fmin_beats_random():
    space = {"x": hp.uniform("x", -10, 10), "y": hp.uniform("y", -10, 10)}

    def obj(p):
        return (p["x"] - 2) ** 2 + (p["y"] + 5) ** 2

    t_tpe, t_rnd = Trials(), Trials()
    fmin(obj, space, algo=tpe.suggest, max_evals=50, trials=t_tpe,
         rstate=np.random.default_rng(0))
    fmin(obj, space, algo=rand.suggest, max_evals=50, trials=t_rnd,
         rstate=np.random.default_rng(0))
    assert min(t_tpe.losses()) <= min(t_rnd.losses()) * 1.5 + 1.0


def test_fmin_deterministic_with_rstate():
    space = {"x": hp.uniform("x", 0, 1)}
    b1 = fmin(lambda p: p["x"], space, max_evals=15,
              rstate=np.random.default_rng(123))
    b2 = fmin(lambda p: p["x"], space, max_evals=15,
              rstate=np.random.default_rng(123))
    assert b1 == b2


def test_fmin_scalar_space_and_dict_result():
    # single-expression space, dict-returning objective (hyperopt/1...py:54-62)
    space = hp.lognormal("C", 0, 1)

    def obj(c):
        return {"loss": abs(c - 1.0), "status": STATUS_OK}

    best = fmin(obj, space, max_evals=30,
                rstate=np.random.default_rng(123))
    assert best["C"] > 0


def test_failed_trials_excluded():
    space = {"x": hp.uniform("x", 0, 1)}
    calls = []

    def obj(p):
        calls.append(p["x"])
        if len(calls) % 3 == 0:
            raise RuntimeError("boom")
        return p["x"]

    t = Trials()
    best = fmin(obj, space, max_evals=12, trials=t,
                rstate=np.random.default_rng(1))
    assert len(t) == 12
    statuses = [r["status"] for r in t.results]
    assert STATUS_FAIL in statuses and STATUS_OK in statuses
    assert best["x"] == min(l for l in t.losses() if l is not None)


def _pool_obj(p):
    return (p["x"] - 0.5) ** 2


def test_gputrials_parallel_cpu():
    """Parallel fan-out path (process pool; device pinning is a no-op on
    CPU). SparkTrials(parallelism=N) alias per hyperopt/1...py:128."""
    from mi355x_scale.tune import SparkTrials
    t = SparkTrials(parallelism=2, use_gpu=False)
    best = fmin(_pool_obj, {"x": hp.uniform("x", 0, 1)}, algo=tpe.suggest,
                max_evals=14, trials=t, rstate=np.random.default_rng(7))
    assert len(t) == 14
    assert 0.0 <= best["x"] <= 1.0


def test_sklearn_svc_demo():
    """The reference's W2 demo shape: SVC/iris accuracy objective
    (hyperopt/1. hyperopt.py:54-62,94-98) run serially."""
    from sklearn import datasets
    from sklearn.model_selection import cross_val_score
    from sklearn.svm import SVC

    iris = datasets.load_iris()
    X, y = iris.data, iris.target

    def objective(c):
        clf = SVC(C=c)
        return -cross_val_score(clf, X, y).mean()

    best = fmin(objective, hp.lognormal("C", 0, 1.0), algo=tpe.suggest,
                max_evals=15, rstate=np.random.default_rng(123))
    assert best["C"] > 0


def test_trials_save_file_resume(tmp_path):
    """the JSONL trial log makes a search resumable —
    10 evals in two halves must total exactly 10 trials, with the resumed
    half's TPE posterior seeded from the saved log."""
    import os
    from mi355x_scale.tune import Trials, fmin, hp, tpe

    path = str(tmp_path / "trials.jsonl")
    space = {"x": hp.uniform("x", -5, 5)}

    def objective(p):
        return (p["x"] - 1.3) ** 2

    fmin(objective, space, algo=tpe.suggest, max_evals=5,
         rstate=np.random.default_rng(0), trials_save_file=path)
    assert os.path.exists(path)
    assert len(Trials.from_jsonl(path)) == 5

    best = fmin(objective, space, algo=tpe.suggest, max_evals=10,
                rstate=np.random.default_rng(1), trials_save_file=path)
    resumed = Trials.from_jsonl(path)
    assert len(resumed) == 10
    assert abs(best["x"] - 1.3) < 2.0
    # the log survives a json round trip with losses intact
    assert all(t["result"]["loss"] is not None for t in resumed.trials)
