"""``fmin`` / ``Trials`` / ``GPUTrials`` — the hyperopt driver surface.

Re-implements the call pattern the reference uses
(``hyperopt/1. hyperopt.py:94-98,128-136``; ``group_apply/02_Fine_Grained_
Demand_Forecasting.py:304-315`` and the nested sequential fmin ``:469``):

    best = fmin(fn=objective, space=space, algo=tpe.suggest,
                max_evals=50, trials=GPUTrials(parallelism=8),
                rstate=np.random.default_rng(123))

TPE suggest runs on the host; trial evaluation fans out one process per
GPU (``HIP_VISIBLE_DEVICES`` pinning), with new proposals conditioned on
completed trials (async dispatch up to ``parallelism``, like SparkTrials).
Failed trials report STATUS_FAIL and are excluded from the TPE posterior
(SURVEY §5.3).
"""
from __future__ import annotations

import os
import traceback
from concurrent.futures import FIRST_COMPLETED, ProcessPoolExecutor, wait
from typing import Any, Callable, Dict, List, Optional

import numpy as np

from .space import bind_params
from .tpe import TPE

STATUS_OK = "ok"
STATUS_FAIL = "fail"


class Trials:
    """Trial store (hyperopt-shaped): ``trials`` is a list of dicts with
    ``result`` / ``misc.vals``; ``losses()``, ``best_trial`` provided."""

    parallelism = 1

    def __init__(self):
        self.trials: List[Dict] = []

    def record(self, tid: int, params: Dict, result: Dict) -> None:
        self.trials.append({
            "tid": tid,
            "misc": {"vals": {k: [v] for k, v in params.items()}},
            "result": result,
            "state": 2,  # JOB_STATE_DONE
        })

    def losses(self) -> List[Optional[float]]:
        return [t["result"].get("loss") for t in self.trials]

    @property
    def results(self) -> List[Dict]:
        return [t["result"] for t in self.trials]

    @property
    def best_trial(self) -> Dict:
        ok = [t for t in self.trials
              if t["result"].get("status") == STATUS_OK
              and t["result"].get("loss") is not None]
        if not ok:
            raise ValueError("no successful trials")
        return min(ok, key=lambda t: t["result"]["loss"])

    def __len__(self):
        return len(self.trials)

    # --- JSONL persistence (the trial log makes a search
    # resumable; hyperopt's fmin(trials_save_file=...) contract) ---

    def to_jsonl(self, path: str) -> None:
        import json
        tmp = f"{path}.tmp"
        with open(tmp, "w") as f:
            for t in self.trials:
                f.write(json.dumps(t, default=float) + "\n")
        os.replace(tmp, path)  # atomic: a crash never truncates the log

    @classmethod
    def from_jsonl(cls, path: str, **kwargs) -> "Trials":
        import json
        tr = cls(**kwargs)
        with open(path) as f:
            tr.trials = [json.loads(line) for line in f if line.strip()]
        return tr


def _default_parallelism() -> int:
    try:
        import torch
        if torch.cuda.is_available():
            return torch.cuda.device_count()
    except Exception:
        pass
    return min(8, os.cpu_count() or 1)


class GPUTrials(Trials):
    """SparkTrials equivalent: trial objectives run in a process pool, one
    worker per GPU (``HIP_VISIBLE_DEVICES`` pinned), losses gathered back
    to the host TPE loop."""

    def __init__(self, parallelism: Optional[int] = None,
                 use_gpu: Optional[bool] = None):
        super().__init__()
        self.parallelism = parallelism or _default_parallelism()
        if use_gpu is None:
            try:
                import torch
                use_gpu = torch.cuda.is_available()
            except Exception:
                use_gpu = False
        self.use_gpu = use_gpu


# SparkTrials(parallelism=N) call-compat alias
SparkTrials = GPUTrials


def _worker_init(device_queue):
    """Pin this pool worker to one GPU before torch is ever imported."""
    try:
        dev = device_queue.get_nowait()
    except Exception:
        dev = None
    if dev is not None:
        os.environ["HIP_VISIBLE_DEVICES"] = str(dev)
        os.environ["CUDA_VISIBLE_DEVICES"] = str(dev)


def _eval_trial_pickled(fn_blob: bytes, args_blob: bytes) -> Dict:
    """Pool entry: fn ships via cloudpickle so closures/lambdas work (the
    serialization contract SparkTrials gives objectives)."""
    import cloudpickle
    fn = cloudpickle.loads(fn_blob)
    args = cloudpickle.loads(args_blob)
    return _eval_trial(fn, args)


def _eval_trial(fn: Callable, args: Any) -> Dict:
    try:
        out = fn(args)
    except Exception:
        return {"status": STATUS_FAIL, "loss": None,
                "error": traceback.format_exc()}
    if isinstance(out, dict):
        out.setdefault("status", STATUS_OK)
        return out
    return {"loss": float(out), "status": STATUS_OK}


def _make_rng(rstate) -> np.random.Generator:
    if rstate is None:
        return np.random.default_rng()
    if isinstance(rstate, np.random.Generator):
        return rstate
    if isinstance(rstate, np.random.RandomState):
        return np.random.default_rng(rstate.randint(2**31))
    return np.random.default_rng(rstate)


def fmin(fn: Callable, space, algo=None, max_evals: int = 10,
         trials: Optional[Trials] = None, rstate=None,
         verbose: bool = False, trials_save_file: Optional[str] = None,
         **_ignored) -> Dict:
    """Minimize ``fn`` over ``space``; returns the best parameter dict.

    ``max_evals`` counts TOTAL trials in ``trials`` (hyperopt semantics):
    passing pre-populated trials — e.g. ``Trials.from_jsonl(...)`` or an
    existing ``trials_save_file`` — resumes the search, with completed
    trials seeding the TPE posterior. ``trials_save_file`` persists the
    trial log (JSONL, atomic rewrite) after every trial.
    """
    rng = _make_rng(rstate)
    if (trials is None and trials_save_file
            and os.path.exists(trials_save_file)):
        trials = Trials.from_jsonl(trials_save_file)
    trials = trials if trials is not None else Trials()
    optimizer = algo() if algo is not None else TPE()
    # seed the posterior from any completed trials (resume path)
    history: List[tuple] = [
        ({k: v[0] for k, v in t["misc"]["vals"].items()},
         t["result"]["loss"])
        for t in trials.trials
        if t["result"].get("status") == STATUS_OK
        and t["result"].get("loss") is not None
    ]
    start = len(trials)

    def _record(tid, params, result):
        trials.record(tid, params, result)
        if result.get("status") == STATUS_OK and result.get("loss") is not None:
            history.append((params, result["loss"]))
        if trials_save_file:
            trials.to_jsonl(trials_save_file)
        if verbose:
            print(f"[fmin] trial {tid}: loss={result.get('loss')} "
                  f"status={result.get('status')} params={params}")
        try:
            from .. import track
            run = track.active_run()
            if run is not None and result.get("loss") is not None:
                run.log_metric("trial_loss", result["loss"], step=tid)
        except Exception:
            pass

    par = getattr(trials, "parallelism", 1)
    if par <= 1:
        for tid in range(start, max_evals):
            params = optimizer.propose(space, history, rng)
            result = _eval_trial(fn, bind_params(space, params))
            _record(tid, params, result)
    else:
        import cloudpickle
        import multiprocessing as mp
        fn_blob = cloudpickle.dumps(fn)
        ctx = mp.get_context("spawn")
        dq = ctx.Queue()
        n_dev = par
        if getattr(trials, "use_gpu", False):
            import torch
            n_dev = max(1, torch.cuda.device_count())
        for i in range(par):
            dq.put(i % n_dev)
        with ProcessPoolExecutor(max_workers=par, mp_context=ctx,
                                 initializer=_worker_init,
                                 initargs=(dq,)) as pool:
            pending = {}
            issued = start
            done_n = start
            while done_n < max_evals:
                while len(pending) < par and issued < max_evals:
                    params = optimizer.propose(space, history, rng)
                    fut = pool.submit(
                        _eval_trial_pickled, fn_blob,
                        cloudpickle.dumps(bind_params(space, params)))
                    pending[fut] = (issued, params)
                    issued += 1
                done, _ = wait(list(pending), return_when=FIRST_COMPLETED)
                for fut in done:
                    tid, params = pending.pop(fut)
                    try:
                        result = fut.result()
                    except Exception:
                        result = {"status": STATUS_FAIL, "loss": None,
                                  "error": traceback.format_exc()}
                    _record(tid, params, result)
                    done_n += 1

    best = trials.best_trial
    vals = {k: v[0] for k, v in best["misc"]["vals"].items()}
    # hyperopt returns option INDICES for hp.choice parameters (so
    # reference-style `options[best[key]]` / `space_eval(space, best)`
    # works); internal trial storage keeps raw values — the one place the
    # index convention surfaces is this return dict.
    from .space import Choice, IntCast, flatten_space
    flat = flatten_space(space)
    out = {}
    for k, v in vals.items():
        expr = flat.get(k)
        base = expr.inner if isinstance(expr, IntCast) else expr
        if isinstance(base, Choice):
            out[k] = int(base.to_internal(v))
        else:
            out[k] = v
    return out
