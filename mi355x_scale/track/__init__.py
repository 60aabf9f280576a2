"""File-based experiment tracking with an MLflow-compatible on-disk layout.

MLflow is the observability spine of the reference (experiment setup
``group_apply/_resources/00-setup.py:71``; run-per-trial logging
``hyperopt/1. hyperopt.py:130``; Lightning MLFlowLogger
``deep_learning/2...py:56-75,365``). mlflow is not installed here, so this
module writes the same ``mlruns/`` directory layout natively:

    mlruns/<experiment_id>/<run_id>/
        meta.yaml
        metrics/<key>      # lines: "<ts_ms> <value> <step>"
        params/<key>       # single value
        tags/<key>

so MLflow UIs/tools pointed at the directory can read the runs.
"""
from __future__ import annotations

import hashlib
import os
import time
import uuid
from typing import Dict, Optional

_DEFAULT_ROOT = os.environ.get("MI355X_MLRUNS", "./mlruns")
_active_experiment = {"name": "Default", "id": "0", "root": _DEFAULT_ROOT}
_active_run_stack = []


def _exp_dir() -> str:
    d = os.path.join(_active_experiment["root"], _active_experiment["id"])
    os.makedirs(d, exist_ok=True)
    meta = os.path.join(d, "meta.yaml")
    if not os.path.exists(meta):
        with open(meta, "w") as f:
            f.write(
                "artifact_location: {loc}\nexperiment_id: '{eid}'\n"
                "lifecycle_stage: active\nname: {name}\n".format(
                    loc=os.path.abspath(d), eid=_active_experiment["id"],
                    name=_active_experiment["name"])
            )
    return d


def set_tracking_root(root: str) -> None:
    _active_experiment["root"] = root


def set_experiment(name: str, experiment_id: Optional[str] = None) -> None:
    """MLflow-compat: select/create the experiment runs land in.

    The id is derived deterministically from the name (md5, not Python's
    per-process salted ``hash``) so every process/run maps the same
    experiment name to the same ``mlruns/<id>/`` directory.
    """
    _active_experiment["name"] = name
    _active_experiment["id"] = experiment_id or str(
        int(hashlib.md5(name.encode()).hexdigest()[:12], 16) % 10**9)
    _exp_dir()


class Run:
    def __init__(self, run_name: Optional[str] = None, nested: bool = False):
        self.run_id = uuid.uuid4().hex
        self.run_name = run_name or f"run-{self.run_id[:8]}"
        self.nested = nested
        self.dir = os.path.join(_exp_dir(), self.run_id)
        for sub in ("metrics", "params", "tags", "artifacts"):
            os.makedirs(os.path.join(self.dir, sub), exist_ok=True)
        self._t0 = int(time.time() * 1000)
        with open(os.path.join(self.dir, "meta.yaml"), "w") as f:
            f.write(
                "artifact_uri: {au}\nexperiment_id: '{eid}'\n"
                "run_id: {rid}\nrun_name: {rn}\nstart_time: {st}\n"
                "status: RUNNING\n".format(
                    au=os.path.abspath(os.path.join(self.dir, "artifacts")),
                    eid=_active_experiment["id"], rid=self.run_id,
                    rn=self.run_name, st=self._t0)
            )

    # -- logging ----------------------------------------------------------
    def log_metric(self, key: str, value: float, step: int = 0) -> None:
        ts = int(time.time() * 1000)
        with open(os.path.join(self.dir, "metrics", key), "a") as f:
            f.write(f"{ts} {value} {step}\n")

    def log_metrics(self, metrics: Dict[str, float], step: int = 0) -> None:
        for k, v in metrics.items():
            self.log_metric(k, v, step)

    def log_param(self, key: str, value) -> None:
        with open(os.path.join(self.dir, "params", key), "w") as f:
            f.write(str(value))

    def log_params(self, params: Dict) -> None:
        for k, v in params.items():
            self.log_param(k, v)

    def set_tag(self, key: str, value) -> None:
        with open(os.path.join(self.dir, "tags", key), "w") as f:
            f.write(str(value))

    def log_artifact(self, local_path: str) -> str:
        import shutil
        dst = os.path.join(self.dir, "artifacts",
                           os.path.basename(local_path))
        shutil.copyfile(local_path, dst)
        return dst

    def end(self, status: str = "FINISHED") -> None:
        meta = os.path.join(self.dir, "meta.yaml")
        with open(meta, "a") as f:
            f.write(f"end_time: {int(time.time() * 1000)}\n"
                    f"final_status: {status}\n")

    def __enter__(self):
        # MLflow layout: nested runs carry the parent run id as a tag
        if _active_run_stack:
            self.set_tag("mlflow.parentRunId", _active_run_stack[-1].run_id)
        _active_run_stack.append(self)
        return self

    def __exit__(self, exc_type, *exc):
        _active_run_stack.pop()
        self.end("FAILED" if exc_type else "FINISHED")
        return False


def start_run(run_name: Optional[str] = None, nested: bool = False) -> Run:
    """MLflow-compat: ``with track.start_run(): ...``."""
    return Run(run_name=run_name, nested=nested)


def active_run() -> Optional[Run]:
    return _active_run_stack[-1] if _active_run_stack else None


def log_metric(key: str, value: float, step: int = 0) -> None:
    r = active_run()
    if r:
        r.log_metric(key, value, step)


def log_param(key: str, value) -> None:
    r = active_run()
    if r:
        r.log_param(key, value)
