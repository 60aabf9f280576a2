"""Workflow job runner — the RUNME.py / Databricks-Workflow equivalent.

The reference deploys a 4-task job DAG with ``timeout_seconds`` and
``max_concurrent_runs`` (``group_apply/RUNME.py:35-106``). Here a job is
a dict of tasks with ``depends_on`` edges executed in topological order
(parallel where the DAG allows), each task a python callable or a
subprocess command, with per-job wall-clock timeout and failure
propagation (downstream tasks of a failed task are skipped).
"""
from __future__ import annotations

import subprocess
import time
import traceback
from concurrent.futures import FIRST_COMPLETED, ThreadPoolExecutor, wait
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional, Union


@dataclass
class Task:
    key: str
    run: Union[Callable[[], object], List[str]]   # callable or argv
    depends_on: List[str] = field(default_factory=list)


@dataclass
class TaskResult:
    key: str
    status: str            # "SUCCESS" | "FAILED" | "SKIPPED"
    seconds: float = 0.0
    error: Optional[str] = None
    output: object = None


class Job:
    def __init__(self, name: str, tasks: List[Task],
                 timeout_seconds: float = 28800.0,   # ref RUNME.py:36
                 max_concurrent_tasks: int = 4):
        self.name = name
        self.tasks = {t.key: t for t in tasks}
        if len(self.tasks) != len(tasks):
            raise ValueError("duplicate task keys")
        for t in tasks:
            for d in t.depends_on:
                if d not in self.tasks:
                    raise ValueError(f"{t.key} depends on unknown {d}")
        self.timeout_seconds = timeout_seconds
        self.max_concurrent_tasks = max_concurrent_tasks

    def _run_one(self, task: Task) -> TaskResult:
        t0 = time.time()
        try:
            if callable(task.run):
                out = task.run()
            else:
                proc = subprocess.run(task.run, capture_output=True,
                                      text=True,
                                      timeout=self.timeout_seconds)
                if proc.returncode != 0:
                    raise RuntimeError(
                        f"exit {proc.returncode}: {proc.stderr[-2000:]}")
                out = proc.stdout
            return TaskResult(task.key, "SUCCESS", time.time() - t0,
                              output=out)
        except Exception:
            return TaskResult(task.key, "FAILED", time.time() - t0,
                              error=traceback.format_exc())

    def run(self) -> Dict[str, TaskResult]:
        deadline = time.time() + self.timeout_seconds
        results: Dict[str, TaskResult] = {}
        pending = dict(self.tasks)
        futures = {}
        with ThreadPoolExecutor(max_workers=self.max_concurrent_tasks) as ex:
            while pending or futures:
                if time.time() > deadline:
                    for k in list(pending):
                        results[k] = TaskResult(k, "SKIPPED",
                                                error="job timeout")
                        del pending[k]
                    break
                # launch every task whose deps succeeded
                for k in list(pending):
                    t = pending[k]
                    deps = [results.get(d) for d in t.depends_on]
                    if any(d and d.status != "SUCCESS" for d in deps):
                        results[k] = TaskResult(
                            k, "SKIPPED",
                            error=f"upstream failed: "
                                  f"{[d.key for d in deps if d and d.status != 'SUCCESS']}")
                        del pending[k]
                        continue
                    if all(d is not None for d in
                           (results.get(x) for x in t.depends_on)):
                        futures[ex.submit(self._run_one, t)] = k
                        del pending[k]
                if not futures:
                    continue
                done, _ = wait(list(futures),
                               return_when=FIRST_COMPLETED,
                               timeout=1.0)
                for f in done:
                    k = futures.pop(f)
                    results[k] = f.result()
        return results
