"""Single-node process launcher — ``TorchDistributor`` call-compatible.

The reference launches DDP via pyspark's TorchDistributor
(``deep_learning/2.distributed-data-loading-petastorm.py:446-448,468-470``):

    TorchDistributor(num_processes=N, local_mode=True, use_gpu=True)
        .run(main_training_loop, *args)

Here the Spark executor pool is the 8 GPUs of one node: ``run`` forks one
process per GPU, sets the torch.distributed rendezvous env
(MASTER_ADDR=127.0.0.1 + free port, RANK/LOCAL_RANK/WORLD_SIZE/NODE_RANK),
pins each child to its GPU, runs ``fn(*args)`` on every rank, and returns
rank 0's result — the same contract TorchDistributor documents.
"""
from __future__ import annotations

import os
import socket
import traceback
from typing import Any, Callable

import torch.multiprocessing as mp


def _free_port() -> int:
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _entry(rank: int, world_size: int, port: int, fn, args, result_q,
           env_extra):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["NODE_RANK"] = "0"
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    for k, v in (env_extra or {}).items():
        os.environ[k] = str(v)
    try:
        out = fn(*args)
        if rank == 0:
            result_q.put(("ok", out))
    except Exception:
        if rank == 0:
            result_q.put(("err", traceback.format_exc()))
        raise


class TorchDistributor:
    """API-compatible with pyspark.ml.torch.distributor.TorchDistributor for
    the single-node shapes the reference uses (local_mode multi-GPU and the
    1-process degenerate case)."""

    def __init__(self, num_processes: int = 1, local_mode: bool = True,
                 use_gpu: bool = True, env: dict = None,
                 max_restarts: int = 0):
        if not local_mode:
            raise NotImplementedError(
                "multi-node launch is out of scope on a single MI355X node; "
                "use local_mode=True"
            )
        self.num_processes = int(num_processes)
        self.use_gpu = use_gpu
        self.env = env or {}
        # Rank-failure handling: a crashed rank fails the
        # gang; with max_restarts > 0 the whole gang is relaunched (the
        # Spark-task-retry equivalent — state recovery is the training
        # loop's checkpoint/resume job).
        self.max_restarts = int(max_restarts)

    def run(self, fn: Callable, *args) -> Any:
        last_err: Exception = None
        for attempt in range(self.max_restarts + 1):
            try:
                return self._run_once(fn, *args)
            except RuntimeError as e:
                last_err = e
                if attempt < self.max_restarts:
                    import sys
                    print(f"[TorchDistributor] gang failed "
                          f"(attempt {attempt + 1}); relaunching: {e}",
                          file=sys.stderr)
        raise last_err

    def _run_once(self, fn: Callable, *args) -> Any:
        if self.num_processes <= 1:
            # Degenerate path, like the reference's direct call
            # (deep_learning/2...py:425-428): run in-process, world size 1.
            os.environ.setdefault("RANK", "0")
            os.environ.setdefault("LOCAL_RANK", "0")
            os.environ.setdefault("WORLD_SIZE", "1")
            return fn(*args)
        ctx = mp.get_context("spawn")
        result_q = ctx.SimpleQueue()
        port = _free_port()
        procs = []
        for rank in range(self.num_processes):
            p = ctx.Process(
                target=_entry,
                args=(rank, self.num_processes, port, fn, args, result_q,
                      self.env),
                daemon=False,
            )
            p.start()
            procs.append(p)
        status, payload = result_q.get()
        for p in procs:
            p.join()
        failed = [i for i, p in enumerate(procs) if p.exitcode != 0]
        if status == "err":
            raise RuntimeError(f"rank 0 failed:\n{payload}")
        if failed:
            raise RuntimeError(f"ranks {failed} exited non-zero")
        return payload
