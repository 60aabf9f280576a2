"""Tracing/profiling helpers.

The reference has only wall-clock prints (``deep_learning/2...py:184``);
here: per-stage timers for the loader/training pipeline, and a
``torch.profiler`` wrapper for the ROCm backend (kineto traces open in
chrome://tracing / perfetto). Kernel-level numbers come from rocprofv3
(see profiles/).
"""
from __future__ import annotations

import contextlib
import time
from collections import defaultdict
from typing import Dict, Optional


class StageTimer:
    """Accumulates wall time per named stage.

        timer = StageTimer()
        with timer("decode"): ...
        print(timer.report())
    """

    def __init__(self):
        self._acc: Dict[str, float] = defaultdict(float)
        self._n: Dict[str, int] = defaultdict(int)

    @contextlib.contextmanager
    def __call__(self, stage: str):
        t0 = time.perf_counter()
        try:
            yield
        finally:
            self._acc[stage] += time.perf_counter() - t0
            self._n[stage] += 1

    def add(self, stage: str, seconds: float) -> None:
        self._acc[stage] += seconds
        self._n[stage] += 1

    def report(self) -> str:
        lines = []
        for k in sorted(self._acc, key=lambda k: -self._acc[k]):
            n = self._n[k]
            tot = self._acc[k]
            lines.append(f"{k:24s} total {tot * 1e3:9.1f} ms  "
                         f"n={n:6d}  avg {tot / n * 1e3:8.3f} ms")
        return "\n".join(lines)

    def reset(self) -> None:
        self._acc.clear()
        self._n.clear()


@contextlib.contextmanager
def torch_profile(out_dir: str = "./torchprof", wait: int = 1,
                  warmup: int = 2, active: int = 5,
                  record_shapes: bool = False):
    """torch.profiler over the ROCm backend; writes a chrome trace per
    worker into ``out_dir``. Usage:

        with torch_profile("./prof") as prof:
            for step ...: train(); prof.step()
    """
    import torch
    from torch.profiler import (ProfilerActivity, profile, schedule,
                                tensorboard_trace_handler)
    activities = [ProfilerActivity.CPU]
    if torch.cuda.is_available():
        activities.append(ProfilerActivity.CUDA)
    with profile(
        activities=activities,
        schedule=schedule(wait=wait, warmup=warmup, active=active,
                          repeat=1),
        on_trace_ready=tensorboard_trace_handler(out_dir),
        record_shapes=record_shapes,
    ) as prof:
        yield prof
