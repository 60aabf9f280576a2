"""Utilities: job DAG runner, per-stage timers, torch.profiler wrapper."""

from .jobs import Job, Task, TaskResult  # noqa: F401
from .profiling import StageTimer, torch_profile  # noqa: F401
