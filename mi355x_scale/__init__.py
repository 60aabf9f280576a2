"""mi355x_scale — MI355X-native distributed-ML scaling framework.

A from-scratch replacement for the capability surface of the Databricks
"Specialist Session: ML at scale" repo (``sebrahimi1988/dss-ml-at-scale``):

* W3 — Petastorm-style Parquet→PyTorch streaming into DDP training
  (reference: ``deep_learning/2.distributed-data-loading-petastorm.py``),
  rebuilt as a pyarrow row-group decode pool feeding a pinned host ring
  buffer with hipMemcpyAsync side-stream H2D on ROCm.
* W1 — ``applyInPandas``-style per-group model fitting (reference:
  ``group_apply/02_Fine_Grained_Demand_Forecasting.py:523-528``), rebuilt
  as a group-gather engine plus a batched CDNA4 HIP fitting kernel.
* W2 — Hyperopt/SparkTrials-style parallel TPE search (reference:
  ``hyperopt/1. hyperopt.py:128-136``), rebuilt as a host-side TPE with a
  one-trial-per-GPU worker pool.

Target hardware: one node of 8× AMD Instinct MI355X (gfx950), RCCL over
xGMI, PyTorch-ROCm.
"""

__version__ = "0.1.0"

from . import data, models, tune, track  # noqa: F401
