"""applyInPandas-compatible per-group execution engine (W1 core).

The reference's pattern (``group_apply/02_Fine_Grained_Demand_
Forecasting.py:520-528``):

    spark.conf.set("...adaptive.enabled", "false")
    df.repartition(n_tasks, "Product", "SKU")
      .groupBy("Product", "SKU")
      .applyInPandas(build_tune_and_score_model, schema=tuning_schema)

Here the Spark JVM shuffle collapses into an in-process group index
(pandas groupby indices — no row movement), and the executor pool is a
local process pool (one worker per core / per GPU for device-backed fns).
Groups are dispatched in chunks to amortize IPC; a crashed fn on one group
never kills the job — failures are reported per group.

    from mi355x_scale.groupby import LocalFrame
    out = (LocalFrame(pdf)
             .repartition(50, "Product", "SKU")      # accepted, advisory
             .groupBy("Product", "SKU")
             .applyInPandas(fn, schema))             # -> pandas DataFrame
"""
from __future__ import annotations

import traceback
from concurrent.futures import ProcessPoolExecutor
from typing import Callable, List, Optional, Sequence, Tuple

import pandas as pd

_TYPE_MAP = {
    "string": "object", "str": "object",
    "date": "datetime64[ns]", "timestamp": "datetime64[ns]",
    "float": "float64", "double": "float64", "real": "float64",
    "int": "int64", "integer": "int64", "bigint": "int64", "long": "int64",
    "smallint": "int64", "boolean": "bool", "bool": "bool",
}


def parse_schema(schema) -> Optional[List[Tuple[str, str]]]:
    """Spark-DDL-style string ("a string, b float") → [(name, pandas dtype)].
    Also accepts a list of (name, dtype) pairs or None."""
    if schema is None:
        return None
    if isinstance(schema, str):
        out = []
        for part in schema.split(","):
            toks = part.strip().split()
            if len(toks) < 2:
                raise ValueError(f"bad schema field {part!r}")
            name, typ = toks[0], toks[1].lower()
            out.append((name, _TYPE_MAP.get(typ, typ)))
        return out
    return [(n, _TYPE_MAP.get(str(t).lower(), str(t))) for n, t in schema]


def _coerce(df: pd.DataFrame, schema: Optional[List[Tuple[str, str]]]
            ) -> pd.DataFrame:
    if schema is None:
        return df
    cols = []
    for name, dtype in schema:
        if name not in df.columns:
            raise KeyError(f"fn output missing schema column {name!r}")
        col = df[name]
        if dtype != "object" and str(col.dtype) != dtype:
            col = col.astype(dtype)
        cols.append(col.rename(name))
    return pd.concat(cols, axis=1)


def _run_chunk(fn: Callable, groups: List[Tuple[tuple, pd.DataFrame]],
               schema) -> Tuple[List[pd.DataFrame], List[Tuple[tuple, str]]]:
    parsed = parse_schema(schema)
    outs, fails = [], []
    for key, gdf in groups:
        try:
            res = fn(gdf)
            outs.append(_coerce(res, parsed))
        except Exception:
            fails.append((key, traceback.format_exc()))
    return outs, fails


def _run_chunk_pickled(payload: bytes):
    # fn is shipped with cloudpickle so lambdas/closures work (the same
    # serialization contract Spark gives applyInPandas UDFs).
    import cloudpickle
    fn, groups, schema = cloudpickle.loads(payload)
    return _run_chunk(fn, groups, schema)


class GroupedFrame:
    def __init__(self, df: pd.DataFrame, keys: Sequence[str],
                 num_workers: Optional[int] = None,
                 chunk_size: Optional[int] = None):
        self.df = df
        self.keys = list(keys)
        self.num_workers = num_workers
        self.chunk_size = chunk_size
        self.failures: List[Tuple[tuple, str]] = []

    def applyInPandas(self, fn: Callable, schema=None) -> pd.DataFrame:
        import os
        groups = [(k if isinstance(k, tuple) else (k,), g)
                  for k, g in self.df.groupby(self.keys, sort=False)]
        n_workers = self.num_workers or min(len(groups), os.cpu_count() or 1)
        self.failures = []
        if n_workers <= 1 or len(groups) <= 1:
            outs, fails = _run_chunk(fn, groups, schema)
            self.failures = fails
        else:
            chunk = self.chunk_size or max(1, len(groups) // (n_workers * 4))
            chunks = [groups[i:i + chunk]
                      for i in range(0, len(groups), chunk)]
            outs, self.failures = [], []
            import multiprocessing as mp
            import cloudpickle
            with ProcessPoolExecutor(
                    max_workers=n_workers,
                    mp_context=mp.get_context("spawn")) as pool:
                futs = [pool.submit(_run_chunk_pickled,
                                    cloudpickle.dumps((fn, c, schema)))
                        for c in chunks]
                for f in futs:
                    o, e = f.result()
                    outs.extend(o)
                    self.failures.extend(e)
        if not outs:
            if self.failures:
                raise RuntimeError(
                    f"all {len(self.failures)} groups failed; first: "
                    f"{self.failures[0][1]}")
            return pd.DataFrame()
        return pd.concat(outs, ignore_index=True)

    # alias matching pyspark's GroupedData.apply
    apply = applyInPandas


class LocalFrame:
    """Minimal DataFrame wrapper carrying the reference's call chain
    (``repartition(n, *keys).groupBy(*keys).applyInPandas(...)``)."""

    def __init__(self, df: pd.DataFrame, num_workers: Optional[int] = None):
        self.df = df
        self.num_workers = num_workers

    def repartition(self, n: int, *cols: str) -> "LocalFrame":
        # Advisory in-process: there is no shuffle; the group index IS the
        # partitioning. Accepted for call compatibility (ref :525).
        return self

    def groupBy(self, *keys: str) -> GroupedFrame:
        return GroupedFrame(self.df, keys, num_workers=self.num_workers)

    groupby = groupBy


def apply_in_pandas(df: pd.DataFrame, keys: Sequence[str], fn: Callable,
                    schema=None, num_workers: Optional[int] = None
                    ) -> pd.DataFrame:
    """Functional shortcut for the chain above."""
    return GroupedFrame(df, keys, num_workers=num_workers).applyInPandas(
        fn, schema)
