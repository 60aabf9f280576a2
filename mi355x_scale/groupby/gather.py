"""Vectorized group gather: long-format frame → dense [G, T] panel.

This is the N1 component's hot half: the reference pays a
Spark JVM hash-shuffle to co-locate each (Product, SKU) group
(``group_apply/02_...py:525-528``); here co-location is a single
vectorized scatter — keys and dates are factorized to integer codes and
values land at ``panel[group_code, time_code]`` in one numpy indexing op
(memory-bound, no per-group Python loop). The output is the time-major
matrix the batched CDNA4 kernel consumes.
"""
from __future__ import annotations

from typing import List, Sequence, Tuple

import numpy as np
import pandas as pd


def panel_from_long(df: pd.DataFrame, keys: Sequence[str], time_col: str,
                    value_col: str,
                    ) -> Tuple[np.ndarray, pd.Index, np.ndarray]:
    """Returns (panel [G,T] float32, group_index, time_values).

    Rows may arrive in any order (the "shuffle" is the scatter). Missing
    (group, time) cells become NaN; duplicate cells keep the last row.
    """
    if len(keys) == 1:
        gcodes, gindex = pd.factorize(df[keys[0]], sort=True)
    else:
        gcodes, gindex = pd.factorize(
            pd.MultiIndex.from_frame(df[list(keys)]), sort=True)
    tcodes, tvals = pd.factorize(df[time_col], sort=True)
    G, T = len(gindex), len(tvals)
    panel = np.full((G, T), np.nan, dtype=np.float32)
    panel[gcodes, tcodes] = df[value_col].to_numpy(dtype=np.float32)
    return panel, gindex, np.asarray(tvals)


def long_from_panel(panel: np.ndarray, gindex, tvals,
                    keys: Sequence[str], time_col: str,
                    value_cols: List[Tuple[str, np.ndarray]]
                    ) -> pd.DataFrame:
    """Inverse: [G,T] matrices back to a long frame (keys × time rows)."""
    G, T = panel.shape
    data = {}
    if isinstance(gindex, pd.MultiIndex):
        for li, name in enumerate(keys):
            data[name] = np.repeat(gindex.get_level_values(li).to_numpy(), T)
    else:
        data[keys[0]] = np.repeat(np.asarray(gindex), T)
    data[time_col] = np.tile(np.asarray(tvals), G)
    for name, mat in value_cols:
        data[name] = mat.reshape(-1)
    return pd.DataFrame(data)
