"""TransformSpec — per-batch pandas transform declaration.

API-compatible with the surface the reference uses
(``deep_learning/2.distributed-data-loading-petastorm.py:310-318``):

    TransformSpec(func, edit_fields=[("data", np.float32, (3,224,224), False),
                                     ("label", np.int64, (1,), False)],
                  selected_fields=["data", "label"])

``func`` receives a pandas DataFrame of one decoded row-group batch and
returns a DataFrame; ``edit_fields`` declares (name, dtype, shape,
nullable) of produced columns so downstream tensor collation knows the
layout without inspecting data; ``selected_fields`` restricts output
column order.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Callable, List, Optional, Sequence, Tuple

import numpy as np


EditField = Tuple[str, np.dtype, tuple, bool]


@dataclass
class TransformSpec:
    func: Optional[Callable] = None
    edit_fields: List[EditField] = field(default_factory=list)
    removed_fields: List[str] = field(default_factory=list)
    selected_fields: Optional[Sequence[str]] = None

    def field_shape(self, name: str):
        for f in self.edit_fields:
            if f[0] == name:
                return tuple(f[2])
        return None

    def field_dtype(self, name: str):
        for f in self.edit_fields:
            if f[0] == name:
                return np.dtype(f[1])
        return None

    def apply(self, pdf):
        out = self.func(pdf) if self.func is not None else pdf
        if isinstance(out, dict):  # dict-of-arrays fast path
            if self.selected_fields is not None:
                out = {k: out[k] for k in self.selected_fields}
            for col in self.removed_fields:
                out.pop(col, None)
            return out
        for col in self.removed_fields:
            if col in out.columns:
                out = out.drop(columns=[col])
        if self.selected_fields is not None:
            out = out[list(self.selected_fields)]
        return out
