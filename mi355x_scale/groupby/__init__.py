"""W1: applyInPandas-compatible group execution engine."""

from .engine import (LocalFrame, GroupedFrame, apply_in_pandas,  # noqa: F401
                     parse_schema)
