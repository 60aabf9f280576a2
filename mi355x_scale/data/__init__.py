"""Data plane: Parquet manifest, streaming row-group reader (Petastorm
``make_batch_reader`` compat), batch loader, pinned/side-stream device
staging, and the synthetic data generators."""

from .manifest import DatasetManifest, RowGroupRef, infer_schema  # noqa: F401
from .transform import TransformSpec  # noqa: F401
from .reader import BatchReader, make_batch_reader  # noqa: F401
from .loader import DataLoader, DeviceLoader  # noqa: F401
from . import generator  # noqa: F401
