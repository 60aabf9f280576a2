"""Dataset manifest: Parquet file discovery, row counts, row-group index.

Replaces the reference's Delta-log resolver
(``deep_learning/2.distributed-data-loading-petastorm.py:99-112``:
``DeltaTable(path).file_uris()`` + add-action row counts) with a plain
pyarrow-metadata scan — no Delta transaction log is needed because the
framework owns its storage layout.
"""
from __future__ import annotations

import os
from dataclasses import dataclass, field
from typing import List, Sequence

import pyarrow.parquet as pq


def _strip_scheme(uri: str) -> str:
    if uri.startswith("file://"):
        return uri[len("file://"):]
    return uri


@dataclass(frozen=True)
class RowGroupRef:
    """One shardable unit of work: a single row group of a parquet file."""
    file_path: str
    row_group: int
    num_rows: int


@dataclass
class DatasetManifest:
    """Index of a parquet dataset directory (or explicit file list).

    API parity with the reference's Delta resolver:
      * ``file_uris()``  — list of parquet file paths
      * ``num_rows``     — exact total row count (the reference sums
        ``num_records`` from Delta add-actions; we sum parquet footers).
    """

    files: List[str]
    row_groups: List[RowGroupRef] = field(default_factory=list)

    @classmethod
    def discover(cls, path_or_paths) -> "DatasetManifest":
        if isinstance(path_or_paths, (list, tuple)):
            files = [_strip_scheme(p) for p in path_or_paths]
        else:
            root = _strip_scheme(path_or_paths)
            if os.path.isdir(root):
                files = sorted(
                    os.path.join(dirpath, f)
                    for dirpath, _, fnames in os.walk(root)
                    for f in fnames
                    if f.endswith(".parquet")
                )
            else:
                files = [root]
        if not files:
            raise FileNotFoundError(f"no parquet files under {path_or_paths!r}")
        man = cls(files=files)
        man._index()
        return man

    def _index(self) -> None:
        self.row_groups = []
        for f in self.files:
            meta = pq.ParquetFile(f).metadata
            for rg in range(meta.num_row_groups):
                self.row_groups.append(
                    RowGroupRef(f, rg, meta.row_group(rg).num_rows)
                )

    def file_uris(self) -> List[str]:
        return list(self.files)

    @property
    def num_rows(self) -> int:
        return sum(rg.num_rows for rg in self.row_groups)

    def shard(self, cur_shard: int, shard_count: int) -> List[RowGroupRef]:
        """Row groups owned by shard ``cur_shard`` of ``shard_count``.

        Round-robin by row-group index — the same contract Petastorm uses
        (reference passes ``cur_shard=rank, shard_count=WORLD_SIZE``,
        ``deep_learning/2...py:249-250,376-377``). Deterministic,
        disjoint, and covers every row group when unioned over shards.
        """
        if shard_count <= 0:
            raise ValueError("shard_count must be positive")
        if not (0 <= cur_shard < shard_count):
            raise ValueError(f"cur_shard {cur_shard} not in [0, {shard_count})")
        return [
            rg for i, rg in enumerate(self.row_groups)
            if i % shard_count == cur_shard
        ]


def infer_schema(manifest: DatasetManifest) -> Sequence[str]:
    """Column names of the dataset (from the first file's footer)."""
    return pq.ParquetFile(manifest.files[0]).schema_arrow.names
