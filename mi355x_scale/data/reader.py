"""Streaming Parquet row-group reader (Petastorm ``make_batch_reader`` compat).

Re-implements, natively, the exact call surface the reference uses
(``deep_learning/2.distributed-data-loading-petastorm.py:246-259``):

    make_batch_reader(parquet_files,
                      transform_spec=...,
                      cur_shard=rank, shard_count=world_size,
                      workers_count=2, reader_pool_type="thread",
                      results_queue_size=20, num_epochs=None)

Design (MI355X-first, no Spark/JVM/petastorm):
  * pyarrow C++ decodes whole row groups on a ``workers_count`` thread
    pool (pyarrow releases the GIL during decode, so threads scale).
  * Decoded batches flow through a **bounded** results queue
    (``results_queue_size``) — the backpressure bound whose OOM
    arithmetic the reference documents at ``deep_learning/2...py:338``
    (workers × queue × rows/rowgroup × rowsize = peak host memory).
  * ``num_epochs=None`` means an **infinite** reader — the contract the
    reference's Trainer relies on (``deep_learning/2...py:218-220``,
    ``limit_train_batches`` imposes epoch length externally). Epoch
    boundaries never starve or deadlock DDP ranks.
  * Sharding is per row group, round-robin (``manifest.shard``).
"""
from __future__ import annotations

import itertools
import queue
import threading
from typing import Dict, List, Optional

import numpy as np
import pyarrow.parquet as pq

from .manifest import DatasetManifest, RowGroupRef
from .transform import TransformSpec

_STOP = object()


class _SpecAsArrowTransform:
    """Adapts a TransformSpec to the process pool's table→dict contract
    (picklable; runs the pandas round-trip inside the worker process)."""

    def __init__(self, spec: TransformSpec):
        self.spec = spec

    def __call__(self, table):
        return self.spec.apply(table.to_pandas())


class BatchReader:
    """Iterator over decoded row-group batches (dict[str, np.ndarray])."""

    def __init__(
        self,
        manifest: DatasetManifest,
        schema_fields: Optional[List[str]] = None,
        transform_spec: Optional[TransformSpec] = None,
        cur_shard: Optional[int] = None,
        shard_count: Optional[int] = None,
        workers_count: int = 4,
        reader_pool_type: str = "thread",
        results_queue_size: int = 10,
        num_epochs: Optional[int] = 1,
        shuffle_row_groups: bool = False,
        seed: Optional[int] = None,
        arrow_transform=None,
    ):
        if reader_pool_type not in ("thread", "process", "dummy"):
            raise ValueError(
                f"reader_pool_type must be 'thread', 'process' or 'dummy', "
                f"got {reader_pool_type!r}")
        if (cur_shard is None) != (shard_count is None):
            raise ValueError("cur_shard and shard_count must be given together")
        self.manifest = manifest
        self.schema_fields = schema_fields
        self.transform_spec = transform_spec
        # arrow_transform: pyarrow.Table -> dict[str, np.ndarray]; the
        # zero-copy fast path (skips pandas entirely). Mutually exclusive
        # with transform_spec.
        self.arrow_transform = arrow_transform
        if arrow_transform is not None and transform_spec is not None:
            raise ValueError("pass either transform_spec or arrow_transform")
        self.num_epochs = num_epochs
        self.workers_count = max(1, workers_count)
        self.reader_pool_type = reader_pool_type
        self.shuffle_row_groups = shuffle_row_groups
        self.seed = seed
        if shard_count is not None:
            self._refs = manifest.shard(cur_shard, shard_count)
        else:
            self._refs = list(manifest.row_groups)
        if not self._refs:
            raise ValueError(
                f"shard {cur_shard}/{shard_count} owns zero row groups "
                f"({len(manifest.row_groups)} total) — write more/smaller row groups"
            )
        self.last_row_consumed = False  # petastorm-compat attribute
        self._counter = itertools.count()  # atomic under the GIL
        self._stop = threading.Event()
        self._results: "queue.Queue" = queue.Queue(maxsize=max(1, results_queue_size))
        self._threads: List[threading.Thread] = []
        self._done_workers = 0
        self._done_lock = threading.Lock()
        self._pf_cache: Dict[str, pq.ParquetFile] = {}
        self._decode_times: List[float] = []
        self._pf_lock = threading.Lock()
        self._proc_pool = None
        if reader_pool_type == "thread":
            for i in range(self.workers_count):
                t = threading.Thread(target=self._worker, name=f"rg-reader-{i}", daemon=True)
                t.start()
                self._threads.append(t)
        elif reader_pool_type == "process":
            from .process_pool import ProcessReaderPool
            at = arrow_transform
            if transform_spec is not None:
                # GIL-heavy user transforms (e.g. per-row JPEG decode)
                # are exactly what the process pool is for: ship the
                # spec via cloudpickle, apply to the decoded pandas
                # batch inside the worker process
                at = _SpecAsArrowTransform(transform_spec)
            self._proc_pool = ProcessReaderPool(
                self._refs, self.workers_count, num_epochs,
                schema_fields, at, results_queue_size)

    # -- work distribution ------------------------------------------------
    def _next_ref(self) -> Optional[RowGroupRef]:
        i = next(self._counter)
        n = len(self._refs)
        epoch, pos = divmod(i, n)
        if self.num_epochs is not None and epoch >= self.num_epochs:
            return None
        if self.shuffle_row_groups:
            rng = np.random.default_rng(
                (self.seed if self.seed is not None else 0) + epoch
            )
            perm = rng.permutation(n)
            pos = int(perm[pos])
        return self._refs[pos]

    def _parquet_file(self, path: str) -> pq.ParquetFile:
        # One ParquetFile handle per file; pyarrow read_row_group is
        # thread-safe for distinct row groups on distinct handles — keep a
        # per-thread handle to stay safe.
        key = f"{threading.get_ident()}:{path}"
        with self._pf_lock:
            pf = self._pf_cache.get(key)
            if pf is None:
                pf = pq.ParquetFile(path)
                self._pf_cache[key] = pf
        return pf

    def _decode(self, ref: RowGroupRef) -> Dict[str, np.ndarray]:
        import os
        if os.environ.get("MI355X_SYNTH_DECODE") == "1":
            # Diagnostic mode: decode each row group once, then replay the
            # cached batch (isolates decode cost from the rest of the
            # pipeline in A/B runs).
            cached = getattr(self, "_synth_batch", None)
            if cached is not None:
                return cached
        pf = self._parquet_file(ref.file_path)
        table = pf.read_row_group(ref.row_group, columns=self.schema_fields)
        if self.arrow_transform is not None:
            out = {k: np.asarray(v)
                   for k, v in self.arrow_transform(table).items()}
            if os.environ.get("MI355X_SYNTH_DECODE") == "1":
                self._synth_batch = out
            return out
        pdf = table.to_pandas()
        if self.transform_spec is not None:
            pdf = self.transform_spec.apply(pdf)
        if isinstance(pdf, dict):  # transform may return dict-of-arrays directly
            return {k: np.asarray(v) for k, v in pdf.items()}
        out: Dict[str, np.ndarray] = {}
        for col in pdf.columns:
            vals = pdf[col].to_numpy()
            if vals.dtype == object:  # column of ndarrays or raw bytes
                first = vals[0]
                if isinstance(first, (bytes, bytearray)):
                    vals = np.stack([np.frombuffer(b, dtype=np.uint8) for b in vals])
                else:
                    vals = np.stack(vals.tolist())
            out[col] = vals
        return out

    # -- worker loop -------------------------------------------------------
    def _worker(self) -> None:
        import os
        import time as _time
        debug = os.environ.get("MI355X_LOADER_DEBUG") == "1"
        try:
            while not self._stop.is_set():
                ref = self._next_ref()
                if ref is None:
                    break
                t0 = _time.perf_counter() if debug else 0.0
                batch = self._decode(ref)
                if debug:
                    self._decode_times.append(_time.perf_counter() - t0)
                while not self._stop.is_set():
                    try:
                        self._results.put(batch, timeout=0.1)
                        break
                    except queue.Full:
                        continue
        finally:
            with self._done_lock:
                self._done_workers += 1
                if self._done_workers == self.workers_count:
                    # All workers finished every epoch: signal end of stream.
                    while not self._stop.is_set():
                        try:
                            self._results.put(_STOP, timeout=0.1)
                            break
                        except queue.Full:
                            continue

    # -- iterator / context manager ---------------------------------------
    def __iter__(self):
        return self

    def __next__(self) -> Dict[str, np.ndarray]:
        if self._proc_pool is not None:
            try:
                return next(self._proc_pool)
            except StopIteration:
                self.last_row_consumed = True
                raise
        if self.reader_pool_type == "dummy":
            ref = self._next_ref()
            if ref is None:
                self.last_row_consumed = True
                raise StopIteration
            return self._decode(ref)
        while True:
            if self._stop.is_set():
                raise StopIteration
            try:
                item = self._results.get(timeout=0.5)
            except queue.Empty:
                continue
            if item is _STOP:
                self.last_row_consumed = True
                raise StopIteration
            return item

    def stop(self) -> None:
        self._stop.set()

    def join(self) -> None:
        for t in self._threads:
            t.join(timeout=5.0)
        self._threads = []

    def close(self) -> None:
        if self._proc_pool is not None:
            self._proc_pool.close()
        self.stop()
        self.join()
        if self._decode_times:
            import sys
            import numpy as _np
            a = _np.array(self._decode_times) * 1e3
            print(f"[reader-debug] decode ms: p50={_np.median(a):.2f} "
                  f"p95={_np.percentile(a, 95):.2f} max={a.max():.2f} "
                  f"n={len(a)}", file=sys.stderr)
            self._decode_times = []

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()
        return False


def make_batch_reader(
    dataset_url_or_urls,
    schema_fields: Optional[List[str]] = None,
    transform_spec: Optional[TransformSpec] = None,
    cur_shard: Optional[int] = None,
    shard_count: Optional[int] = None,
    workers_count: int = 4,
    reader_pool_type: str = "thread",
    results_queue_size: int = 10,
    num_epochs: Optional[int] = 1,
    shuffle_row_groups: bool = False,
    seed: Optional[int] = None,
    **_ignored,
) -> BatchReader:
    """Petastorm-compatible factory (kwarg names preserved verbatim from
    ``deep_learning/2.distributed-data-loading-petastorm.py:246-259``)."""
    manifest = DatasetManifest.discover(dataset_url_or_urls)
    return BatchReader(
        manifest,
        schema_fields=schema_fields,
        transform_spec=transform_spec,
        cur_shard=cur_shard,
        shard_count=shard_count,
        workers_count=workers_count,
        reader_pool_type=reader_pool_type,
        results_queue_size=results_queue_size,
        num_epochs=num_epochs,
        shuffle_row_groups=shuffle_row_groups,
        seed=seed,
    )
