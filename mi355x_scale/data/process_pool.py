"""Process reader pool (``reader_pool_type="process"``).

Petastorm offers thread/process/dummy pools; the reference uses "thread"
(``deep_learning/2...py:347``). The process pool exists for GIL-heavy
user transforms: each worker process owns a slice of the row-group
sequence (index mod workers), decodes + transforms it, and ships batches
back pickled over a bounded mp.Queue. The thread pool remains the
zero-copy fast path for the built-in transforms.
"""
from __future__ import annotations

import multiprocessing as mp
import queue as pyqueue
from typing import Dict, List, Optional

import numpy as np


def _worker_main(wid: int, nworkers: int, refs: List[tuple],
                 num_epochs: Optional[int], schema_fields,
                 transform_blob: Optional[bytes], out_q, stop_ev,
                 queue_slots: int):
    import pyarrow.parquet as pq
    transform = None
    if transform_blob is not None:
        import cloudpickle
        transform = cloudpickle.loads(transform_blob)
    pf_cache: Dict[str, "pq.ParquetFile"] = {}
    n = len(refs)
    i = wid
    try:
        while not stop_ev.is_set():
            epoch = i // n
            if num_epochs is not None and epoch >= num_epochs:
                break
            path, rg = refs[i % n]
            pf = pf_cache.get(path)
            if pf is None:
                pf = pf_cache[path] = pq.ParquetFile(path)
            table = pf.read_row_group(rg, columns=schema_fields)
            if transform is not None:
                out = transform(table)
                out = {k: np.asarray(v) for k, v in out.items()}
            else:
                pdf = table.to_pandas()
                out = {c: pdf[c].to_numpy() for c in pdf.columns}
            while not stop_ev.is_set():
                try:
                    out_q.put(out, timeout=0.1)
                    break
                except pyqueue.Full:
                    continue
            i += nworkers
    finally:
        try:
            out_q.put(None, timeout=5.0)
        except Exception:
            pass


class ProcessReaderPool:
    """Iterator over decoded batches from worker processes."""

    def __init__(self, refs, workers_count: int,
                 num_epochs: Optional[int], schema_fields,
                 arrow_transform, results_queue_size: int):
        ctx = mp.get_context("spawn")
        self._stop_ev = ctx.Event()
        self._q = ctx.Queue(maxsize=max(2, results_queue_size))
        blob = None
        if arrow_transform is not None:
            import cloudpickle
            blob = cloudpickle.dumps(arrow_transform)
        ref_tuples = [(r.file_path, r.row_group) for r in refs]
        self._procs = []
        self.workers_count = workers_count
        for w in range(workers_count):
            p = ctx.Process(
                target=_worker_main,
                args=(w, workers_count, ref_tuples, num_epochs,
                      schema_fields, blob, self._q, self._stop_ev,
                      results_queue_size),
                daemon=True)
            p.start()
            self._procs.append(p)
        self._done = 0

    def __iter__(self):
        return self

    def __next__(self):
        while True:
            if self._stop_ev.is_set():
                raise StopIteration
            try:
                item = self._q.get(timeout=0.5)
            except pyqueue.Empty:
                if all(not p.is_alive() for p in self._procs):
                    raise StopIteration
                continue
            if item is None:
                self._done += 1
                if self._done >= self.workers_count:
                    raise StopIteration
                continue
            return item

    def close(self):
        self._stop_ev.set()
        # drain so writers blocked on put() can see the event and exit
        try:
            while True:
                self._q.get_nowait()
        except pyqueue.Empty:
            pass
        for p in self._procs:
            p.join(timeout=5.0)
            if p.is_alive():
                p.terminate()
