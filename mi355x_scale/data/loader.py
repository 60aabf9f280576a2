"""DataLoader over a BatchReader, plus the device-side streaming stage.

``DataLoader`` mirrors ``petastorm.pytorch.DataLoader(reader, batch_size)``
(reference use: ``deep_learning/2.distributed-data-loading-petastorm.py:
256-259``): slices variable-size row-group batches into fixed-size torch
batches, carrying the remainder across row groups.

``DeviceLoader`` is the MI355X replacement for Petastorm's "just call
``.to(device)`` in the training loop" hidden cost: batches are staged into
**pinned** host buffers (ring of ``depth``) and copied H2D with
``hipMemcpyAsync`` on a dedicated **side stream**, so the copy of batch
k+1 overlaps forward/backward of batch k. Consumer synchronizes via a
recorded event, never a device-wide sync.
"""
from __future__ import annotations

from typing import Dict, Iterator, Optional

import numpy as np
import torch

from .reader import BatchReader

_TORCH_DTYPE = {
    np.dtype(np.float32): torch.float32,
    np.dtype(np.float64): torch.float64,
    np.dtype(np.float16): torch.float16,
    np.dtype(np.int64): torch.int64,
    np.dtype(np.int32): torch.int32,
    np.dtype(np.int16): torch.int16,
    np.dtype(np.uint8): torch.uint8,
    np.dtype(np.int8): torch.int8,
    np.dtype(np.bool_): torch.bool,
}


def _to_tensor(arr: np.ndarray) -> torch.Tensor:
    if arr.dtype == np.dtype(np.uint16):
        arr = arr.astype(np.int32)
    arr = np.ascontiguousarray(arr)
    if not arr.flags.writeable:
        # Zero-copy views of immutable arrow buffers: we only ever read
        # from these tensors (they are staged into pinned buffers), so the
        # "non-writable array" warning is noise.
        import warnings
        with warnings.catch_warnings():
            warnings.simplefilter("ignore")
            return torch.from_numpy(arr)
    return torch.from_numpy(arr)


class DataLoader:
    """Fixed-batch-size iterator of ``dict[str, torch.Tensor]``."""

    def __init__(self, reader: BatchReader, batch_size: int = 1,
                 drop_last: bool = True):
        self.reader = reader
        self.batch_size = int(batch_size)
        self.drop_last = drop_last

    def __iter__(self) -> Iterator[Dict[str, torch.Tensor]]:
        carry: Optional[Dict[str, np.ndarray]] = None
        for rg_batch in self.reader:
            if carry is not None:
                rg_batch = {
                    k: np.concatenate([carry[k], rg_batch[k]], axis=0)
                    for k in rg_batch
                }
                carry = None
            n = len(next(iter(rg_batch.values())))
            pos = 0
            while n - pos >= self.batch_size:
                yield {
                    k: _to_tensor(v[pos:pos + self.batch_size])
                    for k, v in rg_batch.items()
                }
                pos += self.batch_size
            if pos < n:
                carry = {k: v[pos:] for k, v in rg_batch.items()}
        if carry is not None and not self.drop_last:
            yield {k: _to_tensor(v) for k, v in carry.items()}

    def close(self):
        self.reader.close()

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()
        return False


class DeviceLoader:
    """Pinned-ring + side-stream H2D prefetcher wrapping a DataLoader.

    The replacement for Petastorm's synchronous host→device copy: the
    decode threads produce CPU batches; this stage owns ``depth`` pinned
    slots and a non-default HIP stream. ``hipMemcpyAsync`` (issued by
    ``Tensor.copy_(non_blocking=True)`` from pinned memory) for batch k+1
    runs while the consumer computes on batch k; an event recorded on the
    side stream is waited on by the compute stream — no global syncs.
    """

    def __init__(self, loader: DataLoader, device, depth: int = 2,
                 dtype_map: Optional[Dict[str, torch.dtype]] = None,
                 stagers: int = 2):
        self.loader = loader
        self.device = torch.device(device)
        self.depth = max(2, depth)
        self.dtype_map = dtype_map or {}
        self.stagers = max(1, stagers)
        self._use_cuda = self.device.type == "cuda"
        if self._use_cuda:
            # one copy stream per stager thread
            self.copy_streams = [torch.cuda.Stream(device=self.device)
                                 for _ in range(self.stagers)]
        self._pinned: Dict[tuple, Dict[str, torch.Tensor]] = {}
        self._slot_events: Dict[tuple, torch.cuda.Event] = {}

    def _pin_slot(self, slot: int, batch: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
        # A slot may be rewritten only after its previous H2D copy finished
        # (the event is recorded on the copy stream right after the copy).
        prev_ev = self._slot_events.get(slot)
        if prev_ev is not None:
            prev_ev.synchronize()
        pinned = self._pinned.get(slot)
        if pinned is None or any(
            pinned[k].shape != v.shape or pinned[k].dtype != v.dtype
            for k, v in batch.items()
        ):
            pinned = {
                k: torch.empty_like(v, pin_memory=True) for k, v in batch.items()
            }
            self._pinned[slot] = pinned
        for k, v in batch.items():
            pinned[k].copy_(v)
        return pinned

    def __iter__(self):
        if not self._use_cuda:
            for batch in self.loader:
                yield {
                    k: v.to(self.dtype_map.get(k, v.dtype))
                    for k, v in batch.items()
                }
            return
        yield from self._iter_cuda()

    def _iter_cuda(self):
        """Batch assembly + pin-copy + async H2D run on ``stagers``
        background threads (the HIP runtime is thread-safe), each with its
        own pinned-slot ring and copy stream; the consumer thread only
        dequeues ready (event, device_batch) pairs and inserts a stream
        wait — host-side staging never blocks the training step, and with
        2+ stagers a slow batch assembly overlaps the next one."""
        import queue as _q
        import threading

        out_q: "_q.Queue" = _q.Queue(maxsize=self.depth * self.stagers)
        stop = threading.Event()
        device = self.device
        src_it = iter(self.loader)
        src_lock = threading.Lock()

        import os
        import time as _time
        debug = os.environ.get("MI355X_LOADER_DEBUG") == "1"
        stats = {"dequeue": [], "pin": [], "h2d_issue": []} if debug else None
        stats_lock = threading.Lock()

        def _produce(sid: int):
            torch.cuda.set_device(device)
            copy_stream = self.copy_streams[sid]
            try:
                slot = 0
                while True:
                    t0 = _time.perf_counter()
                    with src_lock:
                        try:
                            cpu_batch = next(src_it)
                        except StopIteration:
                            return
                    if stop.is_set():
                        return
                    t1 = _time.perf_counter()
                    pinned = self._pin_slot((sid, slot), cpu_batch)
                    t2 = _time.perf_counter()
                    with torch.cuda.stream(copy_stream):
                        dev = {}
                        for k, v in pinned.items():
                            d = v.to(device, non_blocking=True)
                            want = self.dtype_map.get(k)
                            if want is not None and d.dtype != want:
                                d = d.to(want)
                            dev[k] = d
                        ev = torch.cuda.Event()
                        ev.record(copy_stream)
                    if debug:
                        t3 = _time.perf_counter()
                        with stats_lock:
                            stats["dequeue"].append(t1 - t0)
                            stats["pin"].append(t2 - t1)
                            stats["h2d_issue"].append(t3 - t2)
                    self._slot_events[(sid, slot)] = ev
                    slot = (slot + 1) % self.depth
                    while not stop.is_set():
                        try:
                            out_q.put((ev, dev), timeout=0.1)
                            break
                        except _q.Full:
                            continue
            finally:
                if debug and sid == 0 and stats["dequeue"]:
                    import numpy as _np
                    import sys
                    with stats_lock:
                        for k, v in stats.items():
                            a = _np.array(v) * 1e3
                            print(f"[loader-debug] {k}: "
                                  f"p50={_np.median(a):.2f} "
                                  f"p95={_np.percentile(a, 95):.2f} "
                                  f"max={a.max():.2f} n={len(a)}",
                                  file=sys.stderr)
                while not stop.is_set():
                    try:
                        out_q.put(None, timeout=0.1)
                        break
                    except _q.Full:
                        continue

        threads = [threading.Thread(target=_produce, args=(i,),
                                    name=f"h2d-stager-{i}", daemon=True)
                   for i in range(self.stagers)]
        for t in threads:
            t.start()
        done_sentinels = 0
        try:
            while True:
                item = out_q.get()
                if item is None:
                    done_sentinels += 1
                    if done_sentinels == self.stagers:
                        return
                    continue
                ev, dev = item
                cur = torch.cuda.current_stream(self.device)
                cur.wait_event(ev)
                # The device tensors were allocated on a stager's copy
                # stream; tell the caching allocator the consumer stream
                # uses them, or freeing a batch mid-flight lets the block
                # be recycled for the next H2D while the compute stream is
                # still reading it.
                for t in dev.values():
                    t.record_stream(cur)
                yield dev
        finally:
            stop.set()

    def close(self):
        self.loader.close()

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()
        return False
