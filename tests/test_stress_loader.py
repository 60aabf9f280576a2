"""Producer/consumer stress on the streaming stack (hammer
the bounded-queue pipeline to surface races/deadlocks)."""
import threading

import numpy as np
import pytest

from mi355x_scale.data import BatchReader, DataLoader, DatasetManifest


@pytest.mark.timeout(120)
def test_reader_stress_many_workers_small_queue(image_parquet):
    """Many readers + tiny queue + abrupt close, repeated — no deadlock,
    no lost rows."""
    man = DatasetManifest.discover(image_parquet)
    for trial in range(6):
        r = BatchReader(man, workers_count=8, results_queue_size=1,
                        num_epochs=2)
        total = 0
        with r:
            for b in r:
                total += len(b["label"])
        assert total == 2 * 96

    # abrupt close mid-stream (teardown discipline, ref :277-280)
    for trial in range(6):
        r = BatchReader(man, workers_count=8, results_queue_size=1,
                        num_epochs=None)
        it = iter(r)
        for _ in range(3):
            next(it)
        r.close()   # must join quickly with a full queue
        assert not any(t.is_alive() for t in r._threads)


@pytest.mark.timeout(120)
def test_concurrent_readers_share_dataset(image_parquet):
    """Multiple reader instances over the same files concurrently."""
    man = DatasetManifest.discover(image_parquet)
    totals = []
    lock = threading.Lock()

    def run_one(shard):
        r = BatchReader(man, cur_shard=shard, shard_count=3,
                        workers_count=3, results_queue_size=2, num_epochs=1)
        tot = 0
        with r:
            for b in r:
                tot += len(b["label"])
        with lock:
            totals.append(tot)

    threads = [threading.Thread(target=run_one, args=(s,)) for s in range(3)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=60)
    assert sum(totals) == 96


@pytest.mark.timeout(120)
def test_dataloader_stress_batch_sizes(image_parquet):
    man = DatasetManifest.discover(image_parquet)
    for bs in (1, 7, 16, 96, 100):
        r = BatchReader(man, workers_count=4, results_queue_size=2,
                        num_epochs=1)
        dl = DataLoader(r, batch_size=bs, drop_last=False)
        with dl:
            total = sum(len(b["label"]) for b in dl)
        assert total == 96, f"bs={bs}"
