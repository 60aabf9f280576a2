"""Loader unit tests: manifest, sharding exactness, infinite epochs,
batch slicing — the contracts SURVEY.md §4 requires."""
import numpy as np
import pytest

from mi355x_scale.data import (BatchReader, DataLoader, DatasetManifest,
                               TransformSpec, make_batch_reader)


def test_manifest_counts(image_parquet):
    man = DatasetManifest.discover(image_parquet)
    assert man.num_rows == 96
    assert len(man.files) == 2
    assert len(man.row_groups) == 6  # 96 rows / 16 per group
    assert all(rg.num_rows == 16 for rg in man.row_groups)


def test_shards_disjoint_and_cover(image_parquet):
    man = DatasetManifest.discover(image_parquet)
    seen = []
    for shard in range(3):
        seen += [(rg.file_path, rg.row_group) for rg in man.shard(shard, 3)]
    assert len(seen) == len(man.row_groups)
    assert len(set(seen)) == len(seen)


def test_reader_row_exactness_per_shard(image_parquet):
    """Union of rows over all shards at num_epochs=1 == dataset exactly."""
    total = 0
    for shard in range(2):
        r = make_batch_reader(image_parquet, cur_shard=shard, shard_count=2,
                              workers_count=2, num_epochs=1)
        with r:
            for batch in r:
                total += len(batch["label"])
    assert total == 96


def test_reader_multi_epoch(image_parquet):
    r = make_batch_reader(image_parquet, num_epochs=3, workers_count=3)
    with r:
        total = sum(len(b["label"]) for b in r)
    assert total == 3 * 96


def test_reader_infinite(image_parquet):
    """num_epochs=None never raises StopIteration by itself (the petastorm
    contract the reference's Trainer relies on)."""
    r = make_batch_reader(image_parquet, num_epochs=None, workers_count=2,
                          results_queue_size=4)
    got = 0
    with r:
        for batch in r:
            got += len(batch["label"])
            if got > 96 * 2:  # well past one epoch
                break
    assert got > 96 * 2


def test_reader_dummy_pool(image_parquet):
    r = make_batch_reader(image_parquet, reader_pool_type="dummy",
                          num_epochs=1)
    total = sum(len(b["label"]) for b in r)
    assert total == 96


def test_transform_spec(image_parquet):
    spec = TransformSpec(
        func=lambda pdf: {"label2": pdf["label"].to_numpy() * 2},
    )
    r = make_batch_reader(image_parquet, transform_spec=spec, num_epochs=1,
                          schema_fields=["label"])
    with r:
        b = next(iter(r))
    assert set(b) == {"label2"}
    assert (b["label2"] % 2 == 0).all()


def test_dataloader_batching(image_parquet):
    """Fixed batch size with carry across row groups; drop_last."""
    r = make_batch_reader(image_parquet, num_epochs=1, workers_count=1)
    dl = DataLoader(r, batch_size=28, drop_last=True)
    with dl:
        sizes = [len(b["label"]) for b in dl]
    assert all(s == 28 for s in sizes)
    assert len(sizes) == 96 // 28

    r = make_batch_reader(image_parquet, num_epochs=1, workers_count=1)
    dl = DataLoader(r, batch_size=28, drop_last=False)
    with dl:
        sizes = [len(b["label"]) for b in dl]
    assert sum(sizes) == 96
    assert sizes[-1] == 96 % 28


def test_dataloader_values_roundtrip(image_parquet):
    """Bytes decode to the exact rows written (content integrity)."""
    import pyarrow.parquet as pq
    man = DatasetManifest.discover(image_parquet)
    f0 = man.files[0]
    expected = pq.ParquetFile(f0).read_row_group(0).to_pandas()
    r = BatchReader(DatasetManifest.discover([f0]), num_epochs=1,
                    reader_pool_type="dummy")
    batch = next(iter(r))
    got = batch["image"][0]
    want = np.frombuffer(expected["image"].iloc[0], dtype=np.uint8)
    assert np.array_equal(got, want)
    assert batch["label"][0] == expected["label"].iloc[0]


def test_empty_shard_raises(image_parquet):
    man = DatasetManifest.discover(image_parquet)
    with pytest.raises(ValueError):
        BatchReader(man, cur_shard=7, shard_count=8)  # only 6 row groups


def test_reader_process_pool(image_parquet):
    """reader_pool_type='process' (petastorm's third pool type): same
    row-exactness contract as the thread pool."""
    r = make_batch_reader(image_parquet, reader_pool_type="process",
                          workers_count=2, num_epochs=1,
                          results_queue_size=4)
    with r:
        total = sum(len(b["label"]) for b in r)
    assert total == 96


def test_reader_process_pool_infinite_close(image_parquet):
    r = make_batch_reader(image_parquet, reader_pool_type="process",
                          workers_count=2, num_epochs=None,
                          results_queue_size=2)
    got = 0
    with r:
        for b in r:
            got += len(b["label"])
            if got > 96:
                break
    assert got > 96
