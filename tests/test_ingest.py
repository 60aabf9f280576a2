"""Image ingest: directory → parquet + vocab, consumable by the loader."""
import json
import os

import numpy as np

from mi355x_scale.data.ingest import (build_label_vocab, discover_images,
                                      ingest_image_directory)


def _make_images(root, classes=("cat", "dog"), per_class=5, size=(40, 30)):
    from PIL import Image
    rng = np.random.default_rng(0)
    for c in classes:
        d = os.path.join(root, c)
        os.makedirs(d, exist_ok=True)
        for i in range(per_class):
            arr = rng.integers(0, 256, size=(size[1], size[0], 3),
                               dtype=np.uint8)
            Image.fromarray(arr).save(os.path.join(d, f"img{i}.png"))


def test_ingest_roundtrip(tmp_path):
    src = tmp_path / "src"
    out = tmp_path / "out"
    _make_images(str(src))
    summary = ingest_image_directory(str(src), str(out), image_hw=(32, 32),
                                     rows_per_group=4, rows_per_file=8,
                                     workers=4)
    assert summary["rows"] == 10
    assert summary["num_classes"] == 2
    assert summary["skipped"] == 0
    vocab = json.load(open(out / "label_vocab.json"))
    assert vocab == {"cat": 0, "dog": 1}

    # streamable by the standard loader stack
    from mi355x_scale.data import DatasetManifest
    from mi355x_scale.train.datamodule import ImageStreamDataModule
    import torch
    man = DatasetManifest.discover(str(out))
    assert man.num_rows == 10
    dm = ImageStreamDataModule(str(out), batch_size=5, workers_count=1,
                               image_hw=(32, 32),
                               device=torch.device("cpu"))
    loader = dm.val_dataloader()
    batch = next(iter(loader))
    assert batch["image"].shape == (5, 32, 32, 3)
    assert batch["label"].max() <= 1
    dm.teardown()


def test_vocab_contiguous():
    items = [("a/x.jpg", "z"), ("b/y.jpg", "a"), ("c/z.jpg", "m")]
    v = build_label_vocab(items)
    assert v == {"a": 0, "m": 1, "z": 2}


def test_ingest_jpeg_format_feeds_decode_path(tmp_path):
    """store_format='jpeg': rows are encoded JPEG bytes consumable by
    the image_format='jpeg' streaming loader."""
    import numpy as np
    import torch
    from PIL import Image

    from mi355x_scale.data.ingest import ingest_image_directory
    from mi355x_scale.train import ImageStreamDataModule

    src = tmp_path / "src"
    rng = np.random.default_rng(0)
    for cls in ("cat", "dog"):
        (src / cls).mkdir(parents=True)
        for i in range(6):
            yy, xx = np.mgrid[0:80, 0:90]
            img = np.clip(np.stack([127 + 100 * np.sin(xx / 9.0),
                                    127 + 100 * np.cos(yy / 11.0),
                                    rng.integers(0, 255, (80, 90))],
                                   axis=-1), 0, 255).astype(np.uint8)
            Image.fromarray(img).save(src / cls / f"{i}.jpg")
    out = tmp_path / "ds"
    summary = ingest_image_directory(str(src), str(out), image_hw=(64, 64),
                                     rows_per_group=4, rows_per_file=8,
                                     store_format="jpeg")
    assert summary["rows"] == 12 and summary["skipped"] == 0
    dm = ImageStreamDataModule(str(out), batch_size=4, workers_count=2,
                               image_format="jpeg", image_hw=(64, 64),
                               device=torch.device("cpu"))
    batch = next(iter(dm.train_dataloader()))
    assert batch["image"].shape == (4, 64, 64, 3)
    assert batch["label"].max() <= 1
    dm.teardown()
