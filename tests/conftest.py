import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: needs an MI355X GPU (run with -m gpu on a GPU box)"
    )


def pytest_collection_modifyitems(config, items):
    # Auto-skip gpu tests when no GPU is present and they were selected.
    import torch
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture(scope="session")
def image_parquet(tmp_path_factory):
    """Small synthetic image dataset shared across tests."""
    from mi355x_scale.data.generator import write_image_parquet
    d = tmp_path_factory.mktemp("imgs")
    write_image_parquet(str(d), num_rows=96, image_hw=(32, 32),
                        num_classes=10, rows_per_group=16, rows_per_file=48)
    return str(d)
