"""Full-diagnostic run of the two-rank graph step (NaN forensics)."""
import sys

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from mi355x_scale.parallel import TorchDistributor  # noqa: E402
from tests.test_graph_multirank import _worker  # noqa: E402

if __name__ == "__main__":
    diag = TorchDistributor(num_processes=2, use_gpu=True).run(_worker)
    for k in sorted(diag):
        print(k, "=", diag[k])
