"""Process launch, RCCL bring-up, and shared-memory broadcast."""

from .comm import (  # noqa: F401
    DistContext, init_distributed, topology_probe, assert_xgmi_mesh,
    barrier, destroy, all_reduce_mean,
)
from .launcher import TorchDistributor  # noqa: F401
from .broadcast import Broadcast, broadcast  # noqa: F401
