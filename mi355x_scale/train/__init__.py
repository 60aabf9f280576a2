"""Training loop: ScaleModule/DataModule contracts, DDP Trainer,
checkpoint/resume."""

from .module import ScaleModule, DataModule, ImageClassifier, multiclass_accuracy  # noqa: F401
from .trainer import Trainer  # noqa: F401
from .checkpoint import CheckpointManager  # noqa: F401
from .datamodule import ImageStreamDataModule  # noqa: F401
