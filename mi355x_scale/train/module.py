"""ScaleModule / DataModule — the LightningModule/LightningDataModule
surface the reference's W3 builds on (``deep_learning/2.distributed-data-
loading-petastorm.py:135-208`` model, ``:224-318`` datamodule), without
Lightning.
"""
from __future__ import annotations

from typing import Any, Dict, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..models import build_model


class ScaleModule(nn.Module):
    """Base training module: ``training_step`` / ``validation_step`` /
    ``configure_optimizers`` + ``self.log`` — the subset of the
    LightningModule contract the reference uses."""

    def __init__(self):
        super().__init__()
        self._logged: Dict[str, float] = {}
        self.trainer = None  # set by Trainer.fit
        self.current_epoch = 0
        self.global_step = 0

    def log(self, key: str, value, prog_bar: bool = False,
            sync_dist: bool = False, **_):
        if (torch.is_tensor(value) and value.is_cuda
                and torch.cuda.is_current_stream_capturing()):
            return  # inside hipGraph capture: no host readbacks possible
        v = float(value.detach() if torch.is_tensor(value) else value)
        self._logged[key] = v
        if self.trainer is not None:
            self.trainer._on_module_log(key, v, sync_dist)

    def training_step(self, batch, batch_idx: int):  # pragma: no cover
        raise NotImplementedError

    def validation_step(self, batch, batch_idx: int):  # pragma: no cover
        return None

    def configure_optimizers(self):  # pragma: no cover
        raise NotImplementedError


def multiclass_accuracy(logits: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
    """torchmetrics.functional.accuracy(task="multiclass") equivalent
    (reference: ``deep_learning/2...py:197``)."""
    return (logits.argmax(dim=-1) == target).float().mean()


class ImageClassifier(ScaleModule):
    """ImageNetClassificationModel equivalent (``deep_learning/2...py:
    135-208``): backbone + Adam + cross-entropy train step + val step
    logging loss and multiclass accuracy."""

    # split-backward comm overlap cut (train/graphstep.py): backward of
    # layer3/layer4/fc (~94% of grad bytes) completes first, so its
    # all-reduce overlaps the early-layer backward replay
    comm_overlap_boundary = "model.layer2"

    def __init__(self, model_name: str = "resnet18", num_classes: int = 1000,
                 lr: float = 1e-5, channels_last: bool = True):
        super().__init__()
        self.model = build_model(model_name, num_classes)
        self.lr = lr
        self.channels_last = channels_last

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.model(x)

    def _unpack(self, batch) -> tuple:
        if isinstance(batch, dict):
            x = batch.get("data", batch.get("image"))
            y = batch.get("label")
        else:
            x, y = batch
        if y.dim() > 1:
            y = y.reshape(-1)
        if x.dtype == torch.uint8 and x.dim() == 4 and x.shape[-1] == 3:
            # Raw NHWC uint8 from the streaming loader: fused HIP
            # normalize (GPU) / torch reference (CPU) — ops.preprocess.
            from ..ops import normalize_images
            x = normalize_images(x)
            if x.device.type == "cpu":
                x = x.to(torch.float32)  # CPU convs don't take bf16 well
        elif self.channels_last and x.dim() == 4 and x.device.type == "cuda":
            x = x.to(memory_format=torch.channels_last)
        return x, y

    def training_step(self, batch, batch_idx: int) -> torch.Tensor:
        x, y = self._unpack(batch)
        loss = F.cross_entropy(self(x), y)
        self.log("train_loss", loss)
        return loss

    def validation_step(self, batch, batch_idx: int) -> Dict[str, Any]:
        x, y = self._unpack(batch)
        logits = self(x)
        loss = F.cross_entropy(logits, y)
        acc = multiclass_accuracy(logits, y)
        self.log("val_loss", loss, sync_dist=True)
        self.log("val_accuracy", acc, sync_dist=True)
        return {"val_loss": loss, "val_accuracy": acc}

    def configure_optimizers(self):
        if next(self.parameters()).is_cuda:
            # capturable: the optimizer step can be recorded into the
            # hipGraph train step (Trainer use_hipgraph / GraphedTrainStep)
            return torch.optim.Adam(self.parameters(), lr=self.lr,
                                    foreach=True, capturable=True)
        return torch.optim.Adam(self.parameters(), lr=self.lr)


class DataModule:
    """LightningDataModule-shaped contract: ``train_dataloader`` /
    ``val_dataloader`` return fresh iterables each call (the reference
    re-enters reader contexts per epoch, ``deep_learning/2...py:261-275``);
    ``teardown`` releases reader threads (``:277-280``)."""

    def setup(self, stage: Optional[str] = None) -> None:
        pass

    def train_dataloader(self):  # pragma: no cover
        raise NotImplementedError

    def val_dataloader(self):
        return None

    def teardown(self, stage: Optional[str] = None) -> None:
        pass
