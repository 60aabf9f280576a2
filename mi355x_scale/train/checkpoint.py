"""Checkpoint / resume.

The reference delegates checkpointing to Lightning
(``enable_checkpointing=True`` + ``default_root_dir``,
``deep_learning/2.distributed-data-loading-petastorm.py:407-409``; best
checkpoint path returned at ``:415``). Here: rank-0 writes torch
``state_dict`` checkpoints, keeps ``last.ckpt`` plus the best-k by a
monitored metric, and ``load``/``resume`` restore model+optimizer+progress.
"""
from __future__ import annotations

import json
import os
from typing import Dict, Optional

import torch


class CheckpointManager:
    def __init__(self, root_dir: str, monitor: str = "val_loss",
                 mode: str = "min", keep_best_k: int = 1, rank: int = 0):
        self.root_dir = root_dir
        self.monitor = monitor
        self.mode = mode
        self.keep_best_k = keep_best_k
        self.rank = rank
        self.best_model_path: Optional[str] = None
        self._best: list = []  # [(score, path)]
        if rank == 0:
            os.makedirs(root_dir, exist_ok=True)

    def _is_better(self, a: float, b: float) -> bool:
        return a < b if self.mode == "min" else a > b

    def save(self, model, optimizer, epoch: int, step: int,
             metrics: Dict[str, float]) -> Optional[str]:
        if self.rank != 0:
            return None
        state = {
            "model": model.state_dict(),
            "optimizer": optimizer.state_dict() if optimizer else None,
            "epoch": epoch,
            "step": step,
            "metrics": metrics,
        }
        last = os.path.join(self.root_dir, "last.ckpt")
        torch.save(state, last)
        score = metrics.get(self.monitor)
        if score is not None:
            path = os.path.join(
                self.root_dir,
                f"epoch={epoch}-step={step}-{self.monitor}={score:.4f}.ckpt")
            torch.save(state, path)
            self._best.append((score, path))
            self._best.sort(key=lambda t: t[0],
                            reverse=(self.mode == "max"))
            while len(self._best) > self.keep_best_k:
                _, drop = self._best.pop()
                if os.path.exists(drop):
                    os.unlink(drop)
            self.best_model_path = self._best[0][1]
            with open(os.path.join(self.root_dir, "best.json"), "w") as f:
                json.dump({"best_model_path": self.best_model_path,
                           self.monitor: self._best[0][0]}, f)
        return last

    @staticmethod
    def load(path: str, model, optimizer=None, map_location="cpu") -> Dict:
        state = torch.load(path, map_location=map_location,
                           weights_only=False)
        model.load_state_dict(state["model"])
        if optimizer is not None:
            if state.get("optimizer"):
                optimizer.load_state_dict(state["optimizer"])
            elif hasattr(optimizer, "sync_master_from_params"):
                # weights-only checkpoint: re-seed the optimizer's fp32
                # master from the freshly loaded (bf16) param views, or
                # the first step() reverts the model to its init values
                optimizer.sync_master_from_params()
        return state
