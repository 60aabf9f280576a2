"""End-to-end W3 slice on CPU: parquet → reader → ResNet → Trainer."""
import os

import torch

from mi355x_scale import track
from mi355x_scale.train import (CheckpointManager, ImageClassifier,
                                ImageStreamDataModule, Trainer)


def test_trainer_single_process(image_parquet, tmp_path):
    track.set_tracking_root(str(tmp_path / "mlruns"))
    track.set_experiment("test")
    model = ImageClassifier("resnet18", num_classes=10, lr=1e-3)
    dm = ImageStreamDataModule(
        image_parquet, batch_size=16, workers_count=2,
        image_hw=(32, 32), device=torch.device("cpu"))
    with track.start_run("cpu-slice") as run:
        trainer = Trainer(
            strategy="auto", max_epochs=1, limit_train_batches=3,
            limit_val_batches=2, precision="fp32",
            default_root_dir=str(tmp_path / "ckpt"), logger=run,
            enable_checkpointing=True,
        )
        trainer.fit(model, dm)
    # checkpoints written
    assert os.path.exists(tmp_path / "ckpt" / "last.ckpt")
    assert trainer.checkpoint_callback.best_model_path is not None
    # metrics logged in mlruns layout
    run_dir = run.dir
    assert os.path.exists(os.path.join(run_dir, "metrics", "val_loss"))


def test_checkpoint_roundtrip(tmp_path):
    model = ImageClassifier("resnet18", num_classes=10)
    opt = model.configure_optimizers()
    cm = CheckpointManager(str(tmp_path), rank=0)
    path = cm.save(model, opt, epoch=0, step=5, metrics={"val_loss": 1.0})
    model2 = ImageClassifier("resnet18", num_classes=10)
    state = CheckpointManager.load(path, model2)
    assert state["step"] == 5
    for a, b in zip(model.parameters(), model2.parameters()):
        assert torch.equal(a, b)


def test_trainer_resume(image_parquet, tmp_path):
    """Checkpoint → resume: epoch/step restored, training continues."""
    import torch
    model = ImageClassifier("resnet18", num_classes=10, lr=1e-3)
    dm = ImageStreamDataModule(image_parquet, batch_size=16,
                               workers_count=1, image_hw=(32, 32),
                               device=torch.device("cpu"))
    t1 = Trainer(max_epochs=1, limit_train_batches=2, limit_val_batches=1,
                 precision="fp32", default_root_dir=str(tmp_path))
    t1.fit(model, dm)

    model2 = ImageClassifier("resnet18", num_classes=10, lr=1e-3)
    dm2 = ImageStreamDataModule(image_parquet, batch_size=16,
                                workers_count=1, image_hw=(32, 32),
                                device=torch.device("cpu"))
    t2 = Trainer(max_epochs=2, limit_train_batches=2, limit_val_batches=1,
                 precision="fp32", default_root_dir=str(tmp_path / "b"),
                 resume_from=str(tmp_path / "last.ckpt"))
    t2.fit(model2, dm2)
    assert model2.global_step > 2  # continued past the restored step


def test_nested_runs_carry_parent_tag(tmp_path):
    from mi355x_scale import track
    track.set_tracking_root(str(tmp_path / "mlruns"))
    track.set_experiment("nested-test")
    with track.start_run("outer") as outer:
        with track.start_run("inner", nested=True) as inner:
            inner.log_metric("loss", 1.0)
        tag = (tmp_path / "mlruns").rglob("mlflow.parentRunId")
        [tagfile] = list(tag)
        assert tagfile.read_text() == outer.run_id
