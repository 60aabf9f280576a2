"""W3 walkthrough — `deep_learning/2.distributed-data-loading-petastorm.py`
as a script: synthetic image parquet → streaming datamodule → Trainer,
with the reference's three launch shapes (``:419-470``):

  1. direct single-process call (1 GPU / CPU),
  2. single-node multi-GPU via TorchDistributor(local_mode=True),
  3. (multi-node is out of scope on one MI355X node).
"""
import os
import sys
import tempfile

sys.path.insert(0, __file__.rsplit("/", 2)[0])

import torch

from mi355x_scale import track
from mi355x_scale.data.generator import write_image_parquet
from mi355x_scale.parallel import TorchDistributor
from mi355x_scale.train import ImageClassifier, ImageStreamDataModule, Trainer

# reference hyperparameters (deep_learning/2...py:342-348)
BATCH_SIZE = 212
MAX_EPOCHS = 2
READER_POOL_TYPE = "thread"
WORKERS_COUNT = 2
RESULTS_QUEUE_SIZE = 20

DATA_DIR = os.path.join(tempfile.gettempdir(), "mi355x_example_images")


def main_training_loop(num_tasks: int, num_proc_per_task: int):
    """Runs on EVERY rank (reference main_training_loop :351-415)."""
    from mi355x_scale.parallel.comm import init_distributed, destroy
    ctx = init_distributed()
    use_cuda = torch.cuda.is_available()
    hw, batch = ((224, 224), BATCH_SIZE) if use_cuda else ((32, 32), 8)
    track.set_experiment("distributed_training")
    run = track.start_run(f"rank{ctx.rank}") if ctx.is_main else None

    dm = ImageStreamDataModule(
        DATA_DIR, batch_size=batch,
        workers_count=WORKERS_COUNT,
        reader_pool_type=READER_POOL_TYPE,
        results_queue_size=RESULTS_QUEUE_SIZE,
        cur_shard=ctx.rank if ctx.world_size > 1 else None,
        shard_count=ctx.world_size if ctx.world_size > 1 else None,
        image_hw=hw, device=ctx.device)
    model = ImageClassifier("resnet50" if use_cuda else "resnet18",
                            num_classes=1000, lr=1e-5)
    steps_per_epoch = max(1, dm.num_rows // (batch * ctx.world_size))
    trainer = Trainer(
        strategy="ddp" if ctx.world_size > 1 else "auto",
        max_epochs=MAX_EPOCHS,
        limit_train_batches=min(steps_per_epoch, 4),
        limit_val_batches=2,
        num_sanity_val_steps=0,
        reload_dataloaders_every_n_epochs=1,
        use_distributed_sampler=False,
        precision="bf16-mixed" if use_cuda else "fp32",
        enable_checkpointing=True,
        default_root_dir=os.path.join(tempfile.gettempdir(),
                                      "mi355x_example_ckpt"),
        logger=run,
    )
    trainer.fit(model, dm)
    best = (trainer.checkpoint_callback.best_model_path
            if trainer.checkpoint_callback else None)
    if run:
        run.end()
    destroy()
    return best


def main():
    hw = (224, 224) if torch.cuda.is_available() else (32, 32)
    if not os.path.exists(os.path.join(DATA_DIR, ".done")):
        write_image_parquet(DATA_DIR, num_rows=BATCH_SIZE * 4 if
                            torch.cuda.is_available() else 128,
                            image_hw=hw, rows_per_group=BATCH_SIZE if
                            torch.cuda.is_available() else 16)
        open(os.path.join(DATA_DIR, ".done"), "w").write("1")

    n_gpu = torch.cuda.device_count()
    if n_gpu <= 1:
        # launch shape 1: direct call (ref :425-428)
        best = main_training_loop(1, 1)
    else:
        # launch shape 2: one process per GPU (ref :440-448)
        best = TorchDistributor(num_processes=n_gpu, local_mode=True,
                                use_gpu=True).run(main_training_loop,
                                                  1, n_gpu)
    print("best checkpoint:", best)


if __name__ == "__main__":
    main()
