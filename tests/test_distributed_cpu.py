"""Multi-process distributed plumbing tests on CPU (gloo, world_size 2) —
the loopback process-group strategy SURVEY.md §4 prescribes."""
import os

import numpy as np
import pytest
import torch

from mi355x_scale.parallel import TorchDistributor, broadcast


def _ddp_worker(data_dir: str):
    """Each rank: init gloo, 3 DDP steps over its shard, return params."""
    import torch.distributed as dist
    from mi355x_scale.parallel.comm import init_distributed, destroy
    from mi355x_scale.train import ImageClassifier, ImageStreamDataModule, Trainer

    ctx = init_distributed(backend="gloo")
    torch.manual_seed(0)
    model = ImageClassifier("resnet18", num_classes=10, lr=1e-3)
    dm = ImageStreamDataModule(
        data_dir, batch_size=8, workers_count=1,
        cur_shard=ctx.rank, shard_count=ctx.world_size,
        image_hw=(32, 32), device=torch.device("cpu"))
    trainer = Trainer(strategy="ddp", max_epochs=1, limit_train_batches=3,
                      precision="fp32", enable_checkpointing=False)
    trainer.fit(model, dm)
    # Params must be identical across ranks after DDP steps.
    flat = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
    gathered = [torch.zeros_like(flat) for _ in range(ctx.world_size)]
    dist.all_gather(gathered, flat)
    same = all(torch.equal(gathered[0], g) for g in gathered)
    destroy()
    return bool(same)


def test_ddp_gloo_two_ranks(image_parquet):
    dist_runner = TorchDistributor(num_processes=2, local_mode=True,
                                   use_gpu=False)
    assert dist_runner.run(_ddp_worker, image_parquet) is True


def _allreduce_worker():
    import torch.distributed as dist
    from mi355x_scale.parallel.comm import init_distributed, destroy
    ctx = init_distributed(backend="gloo")
    t = torch.tensor([float(ctx.rank + 1)])
    dist.all_reduce(t)
    destroy()
    return t.item()


def test_allreduce_gloo():
    out = TorchDistributor(num_processes=2, use_gpu=False).run(_allreduce_worker)
    assert out == 3.0  # 1 + 2


def test_distributor_single_process():
    out = TorchDistributor(num_processes=1).run(lambda a, b: a + b, 2, 3)
    assert out == 5


def test_shm_broadcast_roundtrip():
    data = np.arange(1000, dtype=np.float64)
    bc = broadcast(data)
    assert np.array_equal(bc.value, data)
    # pickling ships only the path (closure-capture efficiency)
    import pickle
    blob = pickle.dumps(bc)
    assert len(blob) < 1000
    bc2 = pickle.loads(blob)
    assert np.array_equal(bc2.value, data)
    bc.unpersist()
    assert not os.path.exists(bc.path)


_fail_marker = None


def _flaky_worker(marker_path):
    import os
    # fail on the first gang attempt, succeed on the second
    if not os.path.exists(marker_path):
        if int(os.environ["RANK"]) == 1:
            open(marker_path, "w").write("1")
            raise RuntimeError("injected rank failure")
        import time
        time.sleep(1.0)
        raise RuntimeError("gang abort")
    return "recovered"


def test_launcher_relaunch_after_rank_failure(tmp_path):
    """rank failure -> gang relaunch (max_restarts)."""
    marker = str(tmp_path / "failed_once")
    out = TorchDistributor(num_processes=2, use_gpu=False,
                           max_restarts=1).run(_flaky_worker, marker)
    assert out == "recovered"


def _always_fail(m):
    raise RuntimeError("always")


def test_launcher_no_restart_raises(tmp_path):
    import pytest as _pytest
    with _pytest.raises(RuntimeError):
        TorchDistributor(num_processes=2, use_gpu=False,
                         max_restarts=1).run(_always_fail,
                                             str(tmp_path / "nope"))


def _allreduce_flat_worker():
    import os

    import torch
    import torch.distributed as dist
    from mi355x_scale.train.graphstep import allreduce_flat

    dist.init_process_group("gloo", rank=int(os.environ["RANK"]),
                            world_size=int(os.environ["WORLD_SIZE"]))
    world = dist.get_world_size()
    rank = dist.get_rank()
    flat = torch.full((1000,), float(rank + 1), dtype=torch.float32)
    allreduce_flat(flat, world)  # world>=4 -> bf16-compressed path
    want = float(world * (world + 1) // 2)  # exactly representable in bf16
    ok = torch.allclose(flat, torch.full_like(flat, want))
    dist.destroy_process_group()
    return bool(ok)


def test_allreduce_flat_bf16_compressed_world4():
    """The bf16-compressed flat all-reduce (active at world>=4 — the
    shape the driver's 4/8-GPU scaling run hits over RCCL) sums
    correctly; gloo shares the compression code path."""
    from mi355x_scale.parallel import TorchDistributor
    res = TorchDistributor(num_processes=4, use_gpu=False).run(
        _allreduce_flat_worker)
    assert res is True
