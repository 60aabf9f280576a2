"""Round-2 fix coverage: VERDICT.md weak items 3-5 and ADVICE.md findings."""
import os
import subprocess
import sys

import numpy as np
import pytest
import torch

from mi355x_scale import track
from mi355x_scale.train import ImageClassifier
from mi355x_scale.train.checkpoint import CheckpointManager
from mi355x_scale.train.flat_adam import FlatAdam


def test_experiment_id_deterministic_across_processes(tmp_path):
    """The experiment id must not depend on the per-process hash salt
    (VERDICT weak #3): every process maps the same name to the same
    mlruns/<id>/ directory."""
    code = (
        "from mi355x_scale import track;"
        "track.set_tracking_root(%r);"
        "track.set_experiment('exp-determinism');"
        "print(track._active_experiment['id'])" % str(tmp_path)
    )
    ids = {
        subprocess.run([sys.executable, "-c", code], capture_output=True,
                       text=True, check=True,
                       cwd=os.path.dirname(os.path.dirname(
                           os.path.abspath(__file__)))).stdout.strip()
        for _ in range(2)
    }
    assert len(ids) == 1
    track.set_tracking_root(str(tmp_path))
    track.set_experiment("exp-determinism")
    assert track._active_experiment["id"] == next(iter(ids))


def test_generate_bom_per_sku_dags():
    from mi355x_scale.data.generator import generate_bom
    bom, mapper = generate_bom([f"SKU{i}" for i in range(5)], levels=2)
    assert set(mapper["sku"]) == {f"SKU{i}" for i in range(5)}
    # every SKU's root appears as a material_out and the edge count is
    # bounded by the children range (2..4 per node, 2 levels)
    for root in mapper["final_mat_number"]:
        sub = bom[bom["material_out"] == root]
        assert 2 <= len(sub) <= 4
    assert (bom["qty"] >= 1).all() and (bom["qty"] <= 4).all()


def test_weights_only_resume_syncs_master(tmp_path):
    """ADVICE medium #1: a weights-only checkpoint loaded into a model
    whose FlatAdam(bf16_params) was already constructed must re-seed the
    fp32 master — otherwise the first step() reverts to random init."""
    torch.manual_seed(0)
    src = ImageClassifier("resnet18", num_classes=10, channels_last=False)
    ckpt = str(tmp_path / "w.ckpt")
    torch.save({"model": src.state_dict(), "optimizer": None,
                "epoch": 0, "step": 3, "metrics": {}}, ckpt)

    torch.manual_seed(123)  # different init
    dst = ImageClassifier("resnet18", num_classes=10, channels_last=False)
    opt = FlatAdam(dst.parameters(), bf16_params=True)
    master_before = opt.flat_master.clone()
    CheckpointManager.load(ckpt, dst, opt)
    assert not torch.equal(opt.flat_master, master_before)
    # master's bf16 segment must now equal the loaded weights
    src_bf16 = torch.cat([
        p.detach().reshape(-1) for p in src.parameters() if p.dim() >= 2])
    got = opt.flat_master[:opt.n_bf16]
    # compare at bf16 resolution (the checkpoint stored bf16 views of src? no — src is fp32)
    assert torch.allclose(got, src_bf16, atol=1e-2, rtol=1e-2)
    # and step() must not revert toward the old random master
    for p in dst.parameters():
        p.grad.zero_()
    opt.step()
    assert torch.allclose(opt.flat_master[:opt.n_bf16], src_bf16,
                          atol=1e-2, rtol=1e-2)


def test_fmin_choice_returns_index_and_space_eval():
    """ADVICE low: hyperopt returns option INDICES for hp.choice;
    reference-style options[best[key]] / space_eval must work."""
    from mi355x_scale.tune import fmin, hp, space_eval, tpe

    options = ["rbf", "poly", "linear"]
    space = {"kernel": hp.choice("kernel", options),
             "c": hp.uniform("c", 0.1, 10.0)}

    def obj(p):
        assert p["kernel"] in options  # objective sees the VALUE
        return (p["c"] - 2.0) ** 2 + (0.0 if p["kernel"] == "poly" else 1.0)

    best = fmin(obj, space, algo=tpe.suggest, max_evals=30,
                rstate=np.random.default_rng(123))
    assert isinstance(best["kernel"], int) and 0 <= best["kernel"] < 3
    assert options[best["kernel"]] == "poly"
    evald = space_eval(space, best)
    assert evald["kernel"] == "poly"
    assert abs(evald["c"] - best["c"]) < 1e-12


def _buffer_broadcast_worker():
    import torch
    import torch.distributed as dist
    from mi355x_scale.train import ImageClassifier
    from mi355x_scale.train.trainer import Trainer

    dist.init_process_group("gloo", rank=int(os.environ["RANK"]),
                            world_size=int(os.environ["WORLD_SIZE"]))
    rank = dist.get_rank()
    torch.manual_seed(rank)  # per-rank divergent init AND buffers
    model = ImageClassifier("resnet18", num_classes=10,
                            channels_last=False)
    with torch.no_grad():
        for b in model.buffers():
            if b.is_floating_point():
                b.add_(float(rank))  # force running-stat drift
    t = Trainer(enable_checkpointing=False)

    class _Ctx:  # _broadcast_buffers only touches dist, not ctx
        pass
    t.ctx = _Ctx()
    t._broadcast_buffers(model)
    flat = torch.cat([b.detach().float().reshape(-1)
                      for b in model.buffers()])
    gathered = [torch.zeros_like(flat) for _ in range(2)]
    dist.all_gather(gathered, flat)
    ok = torch.equal(gathered[0], gathered[1])
    dist.destroy_process_group()
    return bool(ok)


def test_bn_buffer_broadcast_two_ranks():
    """VERDICT next-round #7: rank-0 buffer broadcast makes BN running
    stats identical across ranks before eval."""
    from mi355x_scale.parallel import TorchDistributor
    assert TorchDistributor(num_processes=2, use_gpu=False).run(
        _buffer_broadcast_worker) is True
