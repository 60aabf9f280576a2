"""hipGraph-captured step: numerical parity with the eager step."""
import copy

import pytest
import torch

from mi355x_scale.train import ImageClassifier


@pytest.mark.gpu
def test_graphed_gradients_match_eager():
    """One fwd/bwd: captured-graph gradients must match eager gradients
    (direct comparison — Adam's sign-normalized updates would amplify
    benign bf16 noise into lr-scale parameter differences)."""
    from mi355x_scale.train.graphstep import GraphedTrainStep

    torch.manual_seed(0)
    dev = torch.device("cuda:0")
    model_e = ImageClassifier("resnet18", num_classes=10, lr=1e-3).to(dev)
    model_g = copy.deepcopy(model_e)

    g = torch.Generator().manual_seed(42)
    batch = {
        "image": torch.randint(0, 256, (8, 64, 64, 3),
                               dtype=torch.uint8, generator=g).to(dev),
        "label": torch.randint(0, 10, (8,), generator=g).to(dev),
    }

    # eager gradients
    with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
        loss_e = model_e.training_step(batch, 0)
    loss_e.backward()
    eager_flat = torch.cat([p.grad.reshape(-1)
                            for p in model_e.parameters()
                            if p.grad is not None])

    # graphed gradients (restore weights + grads after warmup, replay once)
    state0 = copy.deepcopy(model_g.state_dict())
    opt_g = torch.optim.Adam(model_g.parameters(), lr=0.0, foreach=True,
                             capturable=True)  # lr=0: warmup can't move w
    gs = GraphedTrainStep(model_g, opt_g, batch, world_size=1, warmup=2)
    model_g.load_state_dict(state0)
    gs.step(batch)
    torch.cuda.synchronize()
    graph_flat = torch.cat([p.grad.reshape(-1)
                            for p in model_g.parameters()
                            if p.grad is not None])

    cos = torch.nn.functional.cosine_similarity(
        eager_flat.double(), graph_flat.double(), dim=0).item()
    rel = ((eager_flat - graph_flat).norm() /
           (eager_flat.norm() + 1e-12)).item()
    assert cos > 0.999, f"gradient cosine {cos}"
    assert rel < 2e-2, f"gradient relative L2 {rel}"
    assert abs(loss_e.item() - gs.static_loss.item()) < 5e-2


@pytest.mark.gpu
def test_graphed_step_loss_finite_and_changing():
    from mi355x_scale.train.graphstep import GraphedTrainStep
    torch.manual_seed(0)
    dev = torch.device("cuda:0")
    model = ImageClassifier("resnet18", num_classes=10, lr=1e-2).to(dev)
    opt = torch.optim.Adam(model.parameters(), lr=1e-2, foreach=True, capturable=True)
    b = {"image": torch.randint(0, 256, (8, 64, 64, 3), dtype=torch.uint8,
                                device=dev),
         "label": torch.randint(0, 10, (8,), device=dev)}
    gs = GraphedTrainStep(model, opt, b, world_size=1, warmup=2)
    losses = []
    for _ in range(5):
        loss = gs.step(b)
        torch.cuda.synchronize()
        losses.append(loss.item())
    assert all(torch.isfinite(torch.tensor(losses)))
    assert losses[-1] < losses[0]  # optimizing the same batch must descend


@pytest.mark.gpu
def test_trainer_hipgraph_mode(tmp_path):
    """Trainer with use_hipgraph='auto' on GPU: trains, checkpoints."""
    import os
    from mi355x_scale.data.generator import write_image_parquet
    from mi355x_scale.train import ImageStreamDataModule, Trainer
    d = str(tmp_path / "imgs")
    write_image_parquet(d, num_rows=64, image_hw=(64, 64), num_classes=10,
                        rows_per_group=16, rows_per_file=32)
    model = ImageClassifier("resnet18", num_classes=10, lr=1e-3)
    dm = ImageStreamDataModule(d, batch_size=16, workers_count=2,
                               image_hw=(64, 64))
    trainer = Trainer(max_epochs=1, limit_train_batches=4,
                      limit_val_batches=2,
                      default_root_dir=str(tmp_path / "ckpt"))
    trainer.fit(model, dm)
    assert os.path.exists(tmp_path / "ckpt" / "last.ckpt")


@pytest.mark.gpu
def test_trainer_graph_mode_resume(tmp_path):
    """Graph-mode checkpoints (FlatAdam flat-buffer format) must resume:
    a second Trainer picks up epoch/step and keeps training."""
    import os
    from mi355x_scale.data.generator import write_image_parquet
    from mi355x_scale.train import ImageStreamDataModule, Trainer
    d = str(tmp_path / "imgs")
    write_image_parquet(d, num_rows=64, image_hw=(64, 64), num_classes=10,
                        rows_per_group=16, rows_per_file=32)
    root = str(tmp_path / "ckpt")
    model = ImageClassifier("resnet18", num_classes=10, lr=1e-3)
    dm = ImageStreamDataModule(d, batch_size=16, workers_count=2,
                               image_hw=(64, 64))
    Trainer(max_epochs=1, limit_train_batches=3, limit_val_batches=1,
            default_root_dir=root).fit(model, dm)
    assert os.path.exists(os.path.join(root, "last.ckpt"))

    model2 = ImageClassifier("resnet18", num_classes=10, lr=1e-3)
    dm2 = ImageStreamDataModule(d, batch_size=16, workers_count=2,
                                image_hw=(64, 64))
    t2 = Trainer(max_epochs=2, limit_train_batches=3, limit_val_batches=1,
                 default_root_dir=root,
                 resume_from=os.path.join(root, "last.ckpt"))
    out = t2.fit(model2, dm2)
    assert out.global_step >= 6  # continued past the first run's steps
