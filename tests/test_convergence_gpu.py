"""End-to-end learning sanity on the full fused stack.

Pointwise kernel numerics can pass while training is still broken
(stale running stats, mis-scaled grads, wrong ReLU masks). This trains
the flagship path — fused BN/maxpool/preprocess kernels, hipGraph step,
FlatAdam — on a fixed batch and requires the loss to collapse
(memorization), plus a sanity run of the eager (non-graph) path.
"""
import pytest
import torch

from mi355x_scale.models import resnet18
from mi355x_scale.train import ImageClassifier
from mi355x_scale.train.flat_adam import FlatAdam


def _fixed_batch(dev, n=32, classes=10):
    g = torch.Generator().manual_seed(0)
    return {
        "image": torch.randint(0, 256, (n, 64, 64, 3), dtype=torch.uint8,
                               generator=g).to(dev),
        "label": torch.randint(0, classes, (n,), generator=g).to(dev),
    }


@pytest.mark.gpu
def test_graph_mode_memorizes_fixed_batch():
    from mi355x_scale.train.graphstep import GraphedTrainStep
    dev = torch.device("cuda:0")
    torch.manual_seed(7)
    model = ImageClassifier("resnet18", num_classes=10, lr=1e-3).to(dev)
    model.to(memory_format=torch.channels_last)
    model.trainer = None
    model.log = lambda *a, **k: None
    batch = _fixed_batch(dev)
    opt = FlatAdam(model.parameters(), lr=1e-3)
    graphed = GraphedTrainStep(model, opt, batch, world_size=1, warmup=3)
    first = None
    for i in range(60):
        loss = graphed.step(batch)
        if i == 0:
            first = loss.item()
    torch.cuda.synchronize()
    last = loss.item()
    assert first > 0.5, f"first loss suspiciously low: {first}"
    assert last < 0.35 * first, (
        f"graph-mode training did not learn: {first:.3f} -> {last:.3f}")


@pytest.mark.gpu
def test_eager_mode_memorizes_fixed_batch():
    dev = torch.device("cuda:0")
    torch.manual_seed(7)
    model = ImageClassifier("resnet18", num_classes=10, lr=1e-3).to(dev)
    model.to(memory_format=torch.channels_last)
    model.trainer = None
    model.log = lambda *a, **k: None
    batch = _fixed_batch(dev)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    losses = []
    for _ in range(40):
        with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
            loss = model.training_step(batch, 0)
        opt.zero_grad(set_to_none=True)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < 0.35 * losses[0], (
        f"eager training did not learn: {losses[0]:.3f} -> {losses[-1]:.3f}")
