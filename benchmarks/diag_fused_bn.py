"""Diagnostic: which (if any) BN dispatches take the torch fallback on
the bench-shaped path (bs 212, 224x224, bf16 autocast, graph mode)?

Prints FusedBNReLU2d.gpu_fallbacks after (a) one eager training_step and
(b) GraphedTrainStep construction + one replay.
"""
import sys

import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from mi355x_scale.ops.fused_bn import FusedBNReLU2d  # noqa: E402
from mi355x_scale.train import ImageClassifier  # noqa: E402
from mi355x_scale.train.graphstep import GraphedTrainStep  # noqa: E402


def main():
    assert torch.cuda.is_available()
    dev = torch.device("cuda:0")
    torch.backends.cudnn.benchmark = True
    model = ImageClassifier("resnet18", num_classes=1000, lr=1e-5).to(dev)
    model.to(memory_format=torch.channels_last)
    model.trainer = None
    model.log = lambda *a, **k: None
    batch = {
        "image": torch.randint(0, 256, (212, 224, 224, 3),
                               dtype=torch.uint8, device=dev),
        "label": torch.randint(0, 1000, (212,), device=dev),
    }

    FusedBNReLU2d.gpu_fallbacks.clear()
    with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
        loss = model.training_step(batch, 0)
    loss.backward()
    torch.cuda.synchronize()
    print(f"eager exec fallbacks: {len(FusedBNReLU2d.gpu_fallbacks)}")
    for f in FusedBNReLU2d.gpu_fallbacks[:25]:
        print("  ", f)
    # any leftover nn.BatchNorm2d is a silent MIOpen path
    import torch.nn as nn
    leftovers = [n for n, m in model.named_modules()
                 if isinstance(m, nn.BatchNorm2d)]
    print(f"plain nn.BatchNorm2d modules: {leftovers}")

    # drop the eager autograd graph before capturing (a live
    # AccumulateGrad from a prior iteration breaks graph capture)
    del loss
    FusedBNReLU2d.gpu_fallbacks.clear()
    for p in model.parameters():
        p.grad = None
    opt = torch.optim.Adam(model.parameters(), lr=1e-5, foreach=True,
                           capturable=True)
    g = GraphedTrainStep(model, opt, batch, world_size=1, warmup=3)
    print(f"graphstep build fallbacks: {len(FusedBNReLU2d.gpu_fallbacks)}")
    for f in FusedBNReLU2d.gpu_fallbacks[:25]:
        print("  ", f)
    FusedBNReLU2d.gpu_fallbacks.clear()
    g.step(batch)
    torch.cuda.synchronize()
    print(f"replay fallbacks: {len(FusedBNReLU2d.gpu_fallbacks)}")


if __name__ == "__main__":
    main()
