"""Graph-replay + eager collective interleave, 2 ranks on one GPU (gloo
backend moves the flat-grad all-reduce through host memory, so two ranks
can share cuda:0 — validates the multi-rank graph-mode mechanics the
driver exercises with RCCL at N=2/4/8)."""
import pytest
import torch

from mi355x_scale.parallel import TorchDistributor


def _worker():
    import os
    import torch
    import torch.distributed as dist
    from mi355x_scale.train import ImageClassifier
    from mi355x_scale.train.graphstep import GraphedTrainStep

    dist.init_process_group("gloo", rank=int(os.environ["RANK"]),
                            world_size=int(os.environ["WORLD_SIZE"]))
    dev = torch.device("cuda:0")
    torch.manual_seed(7)  # identical init on both ranks
    model = ImageClassifier("resnet18", num_classes=10, lr=1e-3).to(dev)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3, foreach=True,
                           capturable=True)
    g = torch.Generator().manual_seed(100 + dist.get_rank())  # per-rank data
    batch = {
        "image": torch.randint(0, 256, (4, 64, 64, 3), dtype=torch.uint8,
                               generator=g).to(dev),
        "label": torch.randint(0, 10, (4,), generator=g).to(dev),
    }

    gs = GraphedTrainStep(model, opt, batch, world_size=2, warmup=1)
    for _ in range(3):
        gs.step(batch)
    torch.cuda.synchronize()
    flat = torch.cat([p.detach().reshape(-1).cpu()
                      for p in model.parameters()])
    out = [torch.zeros_like(flat) for _ in range(2)]
    dist.all_gather(out, flat)
    same = torch.equal(out[0], out[1])
    dist.destroy_process_group()
    return bool(same)


@pytest.mark.gpu
def test_graph_step_two_ranks_one_gpu():
    ok = TorchDistributor(num_processes=2, use_gpu=True).run(_worker)
    assert ok is True
