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

    def _cross_rank_diff(t: torch.Tensor) -> float:
        v = t.detach().reshape(-1).cpu().float()
        out = [torch.zeros_like(v) for _ in range(2)]
        dist.all_gather(out, v)
        return (out[0] - out[1]).abs().max().item()

    gs = GraphedTrainStep(model, opt, batch, world_size=2, warmup=1)
    # params must be identical after ctor (seeded init + warmup applied
    # the same all-reduced gradient on both ranks)
    p0 = torch.cat([p.detach().reshape(-1).cpu().float()
                    for p in model.parameters()])
    diag = {"after_ctor": _cross_rank_diff(p0)}
    for i in range(3):
        # manual step with forensics between the phases
        for k, v in batch.items():
            gs.static_batch[k].copy_(v, non_blocking=True)
        gs.g_fwd_bwd.replay()
        torch.cuda.synchronize()
        local_nan = int(gs.grad_buffers[0].isnan().sum().item())
        both = [None, None]
        dist.all_gather_object(both, local_nan)
        diag[f"local_nan_{i}"] = tuple(both)
        diag[f"loss_{i}"] = round(float(gs.static_loss.item()), 4)
        gs._allreduce_grads()
        diag[f"grads_{i}"] = _cross_rank_diff(gs.grad_buffers[0])
        if gs.g_opt is not None:
            gs.g_opt.replay()
        torch.cuda.synchronize()
        pp = torch.cat([p.detach().reshape(-1).cpu().float()
                        for p in model.parameters()])
        diag[f"params_{i}"] = _cross_rank_diff(pp)
    dist.destroy_process_group()
    return diag


@pytest.mark.gpu
def test_graph_step_two_ranks_one_gpu():
    diag = TorchDistributor(num_processes=2, use_gpu=True).run(_worker)
    # both ranks apply the identical all-reduced gradient every step, so
    # parameters stay in lockstep
    assert diag["after_ctor"] == 0.0, diag
    for i in range(3):
        assert diag[f"local_nan_{i}"] == (0, 0), diag
        assert diag[f"grads_{i}"] == 0.0, diag
        assert diag[f"params_{i}"] == 0.0, diag
