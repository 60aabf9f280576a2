"""Graph-replay + eager collective interleave, 2 ranks on one GPU (gloo
backend moves the flat-grad all-reduce through host memory, so two ranks
can share cuda:0 — validates the multi-rank graph-mode mechanics the
driver exercises with RCCL at N=2/4/8: capture, flat bf16/fp32 grad
buffers, the all-reduce strategy ladder, FlatAdam replay, lockstep).

The model is a pure-GEMM MLP on purpose: MIOpen *conv* kernels have a
known co-residency artifact when two processes replay captured graphs on
the SAME physical GPU (deterministic NaN weight-grads from replay 1 —
finite losses, asymmetric per-rank counts under cold concurrent find;
see NOTES_NEXT.md). One-rank-per-GPU — the only production topology —
is unaffected: the 60-replay single-process convergence test covers the
full conv/fused-BN path on the same boxes.
"""
import pytest
import torch

from mi355x_scale.parallel import TorchDistributor


def _worker():
    import os

    import torch
    import torch.distributed as dist
    import torch.nn as nn
    from mi355x_scale.train.flat_adam import FlatAdam
    from mi355x_scale.train.graphstep import GraphedTrainStep

    dist.init_process_group("gloo", rank=int(os.environ["RANK"]),
                            world_size=int(os.environ["WORLD_SIZE"]))
    dev = torch.device("cuda:0")
    torch.manual_seed(7)  # identical init on both ranks

    class MLP(nn.Module):
        def __init__(self):
            super().__init__()
            self.net = nn.Sequential(nn.Linear(64, 256), nn.ReLU(),
                                     nn.Linear(256, 256), nn.ReLU(),
                                     nn.Linear(256, 10))

        def training_step(self, batch, _idx):
            return nn.functional.cross_entropy(self.net(batch["x"]),
                                               batch["y"])

    model = MLP().to(dev)
    # the shipped optimizer: flat bf16 working params + fp32 master
    opt = FlatAdam(model.parameters(), lr=1e-3, bf16_params=True)
    g = torch.Generator().manual_seed(100 + dist.get_rank())  # per-rank data
    batch = {
        "x": torch.randn(32, 64, generator=g).to(dev),
        "y": torch.randint(0, 10, (32,), generator=g).to(dev),
    }

    def _cross_rank_diff(t: torch.Tensor) -> float:
        v = t.detach().reshape(-1).cpu().float()
        out = [torch.zeros_like(v) for _ in range(2)]
        dist.all_gather(out, v)
        return (out[0] - out[1]).abs().max().item()

    gs = GraphedTrainStep(model, opt, batch, world_size=2, warmup=2)
    diag = {"after_ctor": _cross_rank_diff(opt.flat_master)}
    for i in range(4):
        # manual step with forensics between the phases
        for k, v in batch.items():
            gs.static_batch[k].copy_(v, non_blocking=True)
        gs.g_fwd_bwd.replay()
        torch.cuda.synchronize()
        diag[f"local_nan_{i}"] = sum(
            int(b.isnan().sum().item()) for b in gs.grad_buffers)
        diag[f"loss_{i}"] = round(float(gs.static_loss.item()), 4)
        gs._allreduce_grads()
        diag[f"grads_{i}"] = _cross_rank_diff(gs.grad_buffers[0])
        if gs.g_opt is not None:
            gs.g_opt.replay()
        torch.cuda.synchronize()
        diag[f"params_{i}"] = _cross_rank_diff(opt.flat_master)
    dist.destroy_process_group()
    return diag


@pytest.mark.gpu
def test_graph_step_two_ranks_one_gpu():
    diag = TorchDistributor(num_processes=2, use_gpu=True).run(_worker)
    # both ranks apply the identical all-reduced gradient every step, so
    # master parameters stay in lockstep (bf16 grads sum identically)
    assert diag["after_ctor"] == 0.0, diag
    for i in range(4):
        assert diag[f"local_nan_{i}"] == 0, diag
        assert diag[f"grads_{i}"] == 0.0, diag
        assert diag[f"params_{i}"] == 0.0, diag


def _overlap_worker():
    import os

    import torch
    import torch.distributed as dist
    import torch.nn as nn
    from mi355x_scale.train.flat_adam import FlatAdam
    from mi355x_scale.train.graphstep import GraphedTrainStep

    dist.init_process_group("gloo", rank=int(os.environ["RANK"]),
                            world_size=int(os.environ["WORLD_SIZE"]))
    dev = torch.device("cuda:0")
    torch.manual_seed(7)

    class MLP(nn.Module):
        # split-backward overlap cut: backward of net.4 (the tail of the
        # flat buffers) completes first; its all-reduce runs on the comm
        # stream while the early-layer backward graph replays
        comm_overlap_boundary = "net.2"

        def __init__(self):
            super().__init__()
            self.net = nn.Sequential(nn.Linear(64, 256), nn.ReLU(),
                                     nn.Linear(256, 256), nn.ReLU(),
                                     nn.Linear(256, 10))

        def training_step(self, batch, _idx):
            return nn.functional.cross_entropy(self.net(batch["x"]),
                                               batch["y"])

    model = MLP().to(dev)
    opt = FlatAdam(model.parameters(), lr=1e-3, bf16_params=True)
    g = torch.Generator().manual_seed(100 + dist.get_rank())
    batch = {
        "x": torch.randn(32, 64, generator=g).to(dev),
        "y": torch.randint(0, 10, (32,), generator=g).to(dev),
    }
    gs = GraphedTrainStep(model, opt, batch, world_size=2, warmup=2)
    engaged = gs.g_bwd2 is not None
    losses = []
    for _ in range(4):
        loss = gs.step(batch)
        torch.cuda.synchronize()
        losses.append(float(loss.item()))
    v = opt.flat_master.detach().cpu()
    out = [torch.zeros_like(v) for _ in range(2)]
    dist.all_gather(out, v)
    diff = (out[0] - out[1]).abs().max().item()
    dist.destroy_process_group()
    return {"engaged": engaged, "losses": losses, "param_diff": diff}


@pytest.mark.gpu
def test_graph_overlap_two_ranks_one_gpu():
    """The split-backward comm-overlap path at world size 2 (the shape
    the driver's multi-GPU run takes): overlap must engage, losses stay
    finite, and both ranks' masters stay in lockstep."""
    diag = TorchDistributor(num_processes=2, use_gpu=True).run(
        _overlap_worker)
    assert diag["engaged"], diag
    assert all(torch.isfinite(torch.tensor(diag["losses"]))), diag
    assert diag["param_diff"] == 0.0, diag
