"""FlatAdam — Adam over flat fp32 buffers, one fused HIP kernel per step.

The reference's optimizer is ``torch.optim.Adam`` (``deep_learning/
2.distributed-data-loading-petastorm.py:158-161``). PyTorch's capturable
foreach Adam replays ~70 small elementwise kernels per ResNet-18 step
(~1.2 ms on MI355X); FlatAdam lays parameters, gradients and both
moments out as single flat fp32 buffers and steps them with ONE
memory-bound kernel (``ops/csrc/adam.hip``, ~45 us at the HBM3E
roofline).

Construction re-points every parameter's storage into the flat param
buffer (keeping each param's own dense stride order, e.g. channels-last
conv weights) and installs gradient views into the flat grad buffer —
which is therefore also the single-call RCCL all-reduce target for the
graphed DDP step (``train/graphstep.py``).

On CPU (no HIP extension) the same math runs as vectorized torch ops on
the flat buffers — that path is the numerics oracle in the tests.
"""
from __future__ import annotations

import math
from typing import Iterable, Tuple

import torch


class FlatAdam:
    def __init__(self, params: Iterable[torch.nn.Parameter],
                 lr: float = 1e-3, betas: Tuple[float, float] = (0.9, 0.999),
                 eps: float = 1e-8, weight_decay: float = 0.0):
        self.params = [p for p in params if p.requires_grad]
        assert self.params, "no trainable parameters"
        device = self.params[0].device
        dtype = self.params[0].dtype
        assert dtype == torch.float32, "FlatAdam keeps fp32 master params"
        self.lr, self.betas, self.eps = lr, betas, eps
        self.weight_decay = weight_decay
        total = sum(p.numel() for p in self.params)

        self.flat_params = torch.empty(total, dtype=torch.float32,
                                       device=device)
        import os
        if (device.type == "cuda"
                and os.environ.get("MI355X_P2P_ALLREDUCE") == "1"):
            # IPC-shareable grad buffer so the hand-written xGMI p2p
            # all-reduce (parallel/p2p_allreduce.py) can map it from
            # peer ranks; hipIpc handles need the hipMalloc base pointer.
            from ..parallel.p2p_allreduce import alloc_shared
            self.flat_grads = alloc_shared(total, device)
            self.flat_grads.zero_()
        else:
            self.flat_grads = torch.zeros(total, dtype=torch.float32,
                                          device=device)
        off = 0
        for p in self.params:
            n = p.numel()
            # param data moves INTO the flat buffer, keeping the param's
            # own dense stride order (channels_last conv weights stay
            # channels_last; backward then accumulates with no layout
            # permute — the "gradient layout contract")
            pview = self.flat_params[off:off + n].as_strided(
                p.shape, p.stride())
            pview.copy_(p.data)
            p.data = pview
            p.grad = self.flat_grads[off:off + n].as_strided(
                p.shape, p.stride())
            off += n
        self.exp_avg = torch.zeros_like(self.flat_params)
        self.exp_avg_sq = torch.zeros_like(self.flat_params)
        self.step_t = torch.zeros(1, dtype=torch.int32, device=device)
        # graphstep's capturable check reads param_groups
        self.param_groups = [{
            "params": self.params, "lr": lr, "betas": betas, "eps": eps,
            "weight_decay": weight_decay, "capturable": True,
        }]

    @torch.no_grad()
    def step(self) -> None:
        if self.flat_params.is_cuda:
            from ..ops import _C, require_ext
            require_ext()
            _C.adam_step(self.flat_params, self.flat_grads, self.exp_avg,
                         self.exp_avg_sq, self.step_t, self.lr,
                         self.betas[0], self.betas[1], self.eps,
                         self.weight_decay)
            return
        # CPU reference (numerics oracle): identical math, torch ops
        self.step_t += 1
        t = int(self.step_t.item())
        b1, b2 = self.betas
        g = self.flat_grads
        if self.weight_decay:
            g = g + self.weight_decay * self.flat_params
        self.exp_avg.mul_(b1).add_(g, alpha=1 - b1)
        self.exp_avg_sq.mul_(b2).addcmul_(g, g, value=1 - b2)
        bc1 = 1 - b1 ** t
        bc2 = 1 - b2 ** t
        denom = (self.exp_avg_sq / bc2).sqrt_().add_(self.eps)
        self.flat_params.addcdiv_(self.exp_avg, denom, value=-self.lr / bc1)

    def zero_grad(self, set_to_none: bool = False) -> None:
        # grads are views of one buffer; zeroing the buffer is the only
        # correct form (set_to_none would detach the views)
        self.flat_grads.zero_()

    def state_dict(self) -> dict:
        return {
            "flat_params": self.flat_params, "exp_avg": self.exp_avg,
            "exp_avg_sq": self.exp_avg_sq, "step": self.step_t,
            "lr": self.lr, "betas": self.betas, "eps": self.eps,
            "weight_decay": self.weight_decay,
        }

    def load_state_dict(self, sd: dict) -> None:
        if "flat_params" not in sd:
            raise ValueError(
                "checkpoint holds a torch.optim state dict, not FlatAdam's "
                "flat-buffer format — resume graph-mode runs with "
                "graph-mode checkpoints (or load only the model weights)")
        self.flat_params.copy_(sd["flat_params"])
        self.exp_avg.copy_(sd["exp_avg"])
        self.exp_avg_sq.copy_(sd["exp_avg_sq"])
        self.step_t.copy_(sd["step"])
