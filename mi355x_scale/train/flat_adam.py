"""FlatAdam — Adam over flat fp32 buffers, one fused HIP kernel per step.

The reference's optimizer is ``torch.optim.Adam`` (``deep_learning/
2.distributed-data-loading-petastorm.py:158-161``). PyTorch's capturable
foreach Adam replays ~70 small elementwise kernels per ResNet-18 step
(~1.2 ms on MI355X); FlatAdam lays parameters, gradients and both
moments out as single flat fp32 buffers and steps them with ONE
memory-bound kernel (``ops/csrc/adam.hip``, ~45 us at the HBM3E
roofline).

Construction re-points every parameter's storage into the flat buffers
(keeping each param's own dense stride order, e.g. channels-last conv
weights) and installs gradient views into the flat grad buffer(s) —
which are therefore also the single-call RCCL all-reduce targets for
the graphed DDP step (``train/graphstep.py``).

``bf16_params=True`` (GPU): matrix-shaped parameters (conv/linear
weights — everything with dim >= 2) live as bf16 working copies backed
by the fp32 master, exactly the rounding autocast's per-step weight
cast performed — but the ~40 cast kernels per replay disappear, their
gradients arrive and all-reduce natively in bf16 (half the xGMI bytes),
and the fused Adam kernel rewrites the bf16 copies from the updated
master in the same pass. 1-D parameters (BN scale/bias) stay fp32.

On CPU (no HIP extension) the same math runs as vectorized torch ops on
the flat buffers — that path is the numerics oracle in the tests.
"""
from __future__ import annotations

from typing import Iterable, List, Tuple

import torch


class FlatAdam:
    def __init__(self, params: Iterable[torch.nn.Parameter],
                 lr: float = 1e-3, betas: Tuple[float, float] = (0.9, 0.999),
                 eps: float = 1e-8, weight_decay: float = 0.0,
                 bf16_params: bool = False):
        all_params = [p for p in params if p.requires_grad]
        assert all_params, "no trainable parameters"
        device = all_params[0].device
        assert all(p.dtype == torch.float32 for p in all_params), \
            "FlatAdam expects fp32 parameters at construction"
        self.lr, self.betas, self.eps = lr, betas, eps
        self.weight_decay = weight_decay
        self.bf16_params = bool(bf16_params)

        bf16_set: List[torch.nn.Parameter] = []
        fp32_set: List[torch.nn.Parameter] = list(all_params)
        if self.bf16_params:
            bf16_set = [p for p in all_params if p.dim() >= 2]
            fp32_set = [p for p in all_params if p.dim() < 2]
        self.params = bf16_set + fp32_set  # master layout order
        self.n_bf16 = sum(p.numel() for p in bf16_set)
        total = sum(p.numel() for p in self.params)

        self.flat_master = torch.empty(total, dtype=torch.float32,
                                       device=device)
        import os
        if (device.type == "cuda"
                and os.environ.get("MI355X_P2P_ALLREDUCE") == "1"):
            # IPC-shareable grad buffer so the hand-written xGMI p2p
            # all-reduce (parallel/p2p_allreduce.py) can map it from
            # peer ranks; hipIpc handles need the hipMalloc base pointer.
            from ..parallel.p2p_allreduce import alloc_shared
            grads_f32 = alloc_shared(total - self.n_bf16, device)
            grads_f32.zero_()
        else:
            grads_f32 = torch.zeros(total - self.n_bf16,
                                    dtype=torch.float32, device=device)
        self.flat_grads = grads_f32  # fp32 segment (all params if not bf16)
        if self.n_bf16:
            self.flat_pb16 = torch.empty(self.n_bf16, dtype=torch.bfloat16,
                                         device=device)
            self.flat_gb16 = torch.zeros(self.n_bf16, dtype=torch.bfloat16,
                                         device=device)
        else:
            self.flat_pb16 = torch.empty(0, dtype=torch.bfloat16,
                                         device=device)
            self.flat_gb16 = torch.zeros(0, dtype=torch.bfloat16,
                                         device=device)

        # (param, grad-buffer key, offset into that buffer, numel) — the
        # comm-overlap path (train/graphstep.py) uses this to find the
        # flat-buffer split offsets for a given backward bucket boundary
        self.param_layout: List[Tuple[torch.nn.Parameter, str, int, int]] = []
        off = 0
        for p in bf16_set:
            n = p.numel()
            # fp32 master holds the canonical value …
            self.flat_master[off:off + n].as_strided(
                p.shape, p.stride()).copy_(p.data)
            # … the param itself becomes the bf16 working view, grads
            # land in the bf16 flat buffer (param's own stride order)
            pview = self.flat_pb16[off:off + n].as_strided(
                p.shape, p.stride())
            pview.copy_(p.data)
            p.data = pview
            p.grad = self.flat_gb16[off:off + n].as_strided(
                p.shape, p.stride())
            self.param_layout.append((p, "bf16", off, n))
            off += n
        foff = 0
        for p in fp32_set:
            n = p.numel()
            pview = self.flat_master[off:off + n].as_strided(
                p.shape, p.stride())
            pview.copy_(p.data)
            p.data = pview
            p.grad = self.flat_grads[foff:foff + n].as_strided(
                p.shape, p.stride())
            self.param_layout.append((p, "f32", foff, n))
            off += n
            foff += n

        self.exp_avg = torch.zeros_like(self.flat_master)
        self.exp_avg_sq = torch.zeros_like(self.flat_master)
        self.step_t = torch.zeros(1, dtype=torch.int32, device=device)
        # graphstep's capturable check reads param_groups
        self.param_groups = [{
            "params": self.params, "lr": lr, "betas": betas, "eps": eps,
            "weight_decay": weight_decay, "capturable": True,
        }]

    # kept for introspection/back-compat: FlatAdam without bf16 params
    # has exactly one grad buffer
    @property
    def flat_params(self) -> torch.Tensor:
        return self.flat_master

    @property
    def grad_buffers(self) -> List[torch.Tensor]:
        bufs = []
        if self.n_bf16:
            bufs.append(self.flat_gb16)
        if self.flat_grads.numel():
            bufs.append(self.flat_grads)
        return bufs

    @torch.no_grad()
    def step(self) -> None:
        if self.flat_master.is_cuda:
            from ..ops import _C, require_ext
            require_ext()
            if self.n_bf16:
                _C.adam_step_mixed(self.flat_master, self.flat_gb16,
                                   self.flat_grads, self.exp_avg,
                                   self.exp_avg_sq, self.flat_pb16,
                                   self.step_t, self.lr, self.betas[0],
                                   self.betas[1], self.eps,
                                   self.weight_decay)
            else:
                _C.adam_step(self.flat_master, self.flat_grads,
                             self.exp_avg, self.exp_avg_sq, self.step_t,
                             self.lr, self.betas[0], self.betas[1],
                             self.eps, self.weight_decay)
            return
        # CPU reference (numerics oracle): identical math, torch ops
        self.step_t += 1
        t = int(self.step_t.item())
        b1, b2 = self.betas
        if self.n_bf16:
            g = torch.cat([self.flat_gb16.float(), self.flat_grads])
        else:
            g = self.flat_grads
        if self.weight_decay:
            g = g + self.weight_decay * self.flat_master
        self.exp_avg.mul_(b1).add_(g, alpha=1 - b1)
        self.exp_avg_sq.mul_(b2).addcmul_(g, g, value=1 - b2)
        bc1 = 1 - b1 ** t
        bc2 = 1 - b2 ** t
        denom = (self.exp_avg_sq / bc2).sqrt_().add_(self.eps)
        self.flat_master.addcdiv_(self.exp_avg, denom, value=-self.lr / bc1)
        if self.n_bf16:
            self.flat_pb16.copy_(self.flat_master[:self.n_bf16])

    def split_offsets(self, late_params) -> dict:
        """Per-grad-buffer split offset for a two-bucket all-reduce.

        ``late_params``: the params whose grads are ready after the FIRST
        backward stage (the layers closest to the loss). Returns
        ``{"bf16": off, "f32": off}`` such that each buffer's
        ``[off:]`` tail holds exactly the late params' grads. Raises if
        the late set is not a contiguous tail of each buffer (the flat
        layout follows registration order, so any suffix of the model's
        parameter list is).
        """
        late = {id(p) for p in late_params}
        splits = {}
        for key in ("bf16", "f32"):
            entries = [(o, n, id(p) in late)
                       for p, k, o, n in self.param_layout if k == key]
            if not entries:
                continue
            tail = [o for o, n, is_late in entries if is_late]
            split = min(tail) if tail else (
                max(o + n for o, n, _ in entries))
            for o, n, is_late in entries:
                if is_late != (o >= split):
                    raise ValueError(
                        "late params are not a contiguous tail of the "
                        f"{key} grad buffer — comm overlap needs a "
                        "boundary that splits the registration order")
            splits[key] = split
        return splits

    @torch.no_grad()
    def sync_master_from_params(self) -> None:
        """Copy the current parameter values back into the fp32 master.

        Needed after a weights-only resume: ``model.load_state_dict``
        writes the bf16 working views, and without this the master (which
        the next ``step()`` treats as canonical) silently keeps its
        construction-time values. fp32 params are views INTO the master,
        so only the bf16 segment needs the copy-back.
        """
        if self.n_bf16:
            self.flat_master[:self.n_bf16].copy_(self.flat_pb16)

    def zero_grad(self, set_to_none: bool = False) -> None:
        # grads are views of the flat buffers; zeroing the buffers is the
        # only correct form (set_to_none would detach the views)
        if self.n_bf16:
            self.flat_gb16.zero_()
        self.flat_grads.zero_()

    def state_dict(self) -> dict:
        return {
            "flat_params": self.flat_master, "exp_avg": self.exp_avg,
            "exp_avg_sq": self.exp_avg_sq, "step": self.step_t,
            "n_bf16": self.n_bf16, "lr": self.lr, "betas": self.betas,
            "eps": self.eps, "weight_decay": self.weight_decay,
        }

    def load_state_dict(self, sd: dict) -> None:
        if "flat_params" not in sd:
            raise ValueError(
                "checkpoint holds a torch.optim state dict, not FlatAdam's "
                "flat-buffer format — resume graph-mode runs with "
                "graph-mode checkpoints (or load only the model weights)")
        self.flat_master.copy_(sd["flat_params"])
        self.exp_avg.copy_(sd["exp_avg"])
        self.exp_avg_sq.copy_(sd["exp_avg_sq"])
        self.step_t.copy_(sd["step"])
        if self.n_bf16:
            self.flat_pb16.copy_(self.flat_master[:self.n_bf16])
