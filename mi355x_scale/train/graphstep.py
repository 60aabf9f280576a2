"""hipGraph-captured training step.

The MI355X replacement for the reference's Lightning DDP step
(``deep_learning/2.distributed-data-loading-petastorm.py:390-397``):
where Lightning lets NCCL bucket-hook ~400 eager kernel launches per
step, here the whole step replays as two hipGraphs with one flat
RCCL all-reduce between them.

The profiled streaming bench showed the step is launch-bound: ~400 kernel
launches per step from a GIL-contended Python thread leave the GPU ~45%
idle (p50 15.5 ms, p95 80+ ms). The MI355X-idiomatic fix (north star:
"HIP streams and graphs instead of a tracing compiler") is to capture the
whole step — grad-zero, forward, loss, backward, (optimizer) — into one
hipGraph and replay it with a single launch per step.

Data parallism is done manually around the graph (DDP's bucketed hooks
can't fire inside a replay): gradients live in ONE flat buffer
(``p.grad`` are views), the all-reduce is a single RCCL call on that
buffer between the fwd/bwd graph and the optimizer graph. For ResNet-18
(~45 MB fp32 grads) one flat all-reduce over xGMI beats DDP's bucket
pipeline — there is almost no backward left to overlap with after graph
capture.
"""
from __future__ import annotations

from typing import Dict, Optional

import torch
import torch.distributed as dist


def allreduce_flat(flat: torch.Tensor, world_size: int,
                   state: Optional[Dict] = None) -> None:
    """Sum-all-reduce one flat fp32 gradient buffer across ranks.

    Strategy ladder:
      * gloo (CPU tests): host-staged;
      * world >= 4 (``MI355X_ALLREDUCE_BF16`` != "0"): bf16-compressed —
        at 4+ ranks the flat fp32 all-reduce is the inter-step serial
        cost, and halving the xGMI bytes costs two extra passes over the
        buffer (~20 us each for 45 MB). Gradient averaging in bf16 is
        standard DDP practice. Same compression on gloo so the
        world-size-4 CPU test covers the semantics;
      * ``MI355X_P2P_ALLREDUCE=1``: the hand-written direct-xGMI path
        (7 concurrent link reads vs the ring's single-link bound) —
        needs the buffer from p2p_allreduce.alloc_shared;
      * otherwise: stock RCCL ring over xGMI.

    ``state`` carries lazy staging buffers across calls.
    """
    import os
    state = state if state is not None else {}
    compress = (world_size >= 4 and flat.dtype == torch.float32
                and os.environ.get("MI355X_ALLREDUCE_BF16", "1") == "1")
    if dist.get_backend() == "gloo":
        h = (flat.to(torch.bfloat16) if compress else flat).cpu()
        dist.all_reduce(h)
        flat.copy_(h)
        return
    if flat.dtype == torch.bfloat16:
        dist.all_reduce(flat)  # already half-width (bf16-param grads)
        return
    if compress:
        if "bf16" not in state:
            state["bf16"] = torch.empty_like(flat, dtype=torch.bfloat16)
        state["bf16"].copy_(flat)
        dist.all_reduce(state["bf16"])
        flat.copy_(state["bf16"])
        return
    if (os.environ.get("MI355X_P2P_ALLREDUCE") == "1"
            and flat.dtype == torch.float32):
        if "p2p" not in state:
            from ..parallel.p2p_allreduce import P2PAllReduce
            state["p2p"] = P2PAllReduce(flat)
        state["p2p"].all_reduce_()
        return
    dist.all_reduce(flat)  # RCCL over xGMI


class GraphedTrainStep:
    def __init__(self, model, optimizer, example_batch: Dict[str, torch.Tensor],
                 autocast_dtype: Optional[torch.dtype] = torch.bfloat16,
                 world_size: int = 1, warmup: int = 3):
        self.model = model
        self.optimizer = optimizer
        self.world_size = world_size
        self.device = next(model.parameters()).device
        assert self.device.type == "cuda", "graph capture needs a GPU"
        self.autocast_dtype = autocast_dtype
        # MIOpen find mode is REQUIRED under graph capture: immediate-mode
        # can select conv solutions whose workspace preparation runs at
        # find/capture time instead of in-stream, so the first replay is
        # clean and every later replay reads the previous replay's
        # workspace residue (observed as deterministic NaN conv weight
        # grads from step 1 in the two-rank test). benchmark=True records
        # the workspace-zeroing ops inside the captured graph.
        torch.backends.cudnn.benchmark = True

        # Static input buffers (graph replays read these addresses).
        self.static_batch = {
            k: v.to(self.device).clone() for k, v in example_batch.items()
        }

        # Flat gradient buffer(s); every p.grad is a view into one so the
        # cross-rank all-reduce is one call per buffer. FlatAdam
        # (train/flat_adam.py) already owns them with the views installed
        # — reuse so optimizer + all-reduce share one layout (bf16-param
        # mode has two: a bf16 buffer for matrix params, fp32 for the
        # rest).
        if hasattr(optimizer, "grad_buffers"):
            self.grad_buffers = list(optimizer.grad_buffers)
        elif hasattr(optimizer, "flat_grads"):
            self.grad_buffers = [optimizer.flat_grads]
        else:
            params = [p for p in model.parameters() if p.requires_grad]
            total = sum(p.numel() for p in params)
            flat = torch.zeros(total, device=self.device,
                               dtype=torch.float32)
            off = 0
            for p in params:
                # grad views adopt each param's own (dense) stride order —
                # channels-last conv weights get channels-last grad views,
                # so backward accumulates without a per-weight layout
                # permute (the "gradient layout contract") and the flat
                # all-reduce still sums identical byte layouts across
                # ranks.
                p.grad = flat[off:off + p.numel()].as_strided(
                    p.shape, p.stride())
                off += p.numel()
            self.grad_buffers = [flat]
        self.flat_grads = self.grad_buffers[0]  # back-compat alias
        self._ar_states = [dict() for _ in self.grad_buffers]

        def _fwd_bwd():
            for b in self.grad_buffers:
                b.zero_()
            with torch.autocast(device_type="cuda", dtype=autocast_dtype,
                                enabled=autocast_dtype is not None):
                loss = model.training_step(self.static_batch, 0)
            loss.backward()
            return loss

        # Warmup on a side stream (MIOpen find, allocator steady state)
        s = torch.cuda.Stream(device=self.device)
        s.wait_stream(torch.cuda.current_stream(self.device))
        with torch.cuda.stream(s):
            for _ in range(max(1, warmup)):
                _fwd_bwd()
                if world_size > 1:
                    # identical to the replay-time step: keep ranks'
                    # parameters bit-identical through warmup too
                    self._allreduce_grads()
                    for b in self.grad_buffers:
                        b.div_(world_size)
                optimizer.step()
        torch.cuda.current_stream(self.device).wait_stream(s)
        torch.cuda.synchronize(self.device)

        # Capture forward+backward (one graph). thread_local capture
        # mode: the streaming loader's stager threads keep issuing H2D
        # copies on their own streams during capture — global mode would
        # abort the process on their first HIP call.
        self.g_fwd_bwd = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.g_fwd_bwd,
                              capture_error_mode="thread_local"):
            self.static_loss = _fwd_bwd()

        # ...and the optimizer step (second graph; the eager RCCL
        # all-reduce slots between the two replays when world_size > 1).
        # Only optimizers declaring capturable=True are captured — an
        # exception thrown mid-capture (e.g. fused Adam's capturable
        # check) leaves the HIP stream in a broken capture state that
        # later faults, so no try-capture-and-fallback.
        self.g_opt = None
        capturable = all(group.get("capturable", False)
                         for group in optimizer.param_groups)
        if capturable:
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g, capture_error_mode="thread_local"):
                if world_size > 1:
                    for b in self.grad_buffers:
                        b.div_(world_size)
                optimizer.step()
            self.g_opt = g

    def _allreduce_grads(self) -> None:
        for b, st in zip(self.grad_buffers, self._ar_states):
            allreduce_flat(b, self.world_size, st)

    def step(self, batch: Dict[str, torch.Tensor]) -> torch.Tensor:
        for k, v in batch.items():
            self.static_batch[k].copy_(v, non_blocking=True)
        self.g_fwd_bwd.replay()
        if self.world_size > 1:
            self._allreduce_grads()
        if self.g_opt is not None:
            self.g_opt.replay()
        else:
            if self.world_size > 1:
                for b in self.grad_buffers:
                    b.div_(self.world_size)
            self.optimizer.step()
        return self.static_loss
