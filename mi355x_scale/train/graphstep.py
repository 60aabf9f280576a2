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
buffer between the fwd/bwd graph and the optimizer graph.

Comm/compute overlap (round 2): with a boundary module declared
(``model.comm_overlap_boundary``), the backward is split in two captured
graphs at that activation — the classic pipeline cut: the boundary
module's output is detached into a leaf, so ``loss.backward()`` stops
there after producing the LATE layers' grads (the tail of the flat
buffers, ~94% of ResNet-18's bytes), and a second graph continues
``a.backward(a2.grad)`` through the early layers. Between the two
replays the tail all-reduce runs on a dedicated comm stream, overlapped
with the early-backward replay — the overlap DDP's bucket hooks give the
reference for free (``deep_learning/2.distributed-data-loading-
petastorm.py:390-397``), bought back under graph capture.
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

        # Grad buffers by layout key (FlatAdam only) — the comm-overlap
        # path needs named buffers + per-param offsets to split them.
        self._buf_by_key = {}
        if hasattr(optimizer, "param_layout"):
            if getattr(optimizer, "n_bf16", 0):
                self._buf_by_key["bf16"] = optimizer.flat_gb16
            if optimizer.flat_grads.numel():
                self._buf_by_key["f32"] = optimizer.flat_grads
        self._overlap = self._setup_overlap(model, optimizer, world_size)
        self.g_bwd2 = None
        holder = self._overlap["holder"] if self._overlap else None

        def _fwd():
            for b in self.grad_buffers:
                b.zero_()
            with torch.autocast(device_type="cuda", dtype=autocast_dtype,
                                enabled=autocast_dtype is not None):
                return model.training_step(self.static_batch, 0)

        def _fwd_bwd():
            loss = _fwd()
            loss.backward()
            if self._overlap is not None:
                # continue through the early layers from the cut point
                holder["a"].backward(holder["a2"].grad)
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

        # Capture forward+backward. thread_local capture mode: the
        # streaming loader's stager threads keep issuing H2D copies on
        # their own streams during capture — global mode would abort the
        # process on their first HIP call.
        self.g_fwd_bwd = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.g_fwd_bwd,
                              capture_error_mode="thread_local"):
            self.static_loss = _fwd()
            # with the cut installed this stops at the boundary leaf,
            # producing only the LATE layers' grads (buffer tails)
            self.static_loss.backward()
        if self._overlap is not None:
            # capture-time tensors: the hook ran inside the capture above,
            # so a/a2 (and a2.grad, written fresh by graph A's recorded
            # kernels — .grad was None at capture, no accumulation) live
            # in graph A's private pool; graph B must share that pool.
            a, a2 = holder["a"], holder["a2"]
            self.g_bwd2 = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self.g_bwd2, pool=self.g_fwd_bwd.pool(),
                                  capture_error_mode="thread_local"):
                a.backward(a2.grad)
            # replays never run python hooks; remove so eager validation
            # forwards are not rerouted through the detached leaf
            self._overlap["hook_handle"].remove()

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

    def _setup_overlap(self, model, optimizer, world_size):
        """Install the boundary cut for split-backward comm overlap.

        Enabled when the module declares ``comm_overlap_boundary`` (a
        submodule path whose OUTPUT activation is the cut), the optimizer
        exposes a flat layout, and either world_size > 1 or
        ``MI355X_GRAPH_OVERLAP=1`` forces it (the 1-GPU parity test).
        ``MI355X_GRAPH_OVERLAP=0`` or the hand p2p all-reduce path
        disable it (p2p stays monolithic).
        """
        import os
        flag = os.environ.get("MI355X_GRAPH_OVERLAP", "auto")
        if flag == "0":
            return None
        if world_size <= 1 and flag != "1":
            return None
        if os.environ.get("MI355X_P2P_ALLREDUCE") == "1":
            return None
        boundary = getattr(model, "comm_overlap_boundary", None)
        if boundary is None or not self._buf_by_key:
            return None
        try:
            mod = model.get_submodule(boundary)
        except AttributeError:
            return None
        named = list(model.named_parameters())
        prefix = boundary + "."
        idxs = [i for i, (n, _) in enumerate(named)
                if n.startswith(prefix)]
        if not idxs or idxs[-1] + 1 >= len(named):
            return None
        late_params = [p for _, p in named[idxs[-1] + 1:]
                       if p.requires_grad]
        try:
            splits = optimizer.split_offsets(late_params)
        except ValueError:
            return None

        holder: Dict[str, torch.Tensor] = {}

        def _cut(module, inputs, out):
            holder["a"] = out
            a2 = out.detach().requires_grad_(True)
            holder["a2"] = a2
            return a2

        return {
            "holder": holder,
            "hook_handle": mod.register_forward_hook(_cut),
            "splits": splits,
            "stream": torch.cuda.Stream(device=self.device),
            "ev_a": torch.cuda.Event(),
            "ev_b": torch.cuda.Event(),
            "ev_done": torch.cuda.Event(),
            "tail_states": {k: {} for k in self._buf_by_key},
            "head_states": {k: {} for k in self._buf_by_key},
        }

    def _allreduce_grads(self) -> None:
        for b, st in zip(self.grad_buffers, self._ar_states):
            allreduce_flat(b, self.world_size, st)

    def step(self, batch: Dict[str, torch.Tensor]) -> torch.Tensor:
        for k, v in batch.items():
            self.static_batch[k].copy_(v, non_blocking=True)
        self.g_fwd_bwd.replay()
        if self.g_bwd2 is not None and self.world_size > 1:
            # tail (late-layer) all-reduce on the comm stream, overlapped
            # with the early-backward replay on the compute stream
            ov = self._overlap
            cur = torch.cuda.current_stream(self.device)
            comm = ov["stream"]
            ov["ev_a"].record(cur)
            comm.wait_event(ov["ev_a"])
            with torch.cuda.stream(comm):
                for key, buf in self._buf_by_key.items():
                    sp = ov["splits"].get(key, 0)
                    if sp < buf.numel():
                        allreduce_flat(buf[sp:], self.world_size,
                                       ov["tail_states"][key])
            self.g_bwd2.replay()
            ov["ev_b"].record(cur)
            comm.wait_event(ov["ev_b"])
            with torch.cuda.stream(comm):
                for key, buf in self._buf_by_key.items():
                    sp = ov["splits"].get(key, 0)
                    if sp > 0:
                        allreduce_flat(buf[:sp], self.world_size,
                                       ov["head_states"][key])
            ov["ev_done"].record(comm)
            cur.wait_event(ov["ev_done"])
        else:
            if self.g_bwd2 is not None:
                self.g_bwd2.replay()  # split capture, single rank
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
