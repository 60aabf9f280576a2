"""Direct xGMI peer-to-peer all-reduce (SURVEY §2.4 deliverable c).

The latency-optimized alternative to RCCL's ring for the flat-gradient
all-reduce: MI355X GPUs are a full 8-way mesh (7 point-to-point xGMI
links per GPU at ~153 GB/s), so a ring pipeline is bound by one link
while the direct pattern uses all links at once:

  phase 0  rendezvous: every rank allocates its flat buffer from
           ``_p2p.alloc_shared_f32`` and all-gathers hipIpc handles over
           the existing torch.distributed group (one-time setup)
  phase 1  reduce-scatter: rank r pulls every peer's copy of shard r and
           sums it into its own buffer (one kernel, 7 concurrent link
           reads)
  phase 2  all-gather: rank r pulls peer p's reduced shard p
           (7 concurrent hipMemcpyAsync peer reads)

Cross-rank ordering between phases uses tiny RCCL all-reduces as
*stream-level* barriers: a collective kernel on rank r's stream can only
complete once every rank's contribution (stream-ordered after its prior
work) has arrived — no host synchronization anywhere, so the whole
sequence captures into hipGraphs like the rest of the step.

The reference *disables* P2P (``NCCL_P2P_DISABLE=1``,
``deep_learning/2.distributed-data-loading-petastorm.py:362-363``) as a
cloud-GPU workaround; here peer access is asserted at setup instead.

Correctness is covered by a 1-GPU two-process test (IPC to the same
device, ``tests/test_p2p_allreduce.py``); on an 8-GPU node the driver's
bench can enable it via ``MI355X_P2P_ALLREDUCE=1``.
"""
from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist

try:
    from . import _p2p
    HAVE_P2P_EXT = True
except ImportError:  # built by setup.py build_ext --inplace
    _p2p = None
    HAVE_P2P_EXT = False


def alloc_shared(numel: int, device) -> torch.Tensor:
    """IPC-shareable fp32 flat buffer on ``device``."""
    if _p2p is None:
        raise RuntimeError("mi355x_scale.parallel._p2p is not built")
    dev = torch.device(device)
    return _p2p.alloc_shared_f32(numel, dev.index or 0)


class P2PAllReduce:
    """All-reduce of one shared flat fp32 buffer across the local ranks.

    ``flat`` must come from :func:`alloc_shared` (hipIpc handles cover
    whole allocations). After ``all_reduce_()`` every rank's buffer holds
    the element-wise sum.
    """

    def __init__(self, flat: torch.Tensor, group=None):
        if _p2p is None:
            raise RuntimeError("mi355x_scale.parallel._p2p is not built")
        self.flat = flat
        self.group = group
        self.rank = dist.get_rank(group)
        self.world = dist.get_world_size(group)
        n = flat.numel()
        chunk = (n + self.world - 1) // self.world
        self.bounds = [(min(r * chunk, n), min((r + 1) * chunk, n))
                       for r in range(self.world)]
        # one-time IPC handle exchange over the existing process group
        handle = _p2p.get_ipc_handle(flat)
        handles: List[Optional[bytes]] = [None] * self.world
        dist.all_gather_object(handles, handle, group=group)
        self.peer_ptrs = {}
        for r, h in enumerate(handles):
            if r != self.rank:
                self.peer_ptrs[r] = _p2p.open_ipc_handle(h)
        # tiny device scalar reused by the stream-barrier collectives
        self._barrier_t = torch.zeros(1, device=flat.device)

    def _stream_barrier(self) -> None:
        # a 1-element RCCL all-reduce: completion on this rank's stream
        # implies every rank's prior stream work (their contribution is
        # stream-ordered) has finished — a device-level barrier with no
        # host sync. On gloo groups (the 1-GPU two-process test) fall
        # back to an explicit device sync + host barrier.
        if dist.get_backend(self.group) == "gloo":
            torch.cuda.synchronize(self.flat.device)
            dist.barrier(group=self.group)
        else:
            dist.all_reduce(self._barrier_t, group=self.group)

    def all_reduce_(self) -> torch.Tensor:
        if self.world == 1:
            return self.flat
        lo, hi = self.bounds[self.rank]
        srcs = [self.peer_ptrs[r] for r in range(self.world)
                if r != self.rank]
        self._stream_barrier()  # all ranks' buffers fully written
        if hi > lo:
            _p2p.reduce_add(self.flat, srcs, lo, hi - lo)
        self._stream_barrier()  # all shards reduced
        for r in range(self.world):
            if r == self.rank:
                continue
            plo, phi = self.bounds[r]
            if phi > plo:
                _p2p.copy_from_peer(self.flat, self.peer_ptrs[r], plo,
                                    phi - plo)
        self._stream_barrier()  # peers' reduced shards may be reused
        return self.flat

    def close(self) -> None:
        for ptr in self.peer_ptrs.values():
            _p2p.close_ipc_handle(ptr)
        self.peer_ptrs.clear()
