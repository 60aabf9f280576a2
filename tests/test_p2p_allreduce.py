"""Hand-written xGMI p2p all-reduce: correctness on one GPU.

Two processes share cuda:0 and exchange their flat buffers through
hipIpc handles — the identical machinery the 8-GPU path uses over xGMI
(there the opened pointer maps a peer GPU's HBM; here it maps the same
device, which exercises handle export/open, the multi-source reduce
kernel, the shard arithmetic, and the phase barriers end to end).
"""
import multiprocessing as mp
import os

import numpy as np
import pytest
import torch

NUMEL = 1_000_003  # odd size: uneven shards + scalar tail in the kernel


def _worker(rank: int, world: int, port: int, q):
    try:
        import torch.distributed as dist
        from mi355x_scale.parallel.p2p_allreduce import (P2PAllReduce,
                                                         alloc_shared)
        os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                          RANK=str(rank), WORLD_SIZE=str(world))
        dist.init_process_group("gloo", rank=rank, world_size=world)
        torch.cuda.set_device(0)
        flat = alloc_shared(NUMEL, "cuda:0")
        g = torch.Generator().manual_seed(1000 + rank)
        local = torch.randn(NUMEL, generator=g)
        flat.copy_(local)
        ar = P2PAllReduce(flat)
        ar.all_reduce_()
        torch.cuda.synchronize()
        # expected: sum over all ranks' seeded vectors
        want = torch.zeros(NUMEL)
        for r in range(world):
            gg = torch.Generator().manual_seed(1000 + r)
            want += torch.randn(NUMEL, generator=gg)
        err = (flat.cpu() - want).abs().max().item()
        ar.close()
        dist.barrier()  # peers must finish reading before buffers die
        dist.destroy_process_group()
        q.put((rank, err))
    except Exception as e:  # surfaced by the parent
        q.put((rank, f"FAIL: {type(e).__name__}: {e}"))


@pytest.mark.gpu
def test_two_process_one_gpu_allreduce():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29713
    procs = [ctx.Process(target=_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=300) for _ in procs]
    for p in procs:
        p.join(timeout=60)
    for rank, err in results:
        assert isinstance(err, float), f"rank {rank}: {err}"
        assert err < 1e-4, f"rank {rank} max err {err}"


@pytest.mark.gpu
def test_reduce_add_kernel_local():
    """Multi-source reduce kernel against torch on local buffers."""
    from mi355x_scale.parallel import _p2p
    torch.cuda.set_device(0)
    n = 12_345
    dst = _p2p.alloc_shared_f32(n, 0)
    srcs = [_p2p.alloc_shared_f32(n, 0) for _ in range(3)]
    torch.manual_seed(0)
    dst.copy_(torch.randn(n))
    base = dst.clone()
    acc = base.clone()
    ptrs = []
    for s in srcs:
        s.copy_(torch.randn(n))
        acc += s
        ptrs.append(s.data_ptr())
    _p2p.reduce_add(dst, ptrs, 0, n)
    torch.cuda.synchronize()
    torch.testing.assert_close(dst, acc)
    # sub-range with offset
    dst.copy_(base)
    _p2p.reduce_add(dst, ptrs, 100, 57)
    torch.cuda.synchronize()
    torch.testing.assert_close(dst[100:157], acc[100:157])
    torch.testing.assert_close(dst[:100], base[:100])
    torch.testing.assert_close(dst[157:], base[157:])
