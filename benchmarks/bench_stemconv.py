"""Stem conv micro-benchmark: MFMA kernel vs MIOpen at the flagship
shape ([212,3,224,224] bf16 channels-last, 7x7 s2 p3, 3->64)."""
import json
import sys
import time

import torch
import torch.nn.functional as F

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from mi355x_scale.ops.stemconv import _StemConvFn  # noqa: E402


def timeit(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def main():
    torch.backends.cudnn.benchmark = True
    dev = "cuda"
    n, h, w = 212, 224, 224
    g = torch.Generator().manual_seed(0)
    x = torch.randn(n, 3, h, w, generator=g).to(dev).to(torch.bfloat16) \
        .to(memory_format=torch.channels_last)
    wt = (torch.randn(64, 3, 7, 7, generator=g) * 0.05).to(dev) \
        .to(torch.bfloat16).to(memory_format=torch.channels_last)
    x.requires_grad_(False)

    # --- MIOpen paths
    wt_m = wt.clone().requires_grad_(True)
    t_mio_fwd = timeit(lambda: F.conv2d(x, wt_m, None, 2, 3))
    out = F.conv2d(x, wt_m, None, 2, 3)
    dy = torch.randn(out.shape, generator=g).to(dev).to(torch.bfloat16) \
        .to(memory_format=torch.channels_last)

    def mio_wrw():
        wg = torch.ops.aten.convolution_backward(
            dy, x, wt_m, None, [2, 2], [3, 3], [1, 1], False, [0, 0], 1,
            [False, True, False])[1]
        return wg
    t_mio_wrw = timeit(mio_wrw)

    # --- MFMA paths
    t_fwd = timeit(lambda: _StemConvFn.apply(x, wt))

    wt_g = wt.clone().requires_grad_(True)
    out2 = _StemConvFn.apply(x, wt_g)

    def mfma_wrw():
        wt_g.grad = None
        out2.backward(dy, retain_graph=True)
    t_wrw = timeit(mfma_wrw)

    # numerics cross-check at this exact shape
    ref = F.conv2d(x.float(), wt.float(), None, 2, 3)
    got = _StemConvFn.apply(x, wt).float()
    rel = ((got - ref).norm() / ref.norm()).item()

    print(json.dumps({
        "shape": [n, 3, h, w],
        "fwd_us": {"miopen": round(t_mio_fwd, 1), "mfma": round(t_fwd, 1),
                   "speedup": round(t_mio_fwd / t_fwd, 2)},
        "wrw_us": {"miopen": round(t_mio_wrw, 1), "mfma": round(t_wrw, 1),
                   "speedup": round(t_mio_wrw / t_wrw, 2)},
        "fwd_rel_err_vs_fp32": round(rel, 5),
    }))


if __name__ == "__main__":
    main()
