"""Per-shape micro-benchmark: MFMA 3x3/s1 weight grad vs MIOpen, over
the ResNet-18 block-conv family at bs 212."""
import json
import sys
import time

import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from mi355x_scale.ops import _C  # noqa: E402

SHAPES = [(64, 56), (128, 28), (256, 14), (512, 7)]


def timeit(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    torch.backends.cudnn.benchmark = True
    out = {}
    for C, HW in SHAPES:
        g = torch.Generator().manual_seed(C)
        x = torch.randn(212, C, HW, HW, generator=g).cuda() \
            .to(torch.bfloat16).to(memory_format=torch.channels_last)
        dy = torch.randn(212, C, HW, HW, generator=g).cuda() \
            .to(torch.bfloat16).to(memory_format=torch.channels_last)
        wt = (torch.randn(C, C, 3, 3, generator=g) * 0.05).cuda() \
            .to(torch.bfloat16).to(memory_format=torch.channels_last)
        nch = _C.conv3x3_wrw_nchunks(212, HW, C, 0)
        scratch = torch.empty(nch * C * 9 * C, dtype=torch.float32,
                              device="cuda")
        dw = torch.empty_like(wt)

        def mio():
            return torch.ops.aten.convolution_backward(
                dy, x, wt, None, [1, 1], [1, 1], [1, 1], False, [0, 0],
                1, [False, True, False])[1]
        t_mio = timeit(mio)

        def ours():
            _C.conv3x3_wrw(x.permute(0, 2, 3, 1), dy.permute(0, 2, 3, 1),
                           scratch, dw.permute(0, 2, 3, 1))
        t_our = timeit(ours)

        # quick numerics check at this exact shape
        ref = torch.ops.aten.convolution_backward(
            dy.float(), x.float(), wt.float(), None, [1, 1], [1, 1],
            [1, 1], False, [0, 0], 1, [False, True, False])[1]
        ours()
        torch.cuda.synchronize()
        rel = ((dw.float() - ref).norm() / ref.norm()).item()
        out[f"C{C}_hw{HW}"] = {
            "miopen_us": round(t_mio, 1), "mfma_us": round(t_our, 1),
            "speedup": round(t_mio / t_our, 2), "rel_err": round(rel, 5),
        }
        print(f"# C={C} hw={HW}: miopen {t_mio:.0f} us, ours {t_our:.0f} "
              f"us ({t_mio / t_our:.2f}x), rel {rel:.4f}",
              file=sys.stderr)
    print(json.dumps(out))


if __name__ == "__main__":
    main()
