#!/usr/bin/env python3
"""Stem conv (7x7/2, 3->64) microbench: MFMA kernels vs MIOpen.

  python benchmarks/stem_bench.py [batch]
"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from bdbnn_amd import _C


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    nat = _C.native_required()
    cl = lambda t: t.contiguous(memory_format=torch.channels_last)
    N = int(sys.argv[1]) if len(sys.argv) > 1 else 512
    x = cl(torch.randn(N, 3, 224, 224, device="cuda",
                       dtype=torch.bfloat16))
    w = torch.randn(64, 3, 7, 7, device="cuda") * 0.1
    wb = w.to(torch.bfloat16)
    g = cl(torch.randn(N, 64, 112, 112, device="cuda",
                       dtype=torch.bfloat16))
    _, x4 = nat.stem_conv_fwd(x, w)
    print(f"batch {N}; ms per call")
    print("| op | MFMA (ours) | MIOpen |")
    print("|---|---|---|")
    t_f = timeit(lambda: nat.stem_conv_fwd(x, w))
    t_fm = timeit(lambda: torch.nn.functional.conv2d(
        x, wb, None, stride=2, padding=3))
    print(f"| fwd (incl. x4+w4 pack) | {t_f:.3f} | {t_fm:.3f} |")
    t_w = timeit(lambda: nat.stem_conv_wrw(x4, g))
    t_wm = timeit(lambda: torch.ops.aten.convolution_backward(
        g, x, torch.empty(64, 3, 7, 7, device="cuda",
                          dtype=torch.bfloat16), None,
        [2, 2], [3, 3], [1, 1], False, [0, 0], 1,
        [False, True, False])[1])
    print(f"| wrw (incl. slab fold) | {t_w:.3f} | {t_wm:.3f} |")


if __name__ == "__main__":
    main()
