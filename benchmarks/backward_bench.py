#!/usr/bin/env python3
"""Compare the experimental MFMA backward kernels (in-kernel packed
operand decode) against the MIOpen path per ResNet-18 layer shape.

  python benchmarks/backward_bench.py [batch]
"""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from bdbnn_amd import _C
from bdbnn_amd.ops.binarize import binsign, weight_scale


def timeit(fn, iters=20, warmup=3):
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
    N = int(sys.argv[1]) if len(sys.argv) > 1 else 256
    print(f"batch {N}; ms per call\n")
    print("| layer | dgrad MFMA | dgrad MIOpen | wgrad MFMA | wgrad MIOpen |")
    print("|---|---|---|---|---|")
    for (C, H, K) in [(64, 56, 64), (128, 28, 128), (256, 14, 256),
                      (512, 7, 512)]:
        x = torch.randn(N, C, H, H, device="cuda")
        w = torch.randn(K, C, 3, 3, device="cuda")
        g = cl(torch.randn(N, K, H, H, device="cuda", dtype=torch.bfloat16))
        xp = nat.sign_pack_nhwc(cl(x))
        wp, alpha, stab = nat.weight_pack(w)
        xb = cl(binsign(x).to(torch.bfloat16))
        wb = (weight_scale(w) * binsign(w)).to(torch.bfloat16)

        d_mfma = timeit(lambda: nat.conv_dgrad(g, wp, alpha, C))
        d_mi = timeit(lambda: torch.ops.aten.convolution_backward(
            g, xb, wb, None, [1, 1], [1, 1], [1, 1], False, [0, 0], 1,
            [True, False, False])[0])
        w_mfma = timeit(lambda: nat.conv_wgrad(g, xp, C))
        w_mi = timeit(lambda: torch.ops.aten.convolution_backward(
            g, xb, wb, None, [1, 1], [1, 1], [1, 1], False, [0, 0], 1,
            [False, True, False])[1])
        print(f"| {C}x{H}x{H}->{K} | {d_mfma:.3f} | {d_mi:.3f} "
              f"| {w_mfma:.3f} | {w_mi:.3f} |")


if __name__ == "__main__":
    main()
