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
    print("| layer | dgrad2 (v2, wdec+conv+mask fused) | dgrad MIOpen "
          "pipeline (decode+conv+mask) | dgrad MIOpen conv only "
          "| wgrad2 (v2, from bits) | wgrad MIOpen conv only |")
    print("|---|---|---|---|---|")
    for (C, H, K) in [(64, 56, 64), (128, 28, 128), (256, 14, 256),
                      (512, 7, 512)]:
        x = torch.randn(N, C, H, H, device="cuda")
        w = torch.randn(K, C, 3, 3, device="cuda")
        g = cl(torch.randn(N, K, H, H, device="cuda", dtype=torch.bfloat16))
        xp, mp = nat.sign_mask_pack_nhwc(cl(x))
        wp, alpha, stab = nat.weight_pack(w)
        xb = cl(binsign(x).to(torch.bfloat16))
        wb = (weight_scale(w) * binsign(w)).to(torch.bfloat16)

        def dgrad2():
            wd = nat.dgrad_weight_decode(wp, alpha, C)
            return nat.conv_dgrad2(g, wd, mp, C)

        def dgrad_miopen_pipeline():
            # everything dgrad2 replaces: operand decode + igemm + mask
            xb_ = nat.decode_packed(xp, C, True)
            wb_ = nat.weight_decode(wp, alpha, C, True)
            dxb = torch.ops.aten.convolution_backward(
                g, xb_, wb_, None, [1, 1], [1, 1], [1, 1], False, [0, 0],
                1, [True, False, False])[0]
            return nat.mask_mul_packed(dxb, mp, C, True)

        d2 = timeit(dgrad2)
        d_pipe = timeit(dgrad_miopen_pipeline)
        d_mi = timeit(lambda: torch.ops.aten.convolution_backward(
            g, xb, wb, None, [1, 1], [1, 1], [1, 1], False, [0, 0], 1,
            [True, False, False])[0])
        def wgrad2():
            # everything the v2 wgrad path runs per backward: bit repack,
            # all-9-tap block GEMM, transpose + STE-mask finish
            xcp = nat.repack_cplane(xp, C, H)
            dwT = nat.conv_wgrad2(g, xcp, C)
            return nat.wgrad_finish(dwT, w)

        try:
            w2 = timeit(wgrad2)
        except (AttributeError, RuntimeError):
            w2 = float("nan")
        w_mi = timeit(lambda: torch.ops.aten.convolution_backward(
            g, xb, wb, None, [1, 1], [1, 1], [1, 1], False, [0, 0], 1,
            [False, True, False])[1])
        print(f"| {C}x{H}x{H}->{K} | {d2:.3f} | {d_pipe:.3f} | {d_mi:.3f} "
              f"| {w2:.3f} | {w_mi:.3f} |")


if __name__ == "__main__":
    main()
