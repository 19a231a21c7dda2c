#!/usr/bin/env python3
"""dgrad2 with vs without the fused skip-grad acc operand (per layer).

  python benchmarks/dgrad_acc_ab.py [batch]
"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from bdbnn_amd import _C


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
    N = int(sys.argv[1]) if len(sys.argv) > 1 else 512
    print(f"batch {N}; ms per call")
    print("| layer | dgrad2 | dgrad2+acc | separate add | fused win |")
    print("|---|---|---|---|---|")
    for (C, H, K) in [(64, 56, 64), (128, 28, 128), (256, 14, 256),
                      (512, 7, 512)]:
        g = cl(torch.randn(N, K, H, H, device="cuda",
                           dtype=torch.bfloat16))
        w = torch.randn(K, C, 3, 3, device="cuda")
        x = torch.randn(N, C, H, H, device="cuda")
        skip = cl(torch.randn(N, C, H, H, device="cuda",
                              dtype=torch.bfloat16))
        wp, alpha, _ = nat.weight_pack(w)
        _, mp = nat.sign_mask_pack_nhwc(cl(x))
        wd = nat.dgrad_weight_decode(wp, alpha, C)
        t0 = timeit(lambda: nat.conv_dgrad2(g, wd, mp, C))
        t1 = timeit(lambda: nat.conv_dgrad2(g, wd, mp, C, skip))
        dx = nat.conv_dgrad2(g, wd, mp, C)
        t2 = timeit(lambda: dx + skip)
        win = (t0 + t2) - t1
        print(f"| {C}x{H}x{H} | {t0:.3f} | {t1:.3f} | {t2:.3f} "
              f"| {win:+.3f} |")


if __name__ == "__main__":
    main()
