#!/usr/bin/env python3
"""BN-family + conv-stats microbench on MI355X.

Times, per ResNet-18 layer shape at batch 512:
  * xnor_conv_fwd with and without the epilogue stats (the stats
    variant must cost ~nothing or the fused-BN stats hand-off loses),
  * bn_stats / bn fwd / bn backward reduce+apply.

  python benchmarks/bn_bench.py [batch]
"""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from bdbnn_amd import _C


def timeit(fn, iters=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3  # ms


def main():
    nat = _C.native_required()
    cl = lambda t: t.contiguous(memory_format=torch.channels_last)
    N = int(sys.argv[1]) if len(sys.argv) > 1 else 512
    print(f"batch {N}; ms per call\n")
    print("| layer | conv | conv+stats | fwd_prestats | bn_fwd | "
          "bn_fwd+pack | bn_bwd |")
    print("|---|---|---|---|---|---|---|")
    for (C, H, K) in [(64, 56, 64), (128, 28, 128), (256, 14, 256),
                      (512, 7, 512)]:
        x = torch.randn(N, C, H, H, device="cuda")
        w = torch.randn(K, C, 3, 3, device="cuda")
        xp = nat.sign_pack_nhwc(cl(x))
        wp, alpha, stab = nat.weight_pack(w)
        conv = timeit(lambda: nat.xnor_conv_fwd(xp, wp, alpha, stab, C, 1,
                                                1, True, False))
        conv_s = timeit(lambda: nat.xnor_conv_fwd(xp, wp, alpha, stab, C,
                                                  1, 1, True, True))
        out = nat.xnor_conv_fwd(xp, wp, alpha, stab, C, 1, 1, True,
                                False)[0]
        gamma = torch.randn(K, device="cuda").abs() + 0.5
        beta = torch.randn(K, device="cuda") * 0.1
        a = torch.rand(K, device="cuda") * 0.3
        rm = torch.zeros(K, device="cuda")
        rv = torch.ones(K, device="cuda")
        s1 = torch.zeros(32, K, device="cuda")
        s2 = torch.ones(32, K, device="cuda")
        # with pre-stats the fwd skips its stats read pass: the delta
        # isolates the bn_stats kernel's cost
        bns = timeit(lambda: nat.bn_act_fwd_train(out, None, gamma, beta,
                                                  a, rm, rv, 0.1, 1e-5, 1,
                                                  s1, s2, False))
        fwd = timeit(lambda: nat.bn_act_fwd_train(out, None, gamma, beta,
                                                  a, rm, rv, 0.1, 1e-5, 1,
                                                  None, None, False))
        fwdp = timeit(lambda: nat.bn_act_fwd_train(out, None, gamma, beta,
                                                   a, rm, rv, 0.1, 1e-5, 1,
                                                   None, None, True))
        res = nat.bn_act_fwd_train(out, None, gamma, beta, a, rm, rv, 0.1,
                                   1e-5, 1, None, None, False)
        o, z, mean, invstd = res[:4]
        dy = cl(torch.randn_like(o))
        bwd = timeit(lambda: nat.bn_act_bwd(dy, z, out, mean, invstd,
                                            gamma, a, 1, False))
        print(f"| {C}x{H}x{H}->{K} | {conv:.3f} | {conv_s:.3f} | "
              f"{bns:.3f} | {fwd:.3f} | {fwdp:.3f} | {bwd:.3f} |")


if __name__ == "__main__":
    main()
