#!/usr/bin/env python3
"""Per-kernel microbenchmarks on MI355X (clean numbers, no framework
noise).  Prints one markdown table; run via gpurun and commit the output
under profiles/.

  python benchmarks/kernel_bench.py
"""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from bdbnn_amd import _C
from bdbnn_amd.ops.binarize import binsign, weight_scale


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
    assert torch.cuda.is_available()
    nat = _C.native_required()
    cl = lambda t: t.contiguous(memory_format=torch.channels_last)
    rows = []

    # ---- XNOR conv across the ResNet-18 layer shapes (batch 512) ----
    print("## xnor_conv_fwd (batch 512, bf16 out)\n")
    print("| layer shape | ms | T binMAC/s | eff. dense-equiv TFLOP/s |")
    print("|---|---|---|---|")
    for (C, H, K, ks, st) in [(64, 56, 64, 3, 1), (128, 28, 128, 3, 1),
                              (256, 14, 256, 3, 1), (512, 7, 512, 3, 1),
                              (64, 56, 128, 1, 2)]:
        N = 512
        x = torch.randn(N, C, H, H, device="cuda")
        w = torch.randn(K, C, ks, ks, device="cuda")
        xp = nat.sign_pack_nhwc(cl(x))
        wp, alpha, stab = nat.weight_pack(w)
        pad = 1 if ks == 3 else 0
        Ho = (H + 2 * pad - ks) // st + 1
        ms = timeit(lambda: nat.xnor_conv_fwd(xp, wp, alpha, stab, C, st,
                                              pad, True, False))
        macs = N * Ho * Ho * K * C * ks * ks
        tmacs = macs / (ms / 1e3) / 1e12
        rows.append((f"{C}x{H}x{H} -> {K}, {ks}x{ks}/s{st}", ms, tmacs))
        print(f"| {C}x{H}x{H} -> {K}, {ks}x{ks}/s{st} | {ms:.3f} | "
              f"{tmacs:.0f} | {2*tmacs:.0f} |")

    # ---- pack / decode / mask (layer2 shape) ----
    N, C, H = 512, 128, 28
    x = torch.randn(N, C, H, H, device="cuda")
    g = torch.randn(N, C, H, H, device="cuda")
    xcl, gcl = cl(x), cl(g)
    gb = (N * C * H * H * 4) / 1e9
    print("\n## quantizer / BN family (512x128x28x28 fp32 = "
          f"{gb:.2f} GB/tensor)\n")
    print("| kernel | ms | effective TB/s (tensors moved / time) |")
    print("|---|---|---|")

    def row(name, ms, n_tensor_passes):
        print(f"| {name} | {ms:.3f} | {n_tensor_passes * gb / ms:.2f} |")

    sp, mp = nat.sign_mask_pack_nhwc(xcl)
    row("sign_mask_pack (1R)", timeit(lambda: nat.sign_mask_pack_nhwc(xcl)), 1)
    row("decode_packed (1W)", timeit(lambda: nat.decode_packed(sp, C, False)), 1)
    row("mask_mul_packed (1R1W)",
        timeit(lambda: nat.mask_mul_packed(gcl, mp, C, False)), 2)
    row("ste_mask_mul (2R1W)",
        timeit(lambda: nat.ste_mask_mul(gcl, xcl, 0, 0.0, 0.0)), 3)

    gamma = torch.rand(C, device="cuda") + 0.5
    beta = torch.randn(C, device="cuda")
    a = torch.rand(C, device="cuda")
    rm = torch.zeros(C, device="cuda")
    rv = torch.ones(C, device="cuda")
    out = nat.bn_act_fwd_train(xcl, gcl, gamma, beta, a, rm, rv, 0.1,
                               1e-5, 1, None, None, False)
    o, z, mean, invstd = out[:4]
    row("bn stats+finalize (1R)",
        timeit(lambda: nat.bn_act_fwd_train(xcl, None, gamma, beta, None,
                                            rm, rv, 0.1, 1e-5, 0, None,
                                            None, False)), 3)
    row("bn_act_bwd (6R2W)",
        timeit(lambda: nat.bn_act_bwd(gcl, z, xcl, mean, invstd, gamma, a,
                                      1, True)), 8)
    row("prelu_fwd (1R1W)", timeit(lambda: nat.prelu_fwd(xcl, a)), 2)
    dy = cl(torch.randn_like(x))
    row("prelu_bwd (2R1W)", timeit(lambda: nat.prelu_bwd(dy, xcl, a)), 3)


if __name__ == "__main__":
    main()
