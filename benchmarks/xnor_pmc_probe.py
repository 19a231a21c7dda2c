#!/usr/bin/env python3
"""Minimal probe for rocprofv3 --pmc runs: repeats the XNOR conv on the
four ResNet-18 layer shapes (batch 512)."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from bdbnn_amd import _C

nat = _C.native_required()
cl = lambda t: t.contiguous(memory_format=torch.channels_last)
for (C, H, K) in [(64, 56, 64), (128, 28, 128), (256, 14, 256),
                  (512, 7, 512)]:
    x = torch.randn(512, C, H, H, device="cuda")
    w = torch.randn(K, C, 3, 3, device="cuda")
    xp = nat.sign_pack_nhwc(cl(x))
    wp, alpha, stab = nat.weight_pack(w)
    for _ in range(10):
        out = nat.xnor_conv_fwd(xp, wp, alpha, stab, C, 1, 1, True, False)
torch.cuda.synchronize()
print("probe done")
