"""Hand-written MFMA stem convolution (7x7 / stride 2 / pad 3, 3 -> 64).

MIOpen's igemm puts C=3 on the GEMM K-dim and runs ~15x off the MFMA
floor (measured 2.9 ms fwd + 2.8 ms wrw per b2048 step, the largest
non-owned block of the profile).  csrc/stem_conv.hip instead makes the
whole 7x7x(3->4-padded) patch the K-dim (224 slots) so every fragment
is an aligned LDS read.  See SURVEY.md K7 (stem was "acceptable" on
MIOpen; owning it is worth ~4 ms/step at b2048).

``StemConv7x7`` subclasses nn.Conv2d so state_dict / init / repr are
untouched; anything off the fast path falls back to the stock conv.
The stem input never requires grad in this framework, so backward only
produces the weight gradient (loud error otherwise).
"""

import os

import torch
import torch.nn as nn

from .. import _C

# wrw backend: the MFMA pixel-split wrw kernel LOSES to MIOpen at the
# flagship batch (6.06 vs 2.70 ms at b2048, gpurun stem_bench A/B) —
# its per-128-pixel-tile pT im2col rebuild is LDS-bandwidth bound — so
# the weight gradient defaults to MIOpen while the forward (2x faster
# than MIOpen) stays on the owned kernel.  BDBNN_STEM_WRW=mfma for A/B.
_WRW_MFMA = os.environ.get("BDBNN_STEM_WRW", "miopen") == "mfma"


class _StemConvFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w):
        nat = _C.native_required()
        out, x4 = nat.stem_conv_fwd(x, w)
        if ctx.needs_input_grad[0]:
            raise RuntimeError("StemConv7x7: input gradients unsupported "
                               "(the stem input is the data batch)")
        if ctx.needs_input_grad[1]:
            if _WRW_MFMA:
                ctx.save_for_backward(x4)
            else:
                xb = x if x.dtype == torch.bfloat16 else x.to(torch.bfloat16)
                ctx.save_for_backward(
                    xb.contiguous(memory_format=torch.channels_last))
        return out

    @staticmethod
    def backward(ctx, gy):
        (saved,) = ctx.saved_tensors
        if _WRW_MFMA:
            nat = _C.native_required()
            dw = nat.stem_conv_wrw(saved, gy)
        else:
            gb = gy if gy.dtype == torch.bfloat16 else gy.to(torch.bfloat16)
            dw = torch.ops.aten.convolution_backward(
                gb.contiguous(memory_format=torch.channels_last), saved,
                torch.empty(64, 3, 7, 7, device=saved.device,
                            dtype=torch.bfloat16), None,
                [2, 2], [3, 3], [1, 1], False, [0, 0], 1,
                [False, True, False])[1].to(torch.float32)
        return None, dw


class StemConv7x7(nn.Conv2d):
    """Drop-in for the ImageNet ResNet stem nn.Conv2d(3, 64, 7, 2, 3)."""

    def forward(self, x):
        fast = (x.is_cuda and _C.has_native()
                and self.in_channels == 3 and self.out_channels == 64
                and self.kernel_size == (7, 7) and self.stride == (2, 2)
                and self.padding == (3, 3) and self.bias is None
                and x.dim() == 4 and x.size(1) == 3
                and x.size(2) % 2 == 0 and x.size(3) % 2 == 0
                and not x.requires_grad
                # the kernel is bf16: only take fp32 inputs when autocast
                # would have made the stock conv bf16 anyway
                and (x.dtype == torch.bfloat16
                     or torch.is_autocast_enabled()))
        if fast:
            return _StemConvFn.apply(x, self.weight)
        return super().forward(x)
