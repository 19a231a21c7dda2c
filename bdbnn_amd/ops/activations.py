"""Fused activations.  ChannelPReLU replaces nn.PReLU on the GPU path:
torch's prelu_backward on channels_last bf16 was 40% of the whole BD-BNN
training step on MI355X (profiles/r01_bench_b256_kernel_stats.md); the
HIP kernel (csrc/prelu.hip) is one memory-bound pass each way.

State-dict compatible with nn.PReLU (same parameter name/shape).
"""

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import _C


class _ChannelPReLUFn(torch.autograd.Function):
    @staticmethod
    def _fused_ok(x):
        C = x.size(1)
        return (x.is_cuda and x.dim() == 4 and C <= 1024
                and (C & (C - 1)) == 0)

    @staticmethod
    def forward(ctx, x, w):
        ctx.save_for_backward(x, w)
        if _ChannelPReLUFn._fused_ok(x):
            return _C.native_required().prelu_fwd(x, w)
        return F.prelu(x, w.to(x.dtype))

    @staticmethod
    def backward(ctx, g):
        x, w = ctx.saved_tensors
        if _ChannelPReLUFn._fused_ok(x):
            dx, da = _C.native_required().prelu_bwd(g.to(x.dtype), x, w)
            return dx, da.to(w.dtype)
        mask = x > 0
        aw = w.to(x.dtype).view(1, -1, *([1] * (x.dim() - 2)))
        dx = torch.where(mask, g, aw * g)
        red_dims = (0,) + tuple(range(2, x.dim()))
        da = torch.where(mask, torch.zeros_like(x), x * g).sum(dim=red_dims)
        return dx, da.to(w.dtype)


class ChannelPReLU(nn.Module):
    """Per-channel PReLU, fused HIP fwd/bwd on GPU (nn.PReLU-compatible)."""

    def __init__(self, num_parameters, init=0.25):
        super().__init__()
        self.num_parameters = num_parameters
        self.weight = nn.Parameter(torch.full((num_parameters,), float(init)))

    def forward(self, x):
        return _ChannelPReLUFn.apply(x, self.weight)

    def extra_repr(self):
        return f"num_parameters={self.num_parameters}"
