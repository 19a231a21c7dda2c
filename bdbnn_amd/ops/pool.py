"""Fused NHWC max-pool (stem 3x3/s2/p1) with u8 argmax + gather backward
(csrc/pool.hip); drop-in for nn.MaxPool2d on the GPU path."""

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import _C


class _MaxPoolFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, ks, stride, pad):
        nat = _C.native_required()
        out, idx = nat.maxpool_fwd(x, ks, stride, pad)
        ctx.save_for_backward(idx)
        ctx.meta = (x.shape[2], x.shape[3], ks, stride, pad)
        return out

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        H, W, ks, stride, pad = ctx.meta
        dx = _C.native_required().maxpool_bwd(dy, idx, H, W, ks, stride, pad)
        return dx, None, None, None


class FusedMaxPool2d(nn.Module):
    """MaxPool2d(kernel, stride, pad) with the fused NHWC kernels on GPU.

    No parameters — state-dict identical to nn.MaxPool2d.
    """

    def __init__(self, kernel_size=3, stride=2, padding=1):
        super().__init__()
        self.kernel_size = kernel_size
        self.stride = stride
        self.padding = padding

    def forward(self, x):
        C = x.size(1) if x.dim() == 4 else 0
        if (x.is_cuda and x.dim() == 4 and 8 <= C <= 1024
                and (C & (C - 1)) == 0 and _C.has_native()):
            return _MaxPoolFn.apply(x, self.kernel_size, self.stride,
                                    self.padding)
        return F.max_pool2d(x, self.kernel_size, self.stride, self.padding)

    def extra_repr(self):
        return (f"kernel_size={self.kernel_size}, stride={self.stride}, "
                f"padding={self.padding}")
