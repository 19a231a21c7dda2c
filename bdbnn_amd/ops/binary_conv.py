"""1-bit (1W/1A) convolution modules — the compute core of BD-BNN.

The reference's ``models`` package (HardBinaryConv / HardBinaryConv_react /
HardBinaryConv_cifar) is missing from its snapshot; the module contract is
reconstructed from call sites (ref:train.py:30-32,392,
ref:utils/KD_loss.py:6-7,60):

* expose a 4-D ``weight`` nn.Parameter (latent fp weights — kurtosis and
  layer-KD read it directly),
* forward = conv(sign(x), alpha * sign(W)) with straight-through backward,
* accept per-epoch EDE (t, k) injection as module attributes,
* first conv + final FC of a model stay real-valued.

GPU path (CUDA tensors): bit-packed XNOR+popcount convolution via the
in-tree HIP extension (csrc/xnor_conv.hip), activations packed on the fly
(csrc/pack.hip), backward as dense MFMA-bf16 conv on the decoded +-1
operands (MIOpen) with fused STE/EDE mask kernels.  Requires NHWC
(channels_last) activations and C % 32 == 0 for the packed path; the
engine puts models in channels_last.

CPU path: the pure-PyTorch oracle (also the numerics reference for the
kernels' tests).
"""

import os

import torch
import torch.nn as nn
import torch.nn.functional as F

from .binarize import binsign, weight_scale
from .. import _C

_CONV_STATS = os.environ.get("BDBNN_CONV_STATS", "0") == "1"
# hand-written MFMA backward kernels (default ON; BDBNN_MFMA_BWD=0 falls
# back to MIOpen igemm on decoded operands for A/B comparison)
_MFMA_BWD = os.environ.get("BDBNN_MFMA_BWD", "1") != "0"


def _act_grad_mask(x: torch.Tensor, mode: str, t, k) -> torch.Tensor:
    """d sign(x) / dx surrogate used by the backward pass (matches binarize.py)."""
    if t is not None and k is not None:
        th = torch.tanh(float(t) * x)
        return float(k) * float(t) * (1.0 - th * th)
    if mode == "approx":
        neg = (x >= -1) & (x < 0)
        pos = (x >= 0) & (x < 1)
        g = torch.zeros_like(x)
        g = torch.where(neg, 2.0 + 2.0 * x, g)
        g = torch.where(pos, 2.0 - 2.0 * x, g)
        return g
    return (x.abs() <= 1).to(x.dtype)


class BinaryConvFunction(torch.autograd.Function):
    """Fused 1W/1A conv: XNOR+popcount forward, dense-MFMA backward.

    forward:  out = conv2d(sign(x), alpha * sign(w), stride, padding)
    backward: dx = conv_dgrad(g, alpha*sign(w)) * act_mask(x)
              dw = conv_wgrad(sign(x), g) * 1(|w| <= 1)
    (alpha = per-out-channel mean|W|, detached.)
    """

    @staticmethod
    def forward(ctx, x, w, stride, padding, act_mode, t, k,
                want_stats=False, xp_pre=None, mp_pre=None, skip_cell=None):
        ctx.stride = stride
        ctx.padding = padding
        ctx.act_mode = act_mode
        ctx.t = t
        ctx.k = k
        ctx.packed = False
        # deferred residual-skip gradient: the consuming BN's backward
        # (which runs first) stashes its dskip here instead of returning
        # it, and this conv's dgrad adds it in the epilogue — removing
        # autograd's separate grad-accumulation pass over dx
        ctx.skip_cell = skip_cell
        if x.is_cuda:
            nat = _C.native_required()
            wp, alpha, stab = nat.weight_pack(w)  # bits, alpha[K], S[K][T]
            # (grad mode is always off inside Function.forward; use
            # needs_input_grad to detect inference/no-grad calls)
            if not (ctx.needs_input_grad[0] or ctx.needs_input_grad[1]):
                # inference/validation: no mask plane, nothing saved
                xp = (xp_pre if xp_pre is not None else nat.sign_pack_nhwc(
                    x.contiguous(memory_format=torch.channels_last)))
                ctx.packed = None  # backward must never run
            elif act_mode == "ste" and t is None:
                # packed fast path: sign + clip-STE mask bitplanes (either
                # pre-packed by the producing BN's epilogue, or in one
                # pass here); the fp activations are NOT saved — backward
                # works entirely from the 1-bit planes (32x less read
                # traffic)
                if xp_pre is not None and mp_pre is not None:
                    xp, mp = xp_pre, mp_pre
                else:
                    xp, mp = nat.sign_mask_pack_nhwc(
                        x.contiguous(memory_format=torch.channels_last))
                ctx.packed = True
                ctx.x_dtype = x.dtype
                ctx.in_channels = x.shape[1]
                ctx.save_for_backward(w, xp, mp, wp, alpha)
            else:
                xp = (xp_pre if xp_pre is not None else nat.sign_pack_nhwc(
                    x.contiguous(memory_format=torch.channels_last)))
                ctx.save_for_backward(x, w)
            res = nat.xnor_conv_fwd(
                xp, wp, alpha, stab, x.shape[1], stride, padding,
                x.dtype == torch.bfloat16, want_stats)
            if want_stats:
                # per-channel (sum, sumsq) of the output, accumulated in
                # the conv epilogue — consumed by the fused BN (its stats
                # pass is skipped entirely)
                ctx.mark_non_differentiable(res[1], res[2])
                return res[0], res[1], res[2]
            dummy = res[0].new_empty(0)
            ctx.mark_non_differentiable(dummy)
            return res[0], dummy, dummy
        ctx.save_for_backward(x, w)
        xb = binsign(x)
        wb = weight_scale(w) * binsign(w)
        out = F.conv2d(xb, wb, None, stride=stride, padding=padding)
        dummy = out.new_empty(0)
        ctx.mark_non_differentiable(dummy)
        return out, dummy, dummy

    @staticmethod
    def backward(ctx, g, _gs1=None, _gs2=None):
        stride, padding = ctx.stride, ctx.padding
        acc = ctx.skip_cell.pop("g", None) if ctx.skip_cell else None
        if ctx.packed:
            w, xp, mp, wp, alpha = ctx.saved_tensors
            nat = _C.native_required()
            bf16 = g.dtype == torch.bfloat16
            C = ctx.in_channels
            K, kh = w.shape[0], w.shape[2]
            g = g.contiguous(memory_format=torch.channels_last)
            H, W = g.shape[2], g.shape[3]
            # hand-written MFMA dgrad (halo implicit GEMM, clip-STE mask
            # fused into the epilogue) — the default on the hot path
            use2 = (_MFMA_BWD and bf16 and ctx.x_dtype == torch.bfloat16
                    and stride == 1 and padding == 1 and kh == 3
                    and nat.dgrad2_supported(H, W, C, K))
            if use2:
                # dgrad: halo implicit GEMM, clip-STE mask (+ deferred
                # skip grad, if any) fused into the epilogue
                wd = nat.dgrad_weight_decode(wp, alpha, C)
                dx = nat.conv_dgrad2(g, wd, mp, C, acc)
                # wgrad: all-9-tap block GEMM straight from the sign
                # BITS (the dense +-1 activation tensor is never
                # materialized), then transpose + |w|<=1 mask in one pass
                xcp = nat.repack_cplane(xp, C, W)
                dwT = nat.conv_wgrad2(g, xcp, C)
                dw = nat.wgrad_finish(dwT, w.float())
                return (dx, dw.to(w.dtype), None, None, None, None, None,
                        None, None, None, None)
            xb = nat.decode_packed(xp, C, bf16)
            wb = nat.weight_decode(wp, alpha, C, bf16)
            dxb, dwb = torch.ops.aten.convolution_backward(
                g, xb, wb, None,
                [stride, stride], [padding, padding], [1, 1], False, [0, 0],
                1, [True, True, False])[:2]
            dx = nat.mask_mul_packed(dxb, mp, C,
                                     ctx.x_dtype == torch.bfloat16)
            if acc is not None:
                dx = dx + acc.to(dx.dtype)
            dw = nat.ste_mask_mul(dwb, w, 0, 0.0, 0.0)
            return (dx, dw.to(w.dtype), None, None, None, None, None,
                    None, None, None, None)
        x, w = ctx.saved_tensors
        if x.is_cuda:
            nat = _C.native_required()
            # decode +-1 operands in the compute dtype for the dense MFMA pass
            cdt = torch.bfloat16 if g.dtype == torch.bfloat16 else g.dtype
            xb = nat.binsign_decode(x, 1 if cdt == torch.bfloat16 else 0)
            wb = (weight_scale(w) * binsign(w)).to(cdt)
        else:
            xb = binsign(x)
            wb = weight_scale(w) * binsign(w)
        g = g.contiguous(memory_format=torch.channels_last) if g.is_cuda else g
        dxb, dwb = torch.ops.aten.convolution_backward(
            g, xb, wb, None,
            [stride, stride], [padding, padding], [1, 1], False, [0, 0], 1,
            [True, True, False])[:2]
        if x.is_cuda:
            nat = _C.native_required()
            mode_id = {"ste": 0, "approx": 1}[ctx.act_mode]
            if ctx.t is not None and ctx.k is not None:
                mode_id, t, k = 2, float(ctx.t), float(ctx.k)
            else:
                t, k = 0.0, 0.0
            dx = nat.ste_mask_mul(dxb, x, mode_id, t, k)
            dw = nat.ste_mask_mul(dwb, w, 0, 0.0, 0.0)
        else:
            dx = dxb * _act_grad_mask(x, ctx.act_mode, ctx.t, ctx.k)
            dw = dwb * (w.abs() <= 1).to(w.dtype)
        if acc is not None:
            dx = dx + acc.to(dx.dtype)
        return (dx, dw.to(w.dtype), None, None, None, None, None, None,
                None, None, None)


class _HardBinaryConvBase(nn.Module):
    """Shared implementation; subclasses fix the activation-STE default."""

    act_mode = "ste"

    def __init__(self, in_chn, out_chn, kernel_size=3, stride=1, padding=1):
        super().__init__()
        self.in_channels = in_chn
        self.out_channels = out_chn
        self.kernel_size = kernel_size
        self.stride = stride
        self.padding = padding
        self.weight = nn.Parameter(
            torch.empty(out_chn, in_chn, kernel_size, kernel_size))
        # ReActNet-style init scale for latent binary weights
        nn.init.normal_(self.weight, mean=0.0, std=0.1)
        # EDE schedule attrs, injected per-epoch by the engine (ref:train.py:412-415)
        self.t = None
        self.k = None

    def extra_repr(self):
        return (f"{self.in_channels}, {self.out_channels}, "
                f"kernel_size={self.kernel_size}, stride={self.stride}, "
                f"padding={self.padding}, act={self.act_mode}")

    def _apply(self, fn, recurse=True):
        # keep the latent weight NCHW-contiguous even when the model is
        # converted to channels_last: only our own kernels consume it
        # (pack expects NCHW) and a channels_last weight would force a
        # layout copy every step.
        super()._apply(fn, recurse)
        if not self.weight.data.is_contiguous():
            self.weight.data = self.weight.data.contiguous()
        return self

    def _check_prepack(self, x, prepack):
        """Validate a (xp, mp) pair packed by the producing BN's epilogue
        against this conv's input; returns (xp, mp) or (None, None)."""
        if prepack is None or not x.is_cuda:
            return None, None
        xp, mp = prepack
        if (xp.shape[0] == x.shape[0] and xp.shape[1] == x.shape[2]
                and xp.shape[2] == x.shape[3]
                and xp.shape[3] * 32 == self.in_channels):
            return xp, mp
        return None, None

    def forward(self, x, prepack=None, skip_cell=None):
        t = float(self.t) if self.t is not None else None
        k = float(self.k) if self.k is not None else None
        xp, mp = self._check_prepack(x, prepack)
        out, _, _ = BinaryConvFunction.apply(
            x, self.weight, self.stride, self.padding, self.act_mode, t, k,
            False, xp, mp, skip_cell)
        return out

    def forward_with_stats(self, x, prepack=None, skip_cell=None):
        """(out, (s1, s2)|None): per-out-channel sum/sumsq accumulated in
        the conv epilogue, for the fused BN that consumes the output.

        OFF by default, MEASURED twice (r1 flat-[K] atomics, r2 32-way
        sliced + LDS accumulation): any stats work in this epilogue
        costs the conv kernel 1.4-4x (profiles/, bn_bench.md r2) —
        far more than the separate bn_stats read pass it saves, which
        after the 4-deep MLP unroll costs ~0.07 ms/layer at b512.
        BDBNN_CONV_STATS=1 re-enables for experiments."""
        t = float(self.t) if self.t is not None else None
        k = float(self.k) if self.k is not None else None
        want = (x.is_cuda and self.training and _CONV_STATS)
        xp, mp = self._check_prepack(x, prepack)
        out, s1, s2 = BinaryConvFunction.apply(
            x, self.weight, self.stride, self.padding, self.act_mode, t, k,
            want, xp, mp, skip_cell)
        return out, ((s1, s2) if want else None)


class HardBinaryConv(_HardBinaryConvBase):
    """BD-BNN ImageNet binary conv (ref name: models.imagenet.resnet_bi_imagenet_set_2_2)."""
    act_mode = "ste"


class HardBinaryConv_react(_HardBinaryConvBase):
    """ReActNet-style binary conv (ref name: models.imagenet.resnet_bi_imagenet_set_2)."""
    act_mode = "approx"


class HardBinaryConv_cifar(_HardBinaryConvBase):
    """CIFAR binary conv (ref name: models.bin_module.binarized_modules)."""
    act_mode = "ste"
