"""Fused BatchNorm (+ residual add) (+ activation) — K8.

``fused_bn_act(x, bn, act=None, skip=None)`` runs the whole
conv-output -> BN -> (+skip) -> PReLU/ReLU tail of a BD-BNN block as two
kernels forward (stats + normalize/add/act) and two backward (reduce +
apply) on the GPU, replacing the MIOpen BN pipeline + separate add +
activation (profiles/r01_bench_b256_kernel_stats.md).  Parameters stay
in the ordinary ``nn.BatchNorm2d`` / ``ChannelPReLU`` modules, so
checkpoints are unchanged.

CPU (and any non-channels_last corner): exact composition fallback.
"""

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import _C
from .activations import ChannelPReLU

_ACT_NONE, _ACT_PRELU, _ACT_RELU = 0, 1, 2


class _BNActFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, skip, gamma, beta, a, running_mean, running_var,
                momentum, eps, act_kind, training, s1=None, s2=None,
                want_pack=False, defer_cell=None):
        nat = _C.native_required()
        res = nat.bn_act_fwd_train(
            x, skip, gamma, beta, a, running_mean, running_var,
            momentum, eps, act_kind, s1, s2, want_pack)
        out, z, mean, invstd = res[:4]
        ctx.save_for_backward(x, z, mean, invstd, gamma,
                              a if a is not None else torch.empty(0))
        ctx.act_kind = act_kind
        ctx.has_skip = skip is not None
        ctx.has_a = a is not None
        # deferred skip grad: instead of returning dskip (which autograd
        # would add to the conv's dx in a separate full-tensor pass), the
        # backward stashes it in this cell; the conv that produced the
        # skip tensor's other consumer fuses it into its dgrad epilogue.
        ctx.defer_cell = defer_cell if skip is not None else None
        if want_pack:
            # next conv's sign/mask bitplanes, packed in the epilogue
            xpk, mpk = res[4], res[5]
            ctx.mark_non_differentiable(xpk, mpk)
            return out, xpk, mpk
        dummy = out.new_empty(0)
        ctx.mark_non_differentiable(dummy)
        return out, dummy, dummy

    @staticmethod
    def backward(ctx, dy, _gxpk=None, _gmpk=None):
        x, z, mean, invstd, gamma, a = ctx.saved_tensors
        nat = _C.native_required()
        dx, dskip, dgamma, dbeta, da = nat.bn_act_bwd(
            dy, z, x, mean, invstd, gamma,
            a if ctx.has_a else None, ctx.act_kind, ctx.has_skip)
        skip_grad = dskip if ctx.has_skip else None
        if ctx.defer_cell is not None and skip_grad is not None:
            ctx.defer_cell["g"] = skip_grad
            skip_grad = None
        return (dx, skip_grad,
                dgamma, dbeta,
                da if ctx.has_a else None,
                None, None, None, None, None, None, None, None, None,
                None)


def fused_bn_act(x, bn: nn.BatchNorm2d, act=None, skip=None, stats=None,
                 pack=False, defer_skip_cell=None):
    """BN(x) (+skip) then act.  act: None | ChannelPReLU | 'relu'.

    stats: optional (sum, sumsq) per channel of x, pre-accumulated by the
    producing conv's epilogue — skips BN's own stats read pass.

    pack=True: also return the consuming binary conv's (sign, mask)
    bitplanes, packed in the epilogue (the conv then skips its own pack
    read pass).  Returns (out, (xp, mp) | None); only the fused training
    path with C % 32 == 0 produces a pack — callers must handle None."""
    if isinstance(act, ChannelPReLU):
        act_kind, a = _ACT_PRELU, act.weight
    elif act == "relu":
        act_kind, a = _ACT_RELU, None
    elif act is None:
        act_kind, a = _ACT_NONE, None
    else:  # generic module: apply unfused after BN+add
        out = fused_bn_act(x, bn, None, skip, stats=stats,
                           defer_skip_cell=defer_skip_cell)
        out = act(out)
        return (out, None) if pack else out

    C = x.size(1) if x.dim() == 4 else 0
    use_fused = (x.is_cuda and x.dim() == 4 and 0 < C <= 1024
                 and (C & (C - 1)) == 0 and _C.has_native())
    if use_fused and bn.training:
        if bn.track_running_stats and bn.num_batches_tracked is not None:
            bn.num_batches_tracked.add_(1)
        s1, s2 = stats if stats is not None else (None, None)
        want_pack = bool(pack) and C % 32 == 0
        out, xpk, mpk = _BNActFn.apply(
            x, skip, bn.weight, bn.bias, a, bn.running_mean, bn.running_var,
            bn.momentum if bn.momentum is not None else 0.1, bn.eps,
            act_kind, True, s1, s2, want_pack, defer_skip_cell)
        if pack:
            return out, ((xpk, mpk) if want_pack else None)
        return out
    if use_fused and not torch.is_grad_enabled() \
            and bn.running_mean is not None:
        nat = _C.native_required()
        out = nat.bn_act_eval(x, skip, bn.weight, bn.bias, a,
                              bn.running_mean, bn.running_var, bn.eps,
                              act_kind)
        return (out, None) if pack else out

    # composition fallback (CPU / oracle)
    z = bn(x)
    if skip is not None:
        z = z + skip
    if act_kind == _ACT_PRELU:
        out = act(z)
    elif act_kind == _ACT_RELU:
        out = F.relu(z)
    else:
        out = z
    return (out, None) if pack else out
