"""Fused classification losses (K4 logit-KD, K6 cross-entropy) over
(B, C) logits — csrc/loss.hip; exact reference semantics
(ref:utils/KD_loss.py:10-43, ref:train.py:318)."""

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import _C


class _LogitKDFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, s, t):
        nat = _C.native_required()
        out, stats, tc = nat.kd_logit_fwd(s, t)
        ctx.save_for_backward(s, tc, stats)
        return out

    @staticmethod
    def backward(ctx, g):
        s, tc, stats = ctx.saved_tensors
        ds = _C.native_required().kd_logit_bwd(s, tc, stats, 1.0)
        return ds * g, None


def fused_logit_kd(stud_logits, teacher_logits):
    """mean_n(-sum_c softmax(t) * log_softmax(s)); differentiable wrt s."""
    return _LogitKDFn.apply(stud_logits, teacher_logits)


class _CEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, s, y):
        nat = _C.native_required()
        out, stats = nat.ce_fwd(s, y)
        ctx.save_for_backward(s, y, stats)
        return out

    @staticmethod
    def backward(ctx, g):
        s, y, stats = ctx.saved_tensors
        ds = _C.native_required().ce_bwd(s, y, stats, 1.0)
        return ds * g, None


class FusedCrossEntropy(nn.Module):
    """nn.CrossEntropyLoss(mean) with a fused HIP kernel on the GPU."""

    def forward(self, logits, target):
        if (logits.is_cuda and logits.dim() == 2
                and target.dtype == torch.long and _C.has_native()):
            return _CEFn.apply(logits, target)
        return F.cross_entropy(logits, target)
