"""Teacher->student distillation losses (ref:utils/KD_loss.py).

Semantics reproduced exactly:

* ``DistributionLoss`` (ref:KD_loss.py:10-43): logit KD
  loss = mean_n( -sum_c softmax(teacher)_nc * log_softmax(student)_nc )
  (no temperature; teacher output must not require grad).
* ``DistributionLoss_layer`` (ref:KD_loss.py:46-67): weight-space KD —
  for every matched (teacher, student) conv-module pair (excluding
  'module.conv1' and any 'downsample'), accumulate
  KLDivLoss(log_target=True)(W_s, W_t) = mean_e( exp(W_t) * (W_t - W_s) ).
  Raw weights are fed as if they were log-probabilities — that is the
  published behavior and we keep it.
* ``DistributionLoss_layer_cifar_act`` / ``loss_kd``: present-but-uncalled
  variants, kept for API parity.

The engine uses ``WeightKDLoss`` (pair-matching hoisted to setup, one
fused multi-tensor pass per step) instead of the per-step module walk
(ref walks named_modules of BOTH models every batch, KD_loss.py:59-64).
"""

import torch
import torch.nn as nn
import torch.nn.functional as F
from torch.nn.modules import loss as _loss

from .binary_conv import HardBinaryConv, HardBinaryConv_react
from .. import _C


class DistributionLoss(_loss._Loss):
    """Logit-level KD (ref:KD_loss.py:10-43); fused HIP kernel on GPU."""

    def forward(self, stud_output, teacher_output):
        if teacher_output.requires_grad:
            raise ValueError("real network output should not require gradients.")
        if (stud_output.is_cuda and stud_output.dim() == 2
                and _C.has_native()):
            from .losses import fused_logit_kd
            return fused_logit_kd(stud_output, teacher_output)
        log_p_s = F.log_softmax(stud_output, dim=1)
        p_t = F.softmax(teacher_output, dim=1)
        return -(p_t * log_p_s).sum(dim=1).mean()


def _layer_kd_pairs(model_stud, model_teacher):
    """Match same-named conv modules, excluding module.conv1 / downsample.

    Mirrors ref:KD_loss.py:59-64 (isinstance on Conv2d | HardBinaryConv |
    HardBinaryConv_react; name != 'module.conv1'; 'downsample' not in name).
    """
    stud_mods = dict(model_stud.named_modules())
    pairs = []
    for name, module in model_teacher.named_modules():
        if not isinstance(module, (torch.nn.Conv2d, HardBinaryConv, HardBinaryConv_react)):
            continue
        # ref excludes the wrapped stem name 'module.conv1'; we also exclude
        # the bare 'conv1' so unwrapped models keep the same semantics
        if name in ("module.conv1", "conv1") or "downsample" in name:
            continue
        m_s = stud_mods.get(name)
        if m_s is not None and hasattr(m_s, "weight"):
            pairs.append((m_s.weight, module.weight))
    return pairs


def _kl_log_target_mean(w_s, w_t):
    """KLDivLoss(reduction='mean', log_target=True)(input=w_s, target=w_t)."""
    return (torch.exp(w_t) * (w_t - w_s)).mean()


def _match_layout(t, like):
    """Return t with the same dense memory layout as `like`."""
    if t.stride() == like.stride():
        return t
    if like.dim() == 4 and like.is_contiguous(
            memory_format=torch.channels_last):
        return t.contiguous(memory_format=torch.channels_last)
    return t.contiguous()


class DistributionLoss_layer(_loss._Loss):
    """Weight-space KD via per-step module walk (API parity, ref:KD_loss.py:46-67)."""

    def forward(self, stud_output, teacher_output, model_stud, model_teacher, T=1):
        tot = 0
        for w_s, w_t in _layer_kd_pairs(model_stud, model_teacher):
            tot = tot + _kl_log_target_mean(w_s, w_t.detach())
        return tot


class DistributionLoss_layer_cifar_act(_loss._Loss):
    """Temperature variant (ref:KD_loss.py:69-87; never called upstream)."""

    def forward(self, stud_output, teacher_output, model_stud, model_teacher, T=6):
        tot = 0
        for w_s, w_t in _layer_kd_pairs(model_stud, model_teacher):
            tot = tot + F.kl_div(
                F.log_softmax(w_s / T, dim=1),
                F.softmax(w_t.detach() / T, dim=1),
                reduction="mean") * (T * T)
        return tot


def loss_kd(output, teacher_output, T=6):
    """Hinton logit KD with temperature (ref:KD_loss.py:90-100; uncalled upstream)."""
    return F.kl_div(
        F.log_softmax(output / T, dim=1),
        F.softmax(teacher_output / T, dim=1),
        reduction="mean") * (T * T)


class _FusedWeightKD(torch.autograd.Function):
    """sum over pairs of mean_e(exp(wt)*(wt - ws)); d/dws = -exp(wt)/n_e."""

    @staticmethod
    def forward(ctx, n_pairs, *tensors):
        ws = tensors[:n_pairs]
        wt = tensors[n_pairs:]
        if ws[0].is_cuda and _C.has_native():
            nat = _C.native_required()
            # elementwise pairing happens over physical memory: align the
            # teacher tensor's layout to the student's
            wt = tuple(_match_layout(b, a) for a, b in zip(ws, wt))
            out = nat.weight_kd_fwd(list(ws), list(wt))
            ctx.save_for_backward(*wt)
            ctx.native = True
            return out
        ctx.native = False
        ctx.save_for_backward(*wt)
        tot = ws[0].new_zeros(())
        for a, b in zip(ws, wt):
            tot = tot + (torch.exp(b) * (b - a)).mean()
        return tot

    @staticmethod
    def backward(ctx, g):
        wt = ctx.saved_tensors
        grads = []
        if ctx.native:
            nat = _C.native_required()
            # g stays a device scalar — no host sync mid-backward
            grads = list(nat.weight_kd_bwd(list(wt), g.detach()))
        else:
            for b in wt:
                grads.append(-g * torch.exp(b) / b.numel())
        return (None, *grads, *[None] * len(wt))


class WeightKDLoss(nn.Module):
    """Engine fast path for DistributionLoss_layer: pairs matched once."""

    def __init__(self, model_stud, model_teacher):
        super().__init__()
        self.pairs = _layer_kd_pairs(model_stud, model_teacher)
        if not self.pairs:
            raise ValueError("WeightKDLoss: no matched conv pairs")

    def forward(self):
        ws = [p[0] for p in self.pairs]
        wt = [p[1].detach() for p in self.pairs]
        return _FusedWeightKD.apply(len(ws), *ws, *wt)
