"""KD losses: reference-formula parity + fused weight-KD path."""

import torch
import torch.nn.functional as F
import pytest

from bdbnn_amd.ops.kd import (
    DistributionLoss, DistributionLoss_layer, loss_kd, WeightKDLoss,
    _kl_log_target_mean)
from bdbnn_amd.models import imagenet as im
from bdbnn_amd.engine.trainer import _Wrapped


def test_distribution_loss_formula():
    torch.manual_seed(0)
    s = torch.randn(8, 10, requires_grad=True)
    t = torch.randn(8, 10)
    loss = DistributionLoss()(s, t)
    expected = -(F.softmax(t, 1) * F.log_softmax(s, 1)).sum(1).mean()
    assert torch.allclose(loss, expected, atol=1e-6)
    loss.backward()  # differentiable wrt student


def test_distribution_loss_rejects_grad_teacher():
    s = torch.randn(4, 10)
    t = torch.randn(4, 10, requires_grad=True)
    with pytest.raises(ValueError):
        DistributionLoss()(s, t)


def test_kl_log_target_matches_torch():
    a = torch.randn(20)
    b = torch.randn(20)
    ours = _kl_log_target_mean(a, b)
    ref = torch.nn.KLDivLoss(log_target=True)(a, b)
    assert torch.allclose(ours, ref, atol=1e-6)


def test_loss_kd_temperature():
    s = torch.randn(4, 10)
    t = torch.randn(4, 10)
    ref = F.kl_div(F.log_softmax(s / 6, 1), F.softmax(t / 6, 1),
                   reduction="mean") * 36
    assert torch.allclose(loss_kd(s, t, T=6), ref, atol=1e-6)


def test_layer_kd_pairs_and_fused_equivalence():
    torch.manual_seed(1)
    student = _Wrapped(im.resnet18(False))
    teacher = _Wrapped(im.resnet18_real(False))
    walk = DistributionLoss_layer()(None, None, student, teacher)
    fused = WeightKDLoss(student, teacher)()
    assert torch.allclose(walk, fused, atol=1e-5)
    # 16 block convs (19 - 3 downsample), stem excluded
    assert len(WeightKDLoss(student, teacher).pairs) == 16


def test_fused_weight_kd_gradient():
    torch.manual_seed(2)
    student = _Wrapped(im.resnet18(False))
    teacher = _Wrapped(im.resnet18_real(False))
    kd = WeightKDLoss(student, teacher)
    loss = kd()
    loss.backward()
    ws, wt = kd.pairs[0]
    expected = -torch.exp(wt.detach()) / wt.numel()
    assert torch.allclose(ws.grad, expected, atol=1e-6)
