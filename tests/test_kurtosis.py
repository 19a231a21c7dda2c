"""Kurtosis regularizer: parity with the reference formula + fused op
gradient vs autograd."""

import torch

from bdbnn_amd.ops.kurtosis import (
    KurtosisWeight, RidgeRegularization, WeightRegularization,
    kurtosis_loss_fused)


def _ref_kurt(w):
    mean = torch.mean(w)
    std = torch.std(w)  # unbiased (ref:kurtosis.py:25)
    return torch.mean(((w - mean) / std) ** 4)


def test_kurtosis_weight_matches_reference_formula():
    torch.manual_seed(0)
    w = torch.randn(32, 16, 3, 3)
    kw = KurtosisWeight(w, "x", kurtosis_target=1.8)
    kw.fn_regularization()
    expected = (_ref_kurt(w) - 1.8) ** 2
    assert torch.allclose(kw.kurtosis_loss, expected)
    assert torch.allclose(kw.kurtosis, _ref_kurt(w))


def test_fused_loss_modes():
    torch.manual_seed(1)
    ws = [torch.randn(8, 4, 3, 3) for _ in range(3)]
    targets = [1.8, 1.4, 1.2]
    per_layer = [(_ref_kurt(w) - t) ** 2 for w, t in zip(ws, targets)]
    for mode, expected in (
            ("sum", sum(per_layer)),
            ("avg", sum(per_layer) / 3),
            ("max", torch.stack(per_layer).max())):
        loss, kurts = kurtosis_loss_fused(ws, targets, mode=mode)
        assert torch.allclose(loss, expected, atol=1e-5), mode
        assert kurts.shape == (3,)


def test_fused_loss_gradient_vs_autograd():
    torch.manual_seed(2)
    ws = [torch.randn(6, 4, 3, 3, requires_grad=True) for _ in range(2)]
    targets = [1.8, 1.4]
    loss, _ = kurtosis_loss_fused(ws, targets, mode="sum")
    loss.backward()
    analytic = [w.grad.clone() for w in ws]

    ws2 = [w.detach().clone().requires_grad_(True) for w in ws]
    ref = sum((_ref_kurt(w) - t) ** 2 for w, t in zip(ws2, targets))
    ref.backward()
    for a, w2 in zip(analytic, ws2):
        assert torch.allclose(a, w2.grad, atol=1e-4, rtol=1e-4)


def test_ridge_and_weight_regularization():
    w = torch.randn(4, 4, 3, 3)
    r = RidgeRegularization(w, "x")
    r.l2_regularization()
    assert torch.allclose(r.l2_loss, (w ** 2).sum())
    wr = WeightRegularization(w, "x")
    wr.w_regularization()
    assert torch.allclose(wr.wr_loss, torch.norm(w.abs() - 1, p=2))
