"""Quantizer semantics: sign forward, STE/EDE/polynomial backward
(the oracle the HIP kernels are tested against)."""

import pytest
import torch

from bdbnn_amd.ops.binarize import (
    binsign, SignSTE, SignEDE, SignApprox, binarize_weight, weight_scale,
    BinaryActivation, LearnableBias)


def test_binsign_zero_maps_to_plus_one():
    x = torch.tensor([-2.0, -0.0, 0.0, 0.5, 3.0])
    out = binsign(x)
    assert out.tolist() == [-1.0, 1.0, 1.0, 1.0, 1.0]


def test_sign_ste_backward_clip():
    x = torch.tensor([-2.0, -1.0, -0.5, 0.0, 0.5, 1.0, 2.0],
                     requires_grad=True)
    y = SignSTE.apply(x)
    y.backward(torch.ones_like(y))
    assert x.grad.tolist() == [0.0, 1.0, 1.0, 1.0, 1.0, 1.0, 0.0]


def test_sign_ede_backward_formula():
    t, k = 2.0, 1.5
    x = torch.randn(64, requires_grad=True)
    y = SignEDE.apply(x, t, k)
    g = torch.randn(64)
    y.backward(g)
    th = torch.tanh(t * x.detach())
    expected = g * k * t * (1 - th * th)
    assert torch.allclose(x.grad, expected, atol=1e-6)


def test_sign_approx_backward_poly():
    x = torch.tensor([-1.5, -1.0, -0.5, 0.0, 0.5, 0.99, 1.0],
                     requires_grad=True)
    y = SignApprox.apply(x)
    y.backward(torch.ones_like(y))
    expected = [0.0, 0.0, 1.0, 2.0, 1.0, 2 - 2 * 0.99, 0.0]
    assert torch.allclose(x.grad, torch.tensor(expected), atol=1e-6)


def test_weight_scale_per_out_channel():
    w = torch.randn(8, 4, 3, 3)
    a = weight_scale(w)
    assert a.shape == (8, 1, 1, 1)
    assert torch.allclose(a.flatten(), w.abs().mean(dim=(1, 2, 3)))


def test_binarize_weight_forward_and_ste():
    w = torch.randn(6, 3, 3, 3) * 2
    w.requires_grad_(True)
    wb = binarize_weight(w)
    alpha = w.detach().abs().mean(dim=(1, 2, 3), keepdim=True)
    assert torch.allclose(wb.detach(), alpha * binsign(w.detach()))
    g = torch.randn_like(wb)
    wb.backward(g)
    expected = g * (w.detach().abs() <= 1).float()
    assert torch.allclose(w.grad, expected)


def test_binary_activation_ede_injection_overrides():
    act = BinaryActivation(mode="approx")
    act.t, act.k = 0.5, 2.0
    x = torch.randn(16, requires_grad=True)
    y = act(x)
    g = torch.ones(16)
    y.backward(g)
    th = torch.tanh(0.5 * x.detach())
    assert torch.allclose(x.grad, 2.0 * 0.5 * (1 - th * th), atol=1e-6)


def test_learnable_bias_shape():
    lb = LearnableBias(8)
    x = torch.randn(2, 8, 4, 4)
    assert lb(x).shape == x.shape
    assert lb.bias.shape == (1, 8, 1, 1)


def test_channel_prelu_cpu_matches_torch():
    from bdbnn_amd.ops.activations import ChannelPReLU
    torch.manual_seed(9)
    m = ChannelPReLU(8)
    with torch.no_grad():
        m.weight.uniform_(-0.5, 0.5)
    ref = torch.nn.PReLU(8)
    with torch.no_grad():
        ref.weight.copy_(m.weight)
    x = torch.randn(2, 8, 5, 5, requires_grad=True)
    x2 = x.detach().clone().requires_grad_(True)
    out = m(x); out2 = ref(x2)
    assert torch.allclose(out, out2)
    g = torch.randn_like(out)
    out.backward(g); out2.backward(g)
    assert torch.allclose(x.grad, x2.grad, atol=1e-6)
    assert torch.allclose(m.weight.grad, ref.weight.grad, atol=1e-5)
