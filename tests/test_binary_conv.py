"""Binary conv module: oracle semantics on CPU (the GPU XNOR kernel is
tested against this exact composition in test_gpu_kernels.py)."""

import torch
import torch.nn.functional as F

from bdbnn_amd.ops.binarize import binsign, weight_scale
from bdbnn_amd.ops.binary_conv import (
    HardBinaryConv, HardBinaryConv_react, HardBinaryConv_cifar,
    BinaryConvFunction)


def _oracle(x, w, stride, padding):
    xb = binsign(x)
    wb = weight_scale(w) * binsign(w)
    return F.conv2d(xb, wb, None, stride=stride, padding=padding)


def test_forward_matches_oracle():
    torch.manual_seed(0)
    conv = HardBinaryConv(8, 16, 3, 1, 1)
    x = torch.randn(2, 8, 10, 10)
    out = conv(x)
    ref = _oracle(x, conv.weight.detach(), 1, 1)
    assert torch.allclose(out, ref, atol=1e-5)


def test_forward_stride2_1x1():
    torch.manual_seed(1)
    conv = HardBinaryConv(8, 16, 1, 2, 0)
    x = torch.randn(2, 8, 8, 8)
    out = conv(x)
    ref = _oracle(x, conv.weight.detach(), 2, 0)
    assert out.shape == (2, 16, 4, 4)
    assert torch.allclose(out, ref, atol=1e-5)


def test_backward_ste_masks():
    torch.manual_seed(2)
    conv = HardBinaryConv(4, 8, 3, 1, 1)
    with torch.no_grad():
        conv.weight.mul_(3.0)  # push some weights past |1| to exercise the mask
    x = (torch.randn(2, 4, 6, 6) * 2).requires_grad_(True)
    out = conv(x)
    g = torch.randn_like(out)
    out.backward(g)

    # reference: autograd through the composable ops
    x2 = x.detach().clone().requires_grad_(True)
    w2 = conv.weight.detach().clone().requires_grad_(True)
    xb = torch.clamp(x2, -1, 1)
    xb = (binsign(x2) - xb).detach() + xb  # sign with clip-STE
    alpha = w2.detach().abs().mean(dim=(1, 2, 3), keepdim=True)
    wc = torch.clamp(w2, -1, 1)
    wb = (alpha * binsign(w2) - wc).detach() + wc
    ref = F.conv2d(xb, wb, None, 1, 1)
    ref.backward(g)

    assert torch.allclose(x.grad, x2.grad, atol=1e-5)
    assert torch.allclose(conv.weight.grad, w2.grad, atol=1e-5)


def test_backward_ede_activation():
    conv = HardBinaryConv(4, 4, 3, 1, 1)
    conv.t, conv.k = 2.0, 1.0
    x = torch.randn(1, 4, 5, 5, requires_grad=True)
    out = conv(x)
    out.sum().backward()

    x2 = x.detach().clone().requires_grad_(True)
    conv2 = HardBinaryConv(4, 4, 3, 1, 1)
    with torch.no_grad():
        conv2.weight.copy_(conv.weight)
    out2, _, _ = BinaryConvFunction.apply(x2, conv2.weight, 1, 1, "ste",
                                          2.0, 1.0)
    out2.sum().backward()
    assert torch.allclose(x.grad, x2.grad, atol=1e-6)


def test_variants_act_modes():
    assert HardBinaryConv.act_mode == "ste"
    assert HardBinaryConv_react.act_mode == "approx"
    assert HardBinaryConv_cifar.act_mode == "ste"


def test_weight_is_4d_parameter():
    conv = HardBinaryConv_cifar(16, 32, 3, 1, 1)
    assert isinstance(conv.weight, torch.nn.Parameter)
    assert conv.weight.ndim == 4
