"""Model-zoo contracts: constructor namespaces, binary-conv counts,
teacher/student name+shape matching for weight-KD."""

import torch
import pytest

from bdbnn_amd import models
from bdbnn_amd.ops.binary_conv import _HardBinaryConvBase


def _binary_convs(m):
    return [mod for mod in m.modules() if isinstance(mod, _HardBinaryConvBase)]


def test_cifar_namespace_noarg_ctors():
    for name in ("resnet20", "resnet18", "resnet20_real", "resnet18_real"):
        ctor = models.cifar10.__dict__[name]
        m = ctor()
        out = m(torch.randn(2, 3, 32, 32))
        assert out.shape == (2, 10)


def test_imagenet_namespace_pretrained_arg():
    for name in ("resnet18", "resnet34", "resnet18_react", "resnet18_real"):
        ctor = models.imagenet.__dict__[name]
        m = ctor(False)
        assert m(torch.randn(1, 3, 64, 64)).shape == (1, 1000)


def test_resnet18_has_19_binary_convs():
    # 16 block convs + 3 binarized downsamples; stem + fc real
    m = models.imagenet.resnet18(False)
    assert len(_binary_convs(m)) == 19
    assert isinstance(m.conv1, torch.nn.Conv2d)
    assert isinstance(m.fc, torch.nn.Linear)


def test_resnet34_has_35_binary_convs():
    m = models.imagenet.resnet34(False)
    # 32 block convs + 3 downsamples
    assert len(_binary_convs(m)) == 35


def test_parity_import_sites():
    from bdbnn_amd.models.imagenet.resnet_bi_imagenet_set_2_2 import HardBinaryConv
    from bdbnn_amd.models.imagenet.resnet_bi_imagenet_set_2 import HardBinaryConv_react
    from bdbnn_amd.models.bin_module.binarized_modules import HardBinaryConv_cifar
    assert issubclass(HardBinaryConv, _HardBinaryConvBase)
    assert issubclass(HardBinaryConv_react, _HardBinaryConvBase)
    assert issubclass(HardBinaryConv_cifar, _HardBinaryConvBase)


def test_student_teacher_shapes_match_by_name():
    s = models.imagenet.resnet18(False)
    t = models.imagenet.resnet18_real(False)
    t_convs = {n: m.weight.shape for n, m in t.named_modules()
               if isinstance(m, torch.nn.Conv2d)}
    s_convs = {n: m.weight.shape for n, m in s.named_modules()
               if isinstance(m, (torch.nn.Conv2d, _HardBinaryConvBase))}
    assert set(t_convs) == set(s_convs)
    for n in t_convs:
        assert t_convs[n] == s_convs[n], n


def test_backward_through_full_model():
    m = models.imagenet.resnet18(False)
    x = torch.randn(2, 3, 64, 64)
    loss = m(x).sum()
    loss.backward()
    for mod in _binary_convs(m):
        assert mod.weight.grad is not None


def test_react_variant_uses_rprelu():
    from bdbnn_amd.models.resnet_common import RPReLU
    m = models.imagenet.resnet18_react(False)
    assert any(isinstance(mod, RPReLU) for mod in m.modules())


def test_vgg_small():
    m = models.cifar10.vgg_small()
    out = m(torch.randn(2, 3, 32, 32))
    assert out.shape == (2, 10)
    out.sum().backward()
    assert len(_binary_convs(m)) == 5
