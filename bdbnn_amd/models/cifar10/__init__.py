"""CIFAR model constructors — lowercase no-arg names, looked up via
``cifar_models.__dict__[arch]()`` (ref:train.py:50-52,283).

Binarized: resnet18 / resnet20 (1W/1A, HardBinaryConv_cifar).
Real-valued (teacher use): resnet18_real / resnet20_real.
num_classes is a kwarg so the same names serve cifar100.
"""

from ...ops.binary_conv import HardBinaryConv_cifar
from ..vgg_small import vgg_small
from ..resnet_common import (
    CifarResNet,
    BiBasicBlock,
    BasicBlock,
    resnet18_bi,
    _resnet,
)


def resnet20(num_classes=10):
    return CifarResNet(BiBasicBlock, 3, num_classes=num_classes, binary=True,
                       conv_cls=HardBinaryConv_cifar, act="prelu")


def resnet20_real(num_classes=10):
    return CifarResNet(BasicBlock, 3, num_classes=num_classes, binary=False)


def resnet18(num_classes=10):
    """Binarized ResNet-18 with a CIFAR stem (19 binary convs — matches the
    19-entry cifar diffkurt list, ref:train.py:472-475)."""
    m = resnet18_bi(num_classes=num_classes, conv_cls=HardBinaryConv_cifar,
                    act="prelu", stem="cifar")
    return m


def resnet18_real(num_classes=10):
    return _resnet([2, 2, 2, 2], binary=False, num_classes=num_classes,
                   stem="cifar")
