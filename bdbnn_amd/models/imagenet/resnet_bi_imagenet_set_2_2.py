"""BD-BNN ImageNet binarized ResNets (ref module name:
models.imagenet.resnet_bi_imagenet_set_2_2 — 'step 2' of the two-step
recipe with BD-BNN's block; exports HardBinaryConv, ref:KD_loss.py:6).
"""

import warnings

from ...ops.binary_conv import HardBinaryConv  # re-export (parity import site)
from ..resnet_common import resnet18_bi, resnet34_bi


def _check_pretrained(pretrained):
    if pretrained:
        warnings.warn("pretrained weights are not bundled (offline image); "
                      "returning random init")


def resnet18(pretrained=False, num_classes=1000):
    _check_pretrained(pretrained)
    return resnet18_bi(num_classes=num_classes, conv_cls=HardBinaryConv,
                       act="prelu")


def resnet34(pretrained=False, num_classes=1000):
    _check_pretrained(pretrained)
    return resnet34_bi(num_classes=num_classes, conv_cls=HardBinaryConv,
                       act="prelu")
