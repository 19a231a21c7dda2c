"""ReActNet-style ImageNet binarized ResNets (ref module name:
models.imagenet.resnet_bi_imagenet_set_2; exports HardBinaryConv_react,
ref:KD_loss.py:7).  RPReLU activations + polynomial activation STE.
"""

import warnings

from ...ops.binary_conv import HardBinaryConv_react  # re-export (parity import site)
from ..resnet_common import resnet18_bi, resnet34_bi


def _check_pretrained(pretrained):
    if pretrained:
        warnings.warn("pretrained weights are not bundled (offline image); "
                      "returning random init")


def resnet18_react(pretrained=False, num_classes=1000):
    _check_pretrained(pretrained)
    return resnet18_bi(num_classes=num_classes,
                       conv_cls=HardBinaryConv_react, act="rprelu")


def resnet34_react(pretrained=False, num_classes=1000):
    _check_pretrained(pretrained)
    return resnet34_bi(num_classes=num_classes,
                       conv_cls=HardBinaryConv_react, act="rprelu")
