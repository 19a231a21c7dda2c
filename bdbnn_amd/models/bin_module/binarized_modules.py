"""Parity import site for models.bin_module.binarized_modules
(ref:train.py:32: ``from models.bin_module.binarized_modules import
HardBinaryConv_cifar``)."""

from ...ops.binary_conv import (
    HardBinaryConv,
    HardBinaryConv_react,
    HardBinaryConv_cifar,
)
from ...ops.binarize import BinaryActivation, LearnableBias
