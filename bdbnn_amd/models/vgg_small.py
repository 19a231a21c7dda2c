"""Binarized VGG-Small (CIFAR), the second model family of the BD-BNN
paper lineage (XNOR-Net/IR-Net benchmarks).  First conv and the
classifier stay real-valued; the five inner convs are 1W/1A
(HardBinaryConv_cifar) with fused BN+PReLU tails."""

import torch
import torch.nn as nn

from ..ops.binary_conv import HardBinaryConv_cifar
from ..ops.activations import ChannelPReLU
from ..ops.bn_act import fused_bn_act


class _BinConvBlock(nn.Module):
    def __init__(self, cin, cout, pool=False):
        super().__init__()
        self.conv = HardBinaryConv_cifar(cin, cout, 3, 1, 1)
        self.bn = nn.BatchNorm2d(cout)
        self.act = ChannelPReLU(cout)
        self.pool = nn.MaxPool2d(2) if pool else nn.Identity()
        self.has_pool = pool

    def forward(self, x):
        # accept (x, pack) from the previous block's fused BN epilogue
        pk_in = None
        if isinstance(x, tuple):
            x, pk_in = x
        out, stats = self.conv.forward_with_stats(x, prepack=pk_in)
        if self.has_pool:
            # pooled output != BN output: a pack would be stale
            return self.pool(fused_bn_act(out, self.bn, self.act,
                                          stats=stats))
        y, pk = fused_bn_act(out, self.bn, self.act, stats=stats, pack=True)
        return (y, pk) if pk is not None else y


class VGGSmall(nn.Module):
    def __init__(self, num_classes=10):
        super().__init__()
        self.conv0 = nn.Conv2d(3, 128, 3, 1, 1, bias=False)  # real stem
        self.bn0 = nn.BatchNorm2d(128)
        self.block1 = _BinConvBlock(128, 128, pool=True)
        self.block2 = _BinConvBlock(128, 256)
        self.block3 = _BinConvBlock(256, 256, pool=True)
        self.block4 = _BinConvBlock(256, 512)
        self.block5 = _BinConvBlock(512, 512, pool=True)
        self.fc = nn.Linear(512 * 4 * 4, num_classes)

    def forward(self, x):
        x = fused_bn_act(self.conv0(x), self.bn0, "relu")
        x = self.block1(x)
        x = self.block2(x)
        x = self.block3(x)
        x = self.block4(x)
        x = self.block5(x)
        x = torch.flatten(x, 1)
        return self.fc(x)


def vgg_small(num_classes=10):
    return VGGSmall(num_classes=num_classes)
