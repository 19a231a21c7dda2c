"""ResNet skeletons (real-valued and binarized) with torchvision-compatible
module naming.

The weight-space KD (ops/kd.py, ref:utils/KD_loss.py:59-64) matches
teacher and student conv modules BY NAME and needs identical weight
shapes, so both the fp32 teacher and the 1-bit student mirror the
torchvision ResNet layout and names exactly:
conv1 / bn1 / layer{1..4}.{i}.conv1|bn1|conv2|bn2|downsample.0|downsample.1
/ avgpool / fc.  (torchvision itself is not installed in this image;
these are from-scratch implementations.)

Binarized blocks follow the Bi-Real/ReActNet recipe: every binary conv
carries its own residual connection and a PReLU-family activation;
first conv and final fc stay real (ref contract).
ResNet-18 has 19 convs besides the stem (16 block + 3 downsample) —
matching the 19-entry per-layer kurtosis target lists
(ref:train.py:467-470).
"""

import os

import torch
import torch.nn as nn

from ..ops.binary_conv import HardBinaryConv
from ..ops.stem_conv import StemConv7x7
from ..ops.binarize import LearnableBias
from ..ops.activations import ChannelPReLU
from ..ops.bn_act import fused_bn_act
from ..ops.pool import FusedMaxPool2d


# defer each bn's residual-skip gradient into the paired conv's dgrad
# epilogue (kills autograd's separate dx+dskip accumulation pass,
# ~3.3 ms/step at b2048); see ops/bn_act.py / ops/binary_conv.py
_FUSE_SKIP_GRAD = os.environ.get("BDBNN_FUSE_SKIP_GRAD", "1") == "1"


class FusedDownsample(nn.Sequential):
    """conv + BN downsample with the fused BN kernel (names '0'/'1' kept)."""

    def forward(self, x, prepack=None):
        if hasattr(self[0], "forward_with_stats"):
            out, stats = self[0].forward_with_stats(x, prepack=prepack)
            return fused_bn_act(out, self[1], stats=stats)
        return fused_bn_act(self[0](x), self[1])


class RPReLU(nn.Module):
    """ReActNet RPReLU: shift -> PReLU -> shift, per channel."""

    def __init__(self, channels):
        super().__init__()
        self.move1 = LearnableBias(channels)
        self.prelu = ChannelPReLU(channels)
        self.move2 = LearnableBias(channels)

    def forward(self, x):
        return self.move2(self.prelu(self.move1(x)))


def _make_act(kind, channels):
    if kind == "relu":
        return nn.ReLU(inplace=True)
    if kind == "prelu":
        return ChannelPReLU(channels)
    if kind == "rprelu":
        return RPReLU(channels)
    raise ValueError(kind)


class BasicBlock(nn.Module):
    """Real-valued torchvision-style BasicBlock."""

    expansion = 1

    def __init__(self, inplanes, planes, stride=1, downsample=None):
        super().__init__()
        self.conv1 = nn.Conv2d(inplanes, planes, 3, stride, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(planes)
        self.relu = nn.ReLU(inplace=True)
        self.conv2 = nn.Conv2d(planes, planes, 3, 1, 1, bias=False)
        self.bn2 = nn.BatchNorm2d(planes)
        self.downsample = downsample
        self.stride = stride

    def forward(self, x):
        identity = self.downsample(x) if self.downsample is not None else x
        out = fused_bn_act(self.conv1(x), self.bn1, "relu")
        out = fused_bn_act(self.conv2(out), self.bn2, "relu", skip=identity)
        return out


class BiBasicBlock(nn.Module):
    """Binarized BasicBlock: per-conv residual (Bi-Real) + PReLU/RPReLU.

    Module names (conv1/bn1/conv2/bn2/downsample.0/.1) match the real
    block so weight-KD pairs by name with identical shapes.
    """

    expansion = 1

    def __init__(self, inplanes, planes, stride=1, downsample=None,
                 conv_cls=HardBinaryConv, act="prelu"):
        super().__init__()
        self.conv1 = conv_cls(inplanes, planes, 3, stride, 1)
        self.bn1 = nn.BatchNorm2d(planes)
        self.conv2 = conv_cls(planes, planes, 3, 1, 1)
        self.bn2 = nn.BatchNorm2d(planes)
        self.downsample = downsample
        self.stride = stride
        self.act1 = _make_act(act, planes)
        self.act2 = _make_act(act, planes)

    def forward(self, x):
        # accept (x, pack): the previous block's bn2 epilogue packed this
        # block's input bitplanes (conv1 AND the downsample conv share x)
        pk_in = None
        if isinstance(x, tuple):
            x, pk_in = x
        # Each bn's skip tensor is also the paired conv's input, so
        # autograd would sum the two gradient paths in a separate
        # full-tensor add; a shared per-call cell defers the bn's dskip
        # into the conv's dgrad epilogue instead (BDBNN_FUSE_SKIP_GRAD=0
        # restores plain autograd accumulation for A/B).
        defer = _FUSE_SKIP_GRAD and self.training and x.is_cuda
        if self.downsample is not None:
            identity = self.downsample(x, prepack=pk_in)
            cell1 = None  # x's grad paths are conv1 + downsample, not bn1
        else:
            identity = x
            cell1 = {} if defer else None
        o1, st1 = self.conv1.forward_with_stats(x, prepack=pk_in,
                                                skip_cell=cell1)
        # only a plain ChannelPReLU tail writes the exact conv2 input in
        # its epilogue (RPReLU's shifts run after, so no pack there)
        fuse1 = isinstance(self.act1, ChannelPReLU)
        r1 = fused_bn_act(o1, self.bn1, self.act1, skip=identity, stats=st1,
                          pack=fuse1, defer_skip_cell=cell1)
        out, pk1 = r1 if fuse1 else (r1, None)
        cell2 = {} if defer else None
        o2, st2 = self.conv2.forward_with_stats(out, prepack=pk1,
                                                skip_cell=cell2)
        fuse2 = isinstance(self.act2, ChannelPReLU)
        r2 = fused_bn_act(o2, self.bn2, self.act2, skip=out, stats=st2,
                          pack=fuse2, defer_skip_cell=cell2)
        out, pk2 = r2 if fuse2 else (r2, None)
        return (out, pk2) if pk2 is not None else out


class ResNet(nn.Module):
    """Shared trunk.  stem='imagenet' (7x7/2 + maxpool) or 'cifar' (3x3/1)."""

    def __init__(self, block_fn, layers, num_classes=1000, stem="imagenet",
                 width=64, binary=False, conv_cls=None, act="prelu"):
        super().__init__()
        self.binary = binary
        self.inplanes = width
        if stem == "imagenet":
            conv_t = StemConv7x7 if width == 64 else nn.Conv2d
            self.conv1 = conv_t(3, width, 7, 2, 3, bias=False)
        else:
            self.conv1 = nn.Conv2d(3, width, 3, 1, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(width)
        self.relu = nn.ReLU(inplace=True)
        self.maxpool = (FusedMaxPool2d(3, 2, 1) if stem == "imagenet"
                        else nn.Identity())
        stages = [width * (2 ** i) for i in range(len(layers))]
        strides = [1] + [2] * (len(layers) - 1)
        for i, (planes, blocks, stride) in enumerate(zip(stages, layers, strides), 1):
            setattr(self, f"layer{i}",
                    self._make_layer(block_fn, planes, blocks, stride,
                                     conv_cls=conv_cls, act=act))
        self.avgpool = nn.AdaptiveAvgPool2d((1, 1))
        self.fc = nn.Linear(stages[-1], num_classes)

        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out", nonlinearity="relu")
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.constant_(m.weight, 1)
                nn.init.constant_(m.bias, 0)

    def _make_layer(self, block_fn, planes, blocks, stride, conv_cls, act):
        downsample = None
        if stride != 1 or self.inplanes != planes:
            if self.binary:
                ds_conv = conv_cls(self.inplanes, planes, 1, stride, 0)
            else:
                ds_conv = nn.Conv2d(self.inplanes, planes, 1, stride, bias=False)
            downsample = FusedDownsample(ds_conv, nn.BatchNorm2d(planes))
        layers = []
        kw = dict(conv_cls=conv_cls, act=act) if self.binary else {}
        layers.append(block_fn(self.inplanes, planes, stride, downsample, **kw))
        self.inplanes = planes
        for _ in range(1, blocks):
            layers.append(block_fn(self.inplanes, planes, **kw))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.maxpool(fused_bn_act(self.conv1(x), self.bn1, "relu"))
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        x = self.layer4(x) if hasattr(self, "layer4") else x
        if isinstance(x, tuple):   # drop the last block's pack hand-off
            x = x[0]
        x = self.avgpool(x)
        x = torch.flatten(x, 1)
        return self.fc(x)


def _resnet(layers, binary, conv_cls=None, act="prelu", **kw):
    if binary:
        return ResNet(BiBasicBlock, layers, binary=True,
                      conv_cls=conv_cls, act=act, **kw)
    return ResNet(BasicBlock, layers, binary=False, **kw)


# ---------------- ImageNet constructors ----------------

def _warn_no_pretrained(name, pretrained):
    if pretrained:
        import logging
        logging.getLogger("bdbnn").warning(
            "%s(pretrained=True): no pretrained weights ship in this "
            "offline image — the model is RANDOM-INIT; load real weights "
            "with --resume_teacher <ckpt> (the reference loads a "
            "pretrained torchvision teacher, ref:train.py:252-257)", name)


def resnet18_real(pretrained=False, num_classes=1000, stem="imagenet"):
    _warn_no_pretrained("resnet18_real", pretrained)
    return _resnet([2, 2, 2, 2], binary=False, num_classes=num_classes,
                   stem=stem)


def resnet34_real(pretrained=False, num_classes=1000, stem="imagenet"):
    _warn_no_pretrained("resnet34_real", pretrained)
    return _resnet([3, 4, 6, 3], binary=False, num_classes=num_classes,
                   stem=stem)


def resnet18_bi(pretrained=False, num_classes=1000, conv_cls=HardBinaryConv,
                act="prelu", stem="imagenet"):
    return _resnet([2, 2, 2, 2], binary=True, conv_cls=conv_cls, act=act,
                   num_classes=num_classes, stem=stem)


def resnet34_bi(pretrained=False, num_classes=1000, conv_cls=HardBinaryConv,
                act="prelu", stem="imagenet"):
    return _resnet([3, 4, 6, 3], binary=True, conv_cls=conv_cls, act=act,
                   num_classes=num_classes, stem=stem)


# ---------------- CIFAR constructors ----------------

class CifarResNet(nn.Module):
    """3-stage CIFAR ResNet (16/32/64 channels), real or binary."""

    def __init__(self, block_fn, n_per_stage, num_classes=10, binary=False,
                 conv_cls=None, act="prelu"):
        super().__init__()
        self.binary = binary
        self.inplanes = 16
        self.conv1 = nn.Conv2d(3, 16, 3, 1, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(16)
        self.relu = nn.ReLU(inplace=True)
        kw = dict(conv_cls=conv_cls, act=act) if binary else {}
        self.layer1 = self._make_layer(block_fn, 16, n_per_stage, 1, kw)
        self.layer2 = self._make_layer(block_fn, 32, n_per_stage, 2, kw)
        self.layer3 = self._make_layer(block_fn, 64, n_per_stage, 2, kw)
        self.avgpool = nn.AdaptiveAvgPool2d((1, 1))
        self.fc = nn.Linear(64, num_classes)

    def _make_layer(self, block_fn, planes, blocks, stride, kw):
        downsample = None
        if stride != 1 or self.inplanes != planes:
            if self.binary:
                ds_conv = kw["conv_cls"](self.inplanes, planes, 1, stride, 0)
            else:
                ds_conv = nn.Conv2d(self.inplanes, planes, 1, stride, bias=False)
            downsample = FusedDownsample(ds_conv, nn.BatchNorm2d(planes))
        layers = [block_fn(self.inplanes, planes, stride, downsample, **kw)]
        self.inplanes = planes
        for _ in range(1, blocks):
            layers.append(block_fn(self.inplanes, planes, **kw))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = fused_bn_act(self.conv1(x), self.bn1, "relu")
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        if isinstance(x, tuple):   # drop the last block's pack hand-off
            x = x[0]
        x = self.avgpool(x)
        x = torch.flatten(x, 1)
        return self.fc(x)
