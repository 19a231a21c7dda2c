"""Drop-in import shim for the reference's models package: the exact
import sites of the reference (ref:train.py:27-32, ref:utils/KD_loss.py:6-7)
resolve against the MI355X-native implementations."""
import sys

from bdbnn_amd.models import cifar10, imagenet, bin_module

sys.modules[__name__ + ".cifar10"] = cifar10
sys.modules[__name__ + ".imagenet"] = imagenet
sys.modules[__name__ + ".bin_module"] = bin_module
sys.modules[__name__ + ".bin_module.binarized_modules"] = \
    bin_module.binarized_modules
sys.modules[__name__ + ".imagenet.resnet_bi_imagenet_set_2"] = \
    imagenet.resnet_bi_imagenet_set_2
sys.modules[__name__ + ".imagenet.resnet_bi_imagenet_set_2_2"] = \
    imagenet.resnet_bi_imagenet_set_2_2
