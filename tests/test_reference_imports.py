"""The reference's EXACT import lines (ref:train.py:27-44,
ref:utils/KD_loss.py:6-7) must work against this repo unchanged."""


def test_reference_train_py_imports():
    from models.imagenet.resnet_bi_imagenet_set_2 import HardBinaryConv_react  # noqa
    from models.imagenet.resnet_bi_imagenet_set_2_2 import HardBinaryConv  # noqa
    from models.bin_module.binarized_modules import HardBinaryConv_cifar  # noqa
    import models.cifar10 as cifar_models
    import models.imagenet as imagenet_models
    from utils.KD_loss import DistributionLoss, DistributionLoss_layer  # noqa
    from utils.utils import cpt_tk, save_checkpoint, AverageMeter, accuracy  # noqa
    from kurtosis import KurtosisWeight, RidgeRegularization, WeightRegularization  # noqa
    from loader import dataloader_cifar10, dataloader_cifar100, dataloader_imagenet  # noqa

    # constructor lookup pattern of ref:train.py:283,285
    m = cifar_models.__dict__["resnet20"]()
    assert m is not None
    m = imagenet_models.__dict__["resnet18"](False)
    assert m is not None


def test_shimmed_classes_are_the_native_ones():
    from models.imagenet.resnet_bi_imagenet_set_2_2 import HardBinaryConv
    from bdbnn_amd.ops.binary_conv import HardBinaryConv as Native
    assert HardBinaryConv is Native
