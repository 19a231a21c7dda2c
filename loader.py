"""Drop-in import shim for the reference's loader module
(ref:loader.py): same entry points, MI355X-native pipeline underneath."""
from bdbnn_amd.data import (  # noqa: F401
    dataloader_cifar10,
    dataloader_cifar100,
    dataloader_imagenet,
)
