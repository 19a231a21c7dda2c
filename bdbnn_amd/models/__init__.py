"""Model namespaces, mirroring the reference's (missing) models package:

* ``models.cifar10``   — lowercase no-arg constructors (ref:train.py:283)
* ``models.imagenet``  — constructors taking ``pretrained`` (ref:train.py:285)
* ``models.bin_module.binarized_modules`` — HardBinaryConv_cifar
  (ref:train.py:32)
* ``models.imagenet.resnet_bi_imagenet_set_2`` / ``..._set_2_2`` —
  HardBinaryConv_react / HardBinaryConv (ref:train.py:30-31,
  ref:utils/KD_loss.py:6-7)
"""

from . import cifar10
from . import imagenet
from . import bin_module
