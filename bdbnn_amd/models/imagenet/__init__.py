"""ImageNet model constructors (``imagenet_models.__dict__[arch](pretrained)``,
ref:train.py:285)."""

from . import resnet_bi_imagenet_set_2
from . import resnet_bi_imagenet_set_2_2

from .resnet_bi_imagenet_set_2_2 import (
    resnet18,
    resnet34,
)
from .resnet_bi_imagenet_set_2 import (
    resnet18_react,
    resnet34_react,
)
from ..resnet_common import resnet18_real, resnet34_real
