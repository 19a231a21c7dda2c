"""Drop-in import shim (ref:utils/KD_loss.py)."""
from bdbnn_amd.ops.kd import (  # noqa: F401
    DistributionLoss,
    DistributionLoss_layer,
    DistributionLoss_layer_cifar_act,
    loss_kd,
)
