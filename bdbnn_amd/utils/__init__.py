from .utils import (
    cpt_tk,
    find_weight_tensor_by_name,
    save_checkpoint,
    AverageMeter,
    ProgressMeter,
    accuracy,
    correct_counts,
)
from ..ops.kd import (
    DistributionLoss,
    DistributionLoss_layer,
    DistributionLoss_layer_cifar_act,
    loss_kd,
)
