from .binarize import (
    binsign,
    SignSTE,
    SignEDE,
    SignApprox,
    binarize_weight,
    weight_scale,
    BinaryActivation,
    LearnableBias,
)
from .binary_conv import (
    HardBinaryConv,
    HardBinaryConv_react,
    HardBinaryConv_cifar,
    BinaryConvFunction,
)
from .kurtosis import (
    KurtosisWeight,
    RidgeRegularization,
    WeightRegularization,
    kurtosis_loss_fused,
)
from .kd import (
    DistributionLoss,
    DistributionLoss_layer,
    DistributionLoss_layer_cifar_act,
    loss_kd,
)
