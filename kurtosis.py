"""Drop-in import shim for the reference's kurtosis module
(ref:kurtosis.py): same classes, fused multi-tensor kernels available
via bdbnn_amd.ops.kurtosis.kurtosis_loss_fused."""
from bdbnn_amd.ops.kurtosis import (  # noqa: F401
    KurtosisWeight,
    RidgeRegularization,
    WeightRegularization,
)
