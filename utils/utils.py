"""Drop-in import shim (ref:utils/utils.py)."""
from bdbnn_amd.utils.utils import (  # noqa: F401
    cpt_tk,
    find_weight_tensor_by_name,
    save_checkpoint,
    AverageMeter,
    ProgressMeter,
    accuracy,
)
