from .trainer import Trainer, ede_inject, build_kurtosis_table, DIFFKURT_TARGETS
from .checkpoint import save_state, load_state
from .inference import PackedInference
