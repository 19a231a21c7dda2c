from .dist import init_distributed, get_rank, get_world_size, barrier
from .ddp import BucketedDataParallel
