from .dist import (dist_init, dist_finalize, get_rank, get_world_size,
                   allreduce, broadcast, barrier, is_initialized,
                   simple_group_split)
from .ddp import DistModule
