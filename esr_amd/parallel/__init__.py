from .ddp import (init_distributed, is_distributed, get_rank, get_world_size,  # noqa: F401
                  barrier, wrap_ddp, setup_rank0_print)
from .reduce import reduce_tensor, reduce_dict  # noqa: F401
from .watchdog import Watchdog  # noqa: F401
