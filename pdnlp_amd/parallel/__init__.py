from .bootstrap import (  # noqa: F401
    init_distributed, cleanup, spawn, get_rank, get_world_size, get_local_rank,
    barrier,
)
from .ddp import DistributedDataParallel  # noqa: F401
from .dp import DataParallel  # noqa: F401
from .hooks import DistributedOptimizer, broadcast_parameters, broadcast_optimizer_state  # noqa: F401
from .zero import ZeroRedundancyOptimizer, consolidate_zero_checkpoint  # noqa: F401
