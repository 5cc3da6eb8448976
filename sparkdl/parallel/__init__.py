"""sparkdl.parallel — data-parallel machinery (RCCL over xGMI).

Native replacement for the Horovod engine the reference hosts but does
not ship (reference runner_base.py:25-35): process-group bootstrap,
bucketed gradient all-reduce overlapped with backward, and parameter /
optimizer-state broadcast.
"""

from sparkdl.parallel.comm import (  # noqa: F401
    init_process_group, is_initialized, rank, size, local_rank, local_size,
    allreduce_, broadcast_, barrier, shutdown,
)
from sparkdl.parallel.distributed_optimizer import (  # noqa: F401
    DistributedOptimizer, broadcast_parameters, broadcast_optimizer_state,
)
