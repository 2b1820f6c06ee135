"""datafusion_distributed_amd — MI355X-native hash-repartition + RCCL/xGMI exchange for the
datafusion-distributed shuffle hot path.

Layout:
  csrc/       HIP kernels (gfx950) + C-ABI host library (libdd_shuffle.so)
  api.py      ctypes binding + numpy upload/download harness
  shuffle.py  host mirror of the reference's operator seam (TaskKey / window math / pool)
  exchange.py torch.distributed bootstrap for the RCCL communicator
"""

from .shuffle import (  # noqa: F401
    DistributedTaskContext,
    ExecuteTaskRequest,
    NetworkShuffleExec,
    TaskKey,
    WorkerConnectionPool,
    partition_window,
    scale_partitioning,
)
