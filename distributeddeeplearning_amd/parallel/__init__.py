from .api import (  # noqa: F401
    init,
    shutdown,
    is_initialized,
    rank,
    local_rank,
    size,
    local_size,
    allreduce,
    allreduce_,
    allreduce_async,
    allreduce_async_,
    allgather,
    broadcast,
    broadcast_,
    broadcast_async,
    broadcast_async_,
    synchronize,
    barrier,
    broadcast_parameters,
    broadcast_optimizer_state,
)
from .optimizer import DistributedOptimizer  # noqa: F401
from .compression import Compression  # noqa: F401
from .bucket import BucketManager, Bucket  # noqa: F401
