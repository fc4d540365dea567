"""distributeddeeplearning_amd — MI355X-native distributed data-parallel training.

A from-scratch rebuild of the capability surface of
microsoft/DistributedDeepLearning (Horovod-based ResNet/ImageNet DP training
scripts) designed MI355X-first: one process per GPU, gradients bucketed into
flat persistent buffers and ring-all-reduced over RCCL/xGMI on a side HIP
stream overlapped with backward, and the CNN hot path (implicit-GEMM NHWC
conv, fused BN+ReLU, fused SGD/LARS) in hand-written HIP/CDNA4 kernels
(gfx950).

The public distributed API keeps the Horovod call surface
(``init/rank/size/allreduce/broadcast_parameters/DistributedOptimizer/...``,
reference: Horovod ``horovod/torch/__init__.py``; the reference repo's
scripts call exactly this surface from ``HorovodPytorch/src/
imagenet_pytorch_horovod.py``).
"""

__version__ = "0.1.0"

from .parallel.api import (  # noqa: F401
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
from .parallel.optimizer import DistributedOptimizer  # noqa: F401
from .parallel.compression import Compression  # noqa: F401
from . import models  # noqa: F401
from . import ops  # noqa: F401
from .config import TrainConfig, PRESETS  # noqa: F401
