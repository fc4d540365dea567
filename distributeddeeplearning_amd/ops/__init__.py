from . import functional  # noqa: F401
from .modules import (  # noqa: F401
    Conv2d,
    BatchNorm2d,
    MaxPool2d,
    GlobalAvgPool2d,
    Linear,
    Dropout,
    ReLU,
    cast_model_,
)
from .optim import FusedSGD  # noqa: F401
from ._ext import kernels_available, load_kernels_ext, load_comm_ext  # noqa: F401
