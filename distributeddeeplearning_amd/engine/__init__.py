from .schedule import WarmupPiecewiseLR  # noqa: F401
from .loss_scaler import DynamicLossScaler  # noqa: F401
from .trainer import Trainer  # noqa: F401
from . import checkpoint  # noqa: F401
