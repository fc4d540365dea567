from .synthetic import SyntheticImageNet, synthetic_loader  # noqa: F401
