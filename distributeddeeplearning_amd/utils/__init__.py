from .timeline import Timeline  # noqa: F401
