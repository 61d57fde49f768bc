from . import checkpoint, data, health, metrics, profiler  # noqa: F401
