from . import checkpoint, data, health, metrics, profiler, recordio  # noqa: F401
