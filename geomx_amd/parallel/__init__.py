from .trainer import GeoTrainer  # noqa: F401
