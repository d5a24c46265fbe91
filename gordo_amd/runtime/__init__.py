from .trainer import TrainConfig, TrainHistory, Trainer, resolve_device

__all__ = ["Trainer", "TrainConfig", "TrainHistory", "resolve_device"]
