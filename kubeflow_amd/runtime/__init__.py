from .trainer import Trainer, TrainConfig

__all__ = ["Trainer", "TrainConfig"]
