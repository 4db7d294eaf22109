from .trainer import Trainer, TrainConfig, PpTrainer

__all__ = ["Trainer", "TrainConfig", "PpTrainer"]
