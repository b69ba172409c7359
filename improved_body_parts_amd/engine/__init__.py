"""Training / evaluation engine."""
from .trainer import Trainer, SWATrainer, save_checkpoint, load_checkpoint
from .optimizer import FusedSGD

__all__ = ["Trainer", "SWATrainer", "save_checkpoint", "load_checkpoint",
           "FusedSGD"]
