"""Training / evaluation engine."""
from .trainer import Trainer, SWATrainer, save_checkpoint, load_checkpoint
from .optimizer import FusedSGD
from .inference import (predict, find_peaks, find_connections, find_people,
                        process, subsets_to_keypoints, format_results,
                        validation)

__all__ = ["Trainer", "SWATrainer", "save_checkpoint", "load_checkpoint",
           "FusedSGD", "predict", "find_peaks", "find_connections",
           "find_people", "process", "subsets_to_keypoints", "format_results",
           "validation"]
