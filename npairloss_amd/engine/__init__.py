from .solver import CaffeSGD, build_optimizer
from .trainer import Trainer

__all__ = ["CaffeSGD", "build_optimizer", "Trainer"]
