from .optim import FusedAdam
from .trainer import Trainer
from .evaluator import evaluate_aee

__all__ = ["FusedAdam", "Trainer", "evaluate_aee"]
