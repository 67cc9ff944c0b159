from .meters import AverageMeter, ProgressMeter, accuracy
from .checkpoint import save_checkpoint, load_pretrained_backbone, sanity_check
from .lr import adjust_learning_rate
from .metrics import MetricsLogger

__all__ = [
    "AverageMeter",
    "ProgressMeter",
    "accuracy",
    "save_checkpoint",
    "load_pretrained_backbone",
    "sanity_check",
    "adjust_learning_rate",
    "MetricsLogger",
]
