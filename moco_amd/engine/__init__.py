from .trainer import MoCoStepper, train_epoch

__all__ = ["MoCoStepper", "train_epoch"]
