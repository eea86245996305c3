from .metrics import (ConfusionMatrix, accuracy, dice_coeff, dice_loss,
                      multiclass_dice_coeff)
from .scheduler import WarmupScheduler, scale_lr_linear
from .callbacks import Callbacks  # noqa: F401
from .trainer import Trainer, evaluate, throughput_test, train_one_epoch

__all__ = ["accuracy", "ConfusionMatrix", "dice_coeff", "multiclass_dice_coeff",
           "dice_loss", "WarmupScheduler", "scale_lr_linear", "Trainer",
           "train_one_epoch", "evaluate", "throughput_test"]
