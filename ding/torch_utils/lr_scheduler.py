"""LR schedules. Parity: reference ding/torch_utils/lr_scheduler.py."""
import math

from torch.optim.lr_scheduler import LambdaLR


def get_lr_ratio(epoch: int, warmup_epochs: int, learning_rate: float, lr_decay_epochs: int, min_lr: float) -> float:
    if epoch < warmup_epochs:
        return (epoch + 1) / warmup_epochs
    if epoch > lr_decay_epochs:
        return min_lr / learning_rate
    decay_ratio = (epoch - warmup_epochs) / (lr_decay_epochs - warmup_epochs)
    coeff = 0.5 * (1.0 + math.cos(math.pi * decay_ratio))
    return (min_lr + coeff * (learning_rate - min_lr)) / learning_rate


def cos_lr_scheduler(optimizer, learning_rate: float, warmup_epochs: float = 5, lr_decay_epochs: float = 100,
                     min_lr: float = 6e-5) -> LambdaLR:
    return LambdaLR(optimizer, lambda e: get_lr_ratio(e, warmup_epochs, learning_rate, lr_decay_epochs, min_lr))
