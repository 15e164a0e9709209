from .checkpoint import load_checkpoint, save_checkpoint
from .loop import evaluate, train_epoch

__all__ = ["train_epoch", "evaluate", "save_checkpoint", "load_checkpoint"]
