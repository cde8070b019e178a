from .optim import build_optimizer, clip_grads_global_norm
from .checkpoint import save_checkpoint, load_checkpoint, save_options, load_options
from .trainer import train
from .validate import pred_probs

__all__ = [
    "build_optimizer",
    "clip_grads_global_norm",
    "save_checkpoint",
    "load_checkpoint",
    "save_options",
    "load_options",
    "train",
    "pred_probs",
]
