from .config import TrainConfig, mangle_output_dir, validate, get_lr
from .trainer import Trainer, set_seed

__all__ = ["TrainConfig", "mangle_output_dir", "validate", "get_lr", "Trainer", "set_seed"]
