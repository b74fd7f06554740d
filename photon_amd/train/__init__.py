from .scheduler import CosineWithWarmup, build_scheduler
from .timestamp import Timestamp
from .trainer import Trainer

__all__ = ["CosineWithWarmup", "build_scheduler", "Timestamp", "Trainer"]
