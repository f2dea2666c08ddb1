from .lars import LARS
from .scheduler import LinearWarmup, Scheduler
from .builder import build_optimizer, build_lr_schedule

__all__ = ["LARS", "LinearWarmup", "Scheduler", "build_optimizer",
           "build_lr_schedule"]
