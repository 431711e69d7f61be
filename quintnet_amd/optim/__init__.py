from .zero import ZeroRedundancyAdamW, DistributedAdamW
from .zero2 import Zero2AdamW
from .schedule import LRSchedule

__all__ = ["ZeroRedundancyAdamW", "DistributedAdamW", "Zero2AdamW", "LRSchedule"]
