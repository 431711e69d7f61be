from .zero import ZeroRedundancyAdamW, DistributedAdamW
from .zero2 import Zero2AdamW

__all__ = ["ZeroRedundancyAdamW", "DistributedAdamW", "Zero2AdamW"]
