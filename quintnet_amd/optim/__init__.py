from .zero import ZeroRedundancyAdamW, DistributedAdamW

__all__ = ["ZeroRedundancyAdamW", "DistributedAdamW"]
