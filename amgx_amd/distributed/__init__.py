from .manager import DistributedManager

__all__ = ["DistributedManager"]
