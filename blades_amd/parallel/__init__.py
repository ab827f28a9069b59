from .runtime import DistributedRuntime

__all__ = ["DistributedRuntime"]
