from .forest import Forest, ExtendedForest

__all__ = ["Forest", "ExtendedForest"]
