from .isolation_forest import IsolationForest, IsolationForestModel
from .extended_isolation_forest import (
    ExtendedIsolationForest,
    ExtendedIsolationForestModel,
)

__all__ = [
    "IsolationForest",
    "IsolationForestModel",
    "ExtendedIsolationForest",
    "ExtendedIsolationForestModel",
]
