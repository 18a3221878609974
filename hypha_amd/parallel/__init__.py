from .comm import Comm
from .diloco import DiLoCoConfig, DiLoCoWorker, InnerOptConfig, OuterOptConfig, lr_at
from .lean import LeanDiLoCoWorker

__all__ = [
    "Comm",
    "DiLoCoConfig",
    "DiLoCoWorker",
    "InnerOptConfig",
    "OuterOptConfig",
    "lr_at",
    "LeanDiLoCoWorker",
]
