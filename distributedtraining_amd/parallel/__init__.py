from .flat import FlatParams, FusedAdamW
from .comm import CommPlane
from .local_sgd import LocalSGDNode

__all__ = ["FlatParams", "FusedAdamW", "CommPlane", "LocalSGDNode"]
