from .flat import FlatParams, FusedAdamW
from .comm import CommPlane

__all__ = ["FlatParams", "FusedAdamW", "CommPlane", "LocalSGDNode"]


def __getattr__(name):
    # LocalSGDNode imports the role modules, which import this package —
    # resolve lazily to avoid the cycle.
    if name == "LocalSGDNode":
        from .local_sgd import LocalSGDNode
        return LocalSGDNode
    raise AttributeError(name)
