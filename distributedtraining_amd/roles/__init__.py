from .miner import DeltaLoop
from .validator import DeltaValidator
from .averager import ParameterizedAverager

__all__ = ["DeltaLoop", "DeltaValidator", "ParameterizedAverager"]
