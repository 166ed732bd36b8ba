from .core import LPTrainStep

__all__ = ["LPTrainStep"]
