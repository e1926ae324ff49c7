from .scheduler import StreamScheduler
from .similarity import StochasticSimilarityFilter
from .engine import StreamDiffusionEngine

__all__ = ["StreamScheduler", "StochasticSimilarityFilter", "StreamDiffusionEngine"]
