from .timers import StageTimers

__all__ = ["StageTimers"]
