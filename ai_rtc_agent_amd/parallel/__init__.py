from .collectives import (
    broadcast_engine_weights,
    broadcast_module,
    init_distributed,
    is_distributed,
)

__all__ = [
    "broadcast_engine_weights",
    "broadcast_module",
    "init_distributed",
    "is_distributed",
]
