from .arena import ArenaOptimizer  # noqa: F401
from .trainer import Trainer  # noqa: F401
