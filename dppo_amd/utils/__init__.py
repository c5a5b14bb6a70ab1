from .coordinator import Coordinator
from .logging import ScalarLogger
from .timers import PhaseTimers

__all__ = ["Coordinator", "ScalarLogger", "PhaseTimers"]
