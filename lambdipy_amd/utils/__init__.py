from .log import get_logger
from .timing import Timer, phase_trace

__all__ = ["get_logger", "Timer", "phase_trace"]
