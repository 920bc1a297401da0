from .timers import Timer, Timers

__all__ = ["Timer", "Timers"]
