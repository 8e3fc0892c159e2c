from .scheduler import RoundScheduler

__all__ = ["RoundScheduler"]
