from .manager import PerformanceManager

__all__ = ["PerformanceManager"]
