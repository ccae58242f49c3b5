from .strategy import Strategy
from .sampler import BehaviorSampler

__all__ = ["Strategy", "BehaviorSampler"]
