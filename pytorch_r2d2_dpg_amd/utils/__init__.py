from .noise import sigma_ladder, GaussianNoise
from .metrics import MetricsLogger

__all__ = ["sigma_ladder", "GaussianNoise", "MetricsLogger"]
