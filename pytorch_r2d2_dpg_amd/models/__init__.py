from .lstm import FusedLSTM
from .actor_critic import ActorNet, CriticNet, make_models

__all__ = ["FusedLSTM", "ActorNet", "CriticNet", "make_models"]
