"""Env protocol: ``reset() -> obs``, ``step(a) -> (obs, reward, done, trunc)``.

Continuous-control convention: observations are float32 numpy arrays (or
uint8 HWC for pixels), actions are float32 in [-action_scale, action_scale].
``done`` means environment termination (bootstrap with 0), ``trunc`` means
time-limit truncation (bootstrap continues) — the distinction feeds the
n-step TD bootstrap flag.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Tuple

import numpy as np


@dataclass(frozen=True)
class EnvSpec:
    name: str
    obs_shape: tuple
    action_dim: int
    action_scale: float = 1.0
    pixel: bool = False


class Env:
    spec: EnvSpec

    def reset(self) -> np.ndarray:
        raise NotImplementedError

    def step(self, action: np.ndarray) -> Tuple[np.ndarray, float, bool, bool]:
        raise NotImplementedError

    def seed(self, seed: int) -> None:
        raise NotImplementedError
