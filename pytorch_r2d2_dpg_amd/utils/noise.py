"""Exploration noise.

Ape-X-style per-actor scale ladder: sigma_i = sigma_max^(1 + alpha*i/(N-1)),
so actors explore at geometrically spaced intensities; each actor adds
iid Gaussian noise of its sigma to the deterministic action.
"""

from __future__ import annotations

import numpy as np


def sigma_ladder(actor_id: int, num_actors: int, sigma_max: float = 0.4,
                 alpha: float = 7.0) -> float:
    if num_actors <= 1:
        return sigma_max
    expo = 1.0 + alpha * actor_id / (num_actors - 1)
    return float(sigma_max ** expo)


class GaussianNoise:
    def __init__(self, action_dim: int, sigma: float, seed: int = 0):
        self.sigma = sigma
        self.action_dim = action_dim
        self._rng = np.random.default_rng(seed)

    def __call__(self) -> np.ndarray:
        return self._rng.standard_normal(self.action_dim).astype(np.float32) * self.sigma

    def reset(self) -> None:
        pass


class OUNoise:
    """Ornstein-Uhlenbeck process (the DDPG paper's temporally-correlated
    exploration): dx = theta*(mu - x)*dt + sigma*sqrt(dt)*N(0,1).

    dt defaults to 1 (the DDPG-paper implementation convention: one OU step
    per env step), giving stationary std ~ sigma/sqrt(2*theta - theta^2)
    ~ 1.9*sigma at theta=0.15 — with dt=1e-2 the stationary std collapses
    to ~0.18*sigma and exploration is too weak to learn (measured on
    Pendulum)."""

    def __init__(self, action_dim: int, sigma: float, seed: int = 0,
                 theta: float = 0.15, dt: float = 1.0, mu: float = 0.0):
        self.sigma = sigma
        self.theta = theta
        self.dt = dt
        self.mu = mu
        self.action_dim = action_dim
        self._rng = np.random.default_rng(seed)
        self._x = np.full(action_dim, mu, dtype=np.float32)

    def __call__(self) -> np.ndarray:
        self._x = (self._x + self.theta * (self.mu - self._x) * self.dt
                   + self.sigma * np.sqrt(self.dt)
                   * self._rng.standard_normal(self.action_dim).astype(np.float32))
        return self._x.astype(np.float32)

    def reset(self) -> None:
        self._x = np.full(self.action_dim, self.mu, dtype=np.float32)


def make_noise(kind: str, action_dim: int, sigma: float, seed: int = 0):
    if kind == "ou":
        return OUNoise(action_dim, sigma, seed=seed)
    if kind in ("gaussian", "normal"):
        return GaussianNoise(action_dim, sigma, seed=seed)
    raise ValueError(f"unknown noise kind '{kind}' (gaussian | ou)")
