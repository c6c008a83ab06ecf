"""Native Pendulum-v1 (classic-control) implementation.

Known closed-form dynamics (same as gym's Pendulum-v1):
    theta_dd = 3*g/(2*l) * sin(theta) + 3/(m*l^2) * u
    reward   = -(angle_norm(theta)^2 + 0.1*theta_dot^2 + 0.001*u^2)
obs = [cos(theta), sin(theta), theta_dot], action torque in [-2, 2],
episode truncates at 200 steps. Used for the end-to-end learning smoke test
(BASELINE.json config 1: "plumbing, no GPU").
"""

from __future__ import annotations

import numpy as np

from .base import Env, EnvSpec


class PendulumEnv(Env):
    MAX_SPEED = 8.0
    MAX_TORQUE = 2.0
    DT = 0.05
    G = 10.0
    M = 1.0
    L = 1.0

    def __init__(self, seed: int = 0, time_limit: int = 200, **_):
        self.spec = EnvSpec("pendulum", obs_shape=(3,), action_dim=1, action_scale=self.MAX_TORQUE)
        self._rng = np.random.default_rng(seed)
        self._time_limit = time_limit
        self._t = 0
        self._theta = 0.0
        self._theta_dot = 0.0

    def seed(self, seed: int) -> None:
        self._rng = np.random.default_rng(seed)

    def reset(self) -> np.ndarray:
        self._theta = self._rng.uniform(-np.pi, np.pi)
        self._theta_dot = self._rng.uniform(-1.0, 1.0)
        self._t = 0
        return self._obs()

    def step(self, action: np.ndarray):
        u = float(np.clip(np.asarray(action).reshape(-1)[0], -self.MAX_TORQUE, self.MAX_TORQUE))
        th, thd = self._theta, self._theta_dot
        ang = ((th + np.pi) % (2 * np.pi)) - np.pi
        reward = -(ang * ang + 0.1 * thd * thd + 0.001 * u * u)
        thd = thd + (3.0 * self.G / (2.0 * self.L) * np.sin(th) + 3.0 / (self.M * self.L**2) * u) * self.DT
        thd = float(np.clip(thd, -self.MAX_SPEED, self.MAX_SPEED))
        th = th + thd * self.DT
        self._theta, self._theta_dot = th, thd
        self._t += 1
        trunc = self._t >= self._time_limit
        return self._obs(), float(reward), False, trunc

    def _obs(self) -> np.ndarray:
        return np.array([np.cos(self._theta), np.sin(self._theta), self._theta_dot], dtype=np.float32)
