"""Synthetic DM-Control-shaped environments for throughput benchmarks.

Spaces match the DM-Control suite tasks named in BASELINE.json (cheetah-run
obs 17 / act 6; humanoid-run obs 67 / act 21; quadruped-run pixels 84x84x3 /
act 12). Dynamics are a cheap mixing recurrence (obs' = tanh(A obs + B a) +
noise) with a bounded shaped reward, so rollouts are non-trivial but the cost
is dominated by the framework, as a throughput benchmark wants.
"""

from __future__ import annotations

import numpy as np

from .base import Env, EnvSpec

SPECS = {
    "cheetah-run": EnvSpec("cheetah-run", obs_shape=(17,), action_dim=6),
    "humanoid-run": EnvSpec("humanoid-run", obs_shape=(67,), action_dim=21),
    "quadruped-run": EnvSpec("quadruped-run", obs_shape=(78,), action_dim=12),
    "quadruped-run-pixels": EnvSpec(
        "quadruped-run-pixels", obs_shape=(84, 84, 3), action_dim=12, pixel=True
    ),
    "synthetic": EnvSpec("synthetic", obs_shape=(24,), action_dim=6),
}


class SyntheticEnv(Env):
    def __init__(self, name: str, seed: int = 0, time_limit: int = 1000, **_):
        self.spec = SPECS[name]
        self._time_limit = time_limit
        self._rng = np.random.default_rng(seed)
        self._t = 0
        if self.spec.pixel:
            self._dim = 32  # latent dim rendered to pixels
        else:
            self._dim = int(np.prod(self.spec.obs_shape))
        mix_rng = np.random.default_rng(12345)  # dynamics fixed across seeds
        self._A = (mix_rng.standard_normal((self._dim, self._dim)) / np.sqrt(self._dim)).astype(np.float32)
        self._B = (mix_rng.standard_normal((self._dim, self.spec.action_dim)) / np.sqrt(self.spec.action_dim)).astype(np.float32)
        self._state = np.zeros(self._dim, dtype=np.float32)

    def seed(self, seed: int) -> None:
        self._rng = np.random.default_rng(seed)

    def reset(self) -> np.ndarray:
        self._state = self._rng.standard_normal(self._dim).astype(np.float32) * 0.1
        self._t = 0
        return self._obs()

    def step(self, action: np.ndarray):
        a = np.clip(np.asarray(action, dtype=np.float32).reshape(-1), -1.0, 1.0)
        self._state = np.tanh(self._A @ self._state + self._B @ a) + \
            self._rng.standard_normal(self._dim).astype(np.float32) * 0.01
        # bounded [0, 1] reward, DM-Control style: reward tracking a moving target dir
        reward = float(0.5 + 0.5 * np.tanh(self._state[0]))
        self._t += 1
        trunc = self._t >= self._time_limit
        return self._obs(), reward, False, trunc

    def _obs(self) -> np.ndarray:
        if self.spec.pixel:
            # render latent to a cheap deterministic 84x84x3 uint8 image
            x = self._state
            img = np.outer(np.sin(np.linspace(0, 3, 84) + x[0] * 3),
                           np.cos(np.linspace(0, 3, 84) + x[1] * 3))
            img = ((img + 1) * 127.5).astype(np.uint8)
            return np.stack([img, np.roll(img, 7, 0), np.roll(img, 13, 1)], axis=-1)
        return self._state.astype(np.float32)
