"""Real DM-Control adapter (optional dependency).

Wraps a ``dm_control.suite`` task behind the framework's Env protocol:
observation dict flattened to one float32 vector (sorted key order, the
dm_control convention), reward defaulting to 0.0 on the first step, and the
termination/truncation split derived from the timestep discount —
``discount == 0`` on the final step means true termination (bootstrap 0),
anything else at episode end is a time-limit truncation (bootstrap
continues), which is what the n-step fold's done-flag needs.

This sandbox ships without dm_control; ``make_env`` falls back to the
synthetic DM-Control-shaped envs for the BASELINE task names and raises a
clear error otherwise. On a machine with dm_control installed, any
``"domain-task"`` name outside the synthetic set loads the real thing.
"""

from __future__ import annotations

import numpy as np

from .base import Env, EnvSpec


def _flatten(obs_dict) -> np.ndarray:
    parts = [np.asarray(obs_dict[k], dtype=np.float32).reshape(-1)
             for k in sorted(obs_dict)]
    return np.concatenate(parts) if parts else np.zeros(0, np.float32)


class DMControlEnv(Env):
    def __init__(self, domain: str, task: str, seed: int = 0,
                 time_limit: int = 1000, **_):
        from dm_control import suite  # raises ImportError when absent

        self._env = suite.load(domain, task, task_kwargs={"random": seed})
        a_spec = self._env.action_spec()
        obs_spec = self._env.observation_spec()
        obs_dim = int(sum(int(np.prod(v.shape)) if len(v.shape) else 1
                          for v in obs_spec.values()))
        scale = float(np.max(np.abs(np.asarray(a_spec.maximum))))
        self.spec = EnvSpec(f"{domain}-{task}", obs_shape=(obs_dim,),
                            action_dim=int(np.prod(a_spec.shape)),
                            action_scale=scale)
        self._time_limit = time_limit
        self._t = 0

    def reset(self) -> np.ndarray:
        self._t = 0
        ts = self._env.reset()
        return _flatten(ts.observation)

    def step(self, action: np.ndarray):
        ts = self._env.step(np.asarray(action, dtype=np.float32))
        self._t += 1
        last = bool(ts.last())
        done = last and float(ts.discount or 0.0) == 0.0
        trunc = (last and not done) or self._t >= self._time_limit
        reward = float(ts.reward) if ts.reward is not None else 0.0
        return _flatten(ts.observation), reward, done, trunc

    def seed(self, seed: int) -> None:
        # dm_control seeds at construction (task_kwargs["random"])
        pass
