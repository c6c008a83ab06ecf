"""Environment layer.

This sandbox has no gym/dm_control/mujoco (environment
availability constraint), so the framework ships its own env layer behind a
small protocol: a native Pendulum-v1 (known closed-form dynamics) for real
learning tests, and synthetic DM-Control-shaped trajectory generators
(cheetah-run obs 17 / act 6, humanoid-run obs 67 / act 21, quadruped-run
pixels 84x84x3 / act 12) for throughput benchmarks — which is what
BASELINE.json prescribes ("synthetic trajectories / random-init weights").
Real DM-Control plugs into the same protocol if available.
"""

from .base import Env, EnvSpec
from .pendulum import PendulumEnv
from .synthetic import SyntheticEnv, SPECS as SYNTHETIC_SPECS


def make_env(name: str, seed: int = 0, **kw) -> Env:
    if name in ("pendulum", "Pendulum-v1"):
        return PendulumEnv(seed=seed, **kw)
    if name in SYNTHETIC_SPECS:
        return SyntheticEnv(name, seed=seed, **kw)
    # "domain-task" names outside the synthetic set load real DM-Control
    # when the (optional) dependency is installed (envs/dmc.py)
    if "-" in name:
        domain, task = name.split("-", 1)
        try:
            from .dmc import DMControlEnv
            return DMControlEnv(domain, task, seed=seed, **kw)
        except ImportError:
            raise ValueError(
                f"env '{name}': dm_control is not installed in this "
                f"environment (offline sandbox); available: pendulum, "
                f"{', '.join(SYNTHETIC_SPECS)}") from None
    raise ValueError(f"unknown env '{name}' (have: pendulum, {', '.join(SYNTHETIC_SPECS)})")


__all__ = ["Env", "EnvSpec", "PendulumEnv", "SyntheticEnv", "make_env", "SYNTHETIC_SPECS"]
