"""Run configuration.

One dataclass-backed config with TOML file + CLI overrides (SURVEY.md §5
"Config / flag system": the reference used constants + argparse in ``main.py``;
this framework ships the five BASELINE.json configs under ``configs/``).

Numeric defaults follow SURVEY.md §2.1 (R2D2 paper values with DPG-style soft
target updates): seq 80 / burn-in 40 / overlap 40, n-step 5, γ 0.997,
priority p = η·max|δ| + (1−η)·mean|δ| with η = 0.9, PER α = 0.9, β = 0.6
(the paper pairs α = 0.9 with IS exponent 0.6), Polyak τ = 5e-3, Adam lr 1e-4.
"""

from __future__ import annotations

import dataclasses
import json
from dataclasses import dataclass, field

try:  # py3.11+ stdlib, else the wheelhouse 'tomli'
    import tomllib as _toml
except ImportError:  # pragma: no cover
    import tomli as _toml  # type: ignore


@dataclass
class EnvConfig:
    name: str = "pendulum"          # pendulum | cheetah-run | humanoid-run | quadruped-run-pixels | synthetic
    pixel: bool = False             # 84x84x3 pixel observations (Conv+LSTM trunk)
    action_repeat: int = 1
    time_limit: int = 1000          # steps per episode (truncation, not termination)
    seed: int = 0


@dataclass
class ModelConfig:
    hidden_size: int = 256          # LSTM hidden (256-512)
    trunk_sizes: tuple = (256, 256)  # MLP trunk layer widths (state obs)
    conv_channels: tuple = (32, 64, 64)  # DQN-style conv stack for 84x84 pixels
    dtype: str = "fp32"             # learner compute dtype: fp32 | bf16


@dataclass
class ReplayConfig:
    capacity: int = 65536           # sequences per shard
    seq_len: int = 80               # stored sequence length (burn-in + train window)
    burn_in: int = 40
    overlap: int = 40               # adjacent-sequence overlap (= seq_len - stride)
    n_step: int = 5
    gamma: float = 0.997
    priority_eta: float = 0.9       # p = eta*max|d| + (1-eta)*mean|d|
    per_alpha: float = 0.9
    per_beta: float = 0.6           # importance-sampling exponent. The R2D2 paper
                                    # ran alpha=0.9 WITH beta=0.6 (Ape-X values);
                                    # set 0 to disable IS correction entirely.
    min_fill: int = 256             # sequences before learning starts
    reward_scale: float = 1.0       # rewards are scaled at actor-push time
    fp8_obs: bool = False           # store observations as fp8 e4m3 in the GPU shard
    # Time-limit handling. False (default) = partial-episode bootstrapping:
    # truncation keeps done=0 and the learner bootstraps from the target
    # net's value of the last real state (via the repeated-obs pad) — the
    # correct treatment of arbitrary time limits (DM-Control-style tasks).
    # True = treat the time limit as termination (classic episodic trick):
    # anchors the value scale at episode ends, which grounds Q thousands of
    # steps faster on short-horizon tasks like Pendulum (measured: PEB's Q
    # sinks unanchored toward r/(1-gamma) while the policy stays degenerate
    # at the smoke test's step budget; the episodic objective solves it).
    timeout_terminal: bool = False


@dataclass
class LearnerConfig:
    batch_size: int = 64            # sequences per rank per step
    lr_actor: float = 1e-4
    lr_critic: float = 1e-4
    adam_eps: float = 1e-8
    critic_weight_decay: float = 0.0  # DDPG-paper L2 on the critic
    tau: float = 5e-3               # Polyak soft-update rate
    target_update_every: int = 1
    value_rescale: bool = False     # h(x)=sign(x)(sqrt(|x|+1)-1)+eps*x (R2D2-DQN detail, default off)
    max_grad_norm: float = 40.0     # 0 disables clipping
    steps: int = 100_000
    sync_every: int = 100           # publish weights to actors every K steps
    checkpoint_every: int = 5000
    checkpoint_dir: str = "checkpoints"
    log_every: int = 100
    seed: int = 0


@dataclass
class ActorConfig:
    num_actors: int = 4
    envs_per_actor: int = 1         # vectorized actor: envs batched per process
    noise: str = "gaussian"         # gaussian | ou (Ornstein-Uhlenbeck, DDPG paper)
    sigma_max: float = 0.4          # Ape-X ladder: sigma_i = sigma_max^(1 + alpha*i/(N-1))
    sigma_alpha: float = 7.0
    eval_actors: int = 0            # noise-free actors for evaluation logging
    max_env_steps: int = 0          # 0 = unbounded
    torch_threads: int = 1          # intra-op threads per actor process
                                    # (vectorized actors batch K envs through
                                    # one CPU forward; >1 thread speeds the
                                    # policy GEMMs when cores outnumber
                                    # processes)


@dataclass
class Config:
    env: EnvConfig = field(default_factory=EnvConfig)
    model: ModelConfig = field(default_factory=ModelConfig)
    replay: ReplayConfig = field(default_factory=ReplayConfig)
    learner: LearnerConfig = field(default_factory=LearnerConfig)
    actor: ActorConfig = field(default_factory=ActorConfig)

    # ---- (de)serialization -------------------------------------------------
    @staticmethod
    def from_dict(d: dict) -> "Config":
        cfg = Config()
        for section_name, section in d.items():
            if not hasattr(cfg, section_name):
                raise KeyError(f"unknown config section [{section_name}]")
            sec = getattr(cfg, section_name)
            for k, v in section.items():
                if not hasattr(sec, k):
                    raise KeyError(f"unknown key {section_name}.{k}")
                cur = getattr(sec, k)
                if isinstance(cur, tuple):
                    v = tuple(v)
                setattr(sec, k, v)
        return cfg

    @staticmethod
    def from_toml(path: str) -> "Config":
        with open(path, "rb") as f:
            return Config.from_dict(_toml.load(f))

    def to_dict(self) -> dict:
        return dataclasses.asdict(self)

    def to_json(self) -> str:
        return json.dumps(self.to_dict(), indent=2)

    def apply_overrides(self, overrides: list) -> "Config":
        """Apply ``section.key=value`` CLI overrides in place."""
        for ov in overrides:
            key, _, raw = ov.partition("=")
            section_name, _, k = key.partition(".")
            sec = getattr(self, section_name)
            cur = getattr(sec, k)
            if isinstance(cur, bool):
                val = raw.lower() in ("1", "true", "yes", "on")
            elif isinstance(cur, int):
                val = int(raw)
            elif isinstance(cur, float):
                val = float(raw)
            elif isinstance(cur, tuple):
                val = tuple(int(x) for x in raw.strip("()[] ").split(",") if x)
            else:
                val = raw
            setattr(sec, k, val)
        return self

    # ---- derived -----------------------------------------------------------
    @property
    def train_len(self) -> int:
        """Training-window length: sequence minus burn-in prefix."""
        return self.replay.seq_len - self.replay.burn_in

    @property
    def seq_stride(self) -> int:
        return self.replay.seq_len - self.replay.overlap
