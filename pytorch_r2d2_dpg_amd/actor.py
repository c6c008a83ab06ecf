"""Actor — CPU rollout loop.

Owns local fp32 CPU copies of ActorNet + CriticNet (the critic provides
step-wise Q values for initial priorities), steps the env with
a = mu(o, h) + N(0, sigma_i), maintains both LSTM states across steps,
feeds the sequence cutter, and periodically pulls fresh learner weights.

Runs in-process (single-process mode, tests) or as the body of an actor
process (parallel/actor_pool.py).
"""

from __future__ import annotations

from typing import List, Optional

import numpy as np
import torch

from .config import Config
from .envs import make_env
from .models import make_models
from .replay.sequence import Sequence, SequenceCutter
from .utils.noise import make_noise, sigma_ladder


class ActorWorker:
    def __init__(self, cfg: Config, actor_id: int = 0, sigma: Optional[float] = None,
                 eval_mode: bool = False):
        self.cfg = cfg
        self.actor_id = actor_id
        self.eval_mode = eval_mode
        self.env = make_env(cfg.env.name, seed=cfg.env.seed + 1000 * actor_id + 1,
                            time_limit=cfg.env.time_limit)
        spec = self.env.spec
        self.spec = spec
        self.actor_net, self.critic_net, _, _ = make_models(spec, cfg.model, device="cpu")
        self.actor_net.eval()
        self.critic_net.eval()
        if sigma is None:
            sigma = 0.0 if eval_mode else sigma_ladder(
                actor_id, max(cfg.actor.num_actors, 1), cfg.actor.sigma_max, cfg.actor.sigma_alpha)
        self.sigma = sigma
        self.noise = make_noise(cfg.actor.noise, spec.action_dim, sigma,
                                seed=cfg.env.seed + actor_id + 7777)
        R = cfg.replay
        self.cutter = SequenceCutter(R.seq_len, R.overlap, R.burn_in, R.n_step,
                                     R.gamma, R.priority_eta, cfg.model.hidden_size)
        self._obs = self.env.reset()
        H = cfg.model.hidden_size
        self._state = (torch.zeros(1, H), torch.zeros(1, H),
                       torch.zeros(1, H), torch.zeros(1, H))  # h_a, c_a, h_c, c_c
        self.episode_return = 0.0
        self.episode_returns: List[float] = []
        self.env_steps = 0

    # -------------------------------------------------------------- weights
    def load_weights(self, actor_sd: dict, critic_sd: dict) -> None:
        self.actor_net.load_state_dict(actor_sd)
        self.critic_net.load_state_dict(critic_sd)

    # ----------------------------------------------------------------- step
    @torch.no_grad()
    def rollout(self, num_steps: int) -> List[Sequence]:
        """Run `num_steps` env steps; return the sequences cut along the way."""
        out: List[Sequence] = []
        cap = self.cfg.actor.max_env_steps
        if cap and self.env_steps >= cap:
            return out
        h_a, c_a, h_c, c_c = self._state
        for _ in range(num_steps):
            snapshot = np.stack([h_a[0].numpy(), c_a[0].numpy(),
                                 h_c[0].numpy(), c_c[0].numpy()])
            obs_t = torch.from_numpy(np.ascontiguousarray(self._obs)).unsqueeze(0)
            if self.spec.pixel:
                a_feat = obs_t
            else:
                a_feat = obs_t.float()
            mu, (h_a, c_a) = self.actor_net.step(a_feat, (h_a, c_a))
            a = mu[0].numpy()
            if not self.eval_mode:
                a = np.clip(a + self.noise(),
                            -self.spec.action_scale, self.spec.action_scale)
            q, (h_c, c_c) = self.critic_net.step(
                a_feat, torch.from_numpy(a).unsqueeze(0), (h_c, c_c))
            # action repeat (env wrapper semantic): apply a for k frames,
            # accumulate reward, stop early on termination
            r = 0.0
            done = trunc = False
            for _ in range(max(self.cfg.env.action_repeat, 1)):
                next_obs, rk, done, trunc = self.env.step(a)
                r += rk
                if done or trunc:
                    break
            # timeout_terminal: treat the time limit as termination (config
            # choice — see ReplayConfig); else PEB (bootstrap through pad)
            d_eff = done or (trunc and self.cfg.replay.timeout_terminal)
            self.episode_return += r
            self.env_steps += 1
            out.extend(self.cutter.push(self._obs, a, r * self.cfg.replay.reward_scale,
                                        d_eff, float(q[0]), snapshot))
            self._obs = next_obs
            if done or trunc:
                if trunc and not d_eff:
                    out.extend(self.cutter.flush())
                self.episode_returns.append(self.episode_return)
                self.episode_return = 0.0
                self._obs = self.env.reset()
                self.noise.reset()
                H = self.cfg.model.hidden_size
                h_a = torch.zeros(1, H); c_a = torch.zeros(1, H)
                h_c = torch.zeros(1, H); c_c = torch.zeros(1, H)
        self._state = (h_a, c_a, h_c, c_c)
        return out


class VectorActorWorker:
    """Vectorized actor: ONE process steps E envs with BATCHED policy
    inference (obs [E, ...] through one forward) — removes the per-env
    Python/torch overhead that caps single-env actors at a few hundred
    steps/s for H=512 nets. Sequences, priorities and episode accounting
    are per-env (each env has its own cutter and noise stream).

    `rollout(n)` runs n batched steps = n*E env frames.
    """

    def __init__(self, cfg: Config, actor_id: int = 0, num_envs: int = 8,
                 eval_mode: bool = False):
        self.cfg = cfg
        self.actor_id = actor_id
        self.eval_mode = eval_mode
        self.num_envs = num_envs
        self.envs = [make_env(cfg.env.name,
                              seed=cfg.env.seed + 1000 * actor_id + 17 * e + 1,
                              time_limit=cfg.env.time_limit)
                     for e in range(num_envs)]
        spec = self.envs[0].spec
        self.spec = spec
        self.actor_net, self.critic_net, _, _ = make_models(spec, cfg.model, device="cpu")
        self.actor_net.eval()
        self.critic_net.eval()
        base_sigma = 0.0 if eval_mode else sigma_ladder(
            actor_id, max(cfg.actor.num_actors, 1), cfg.actor.sigma_max,
            cfg.actor.sigma_alpha)
        self.noise = [make_noise(cfg.actor.noise, spec.action_dim, base_sigma,
                                 seed=cfg.env.seed + actor_id * 131 + e + 7777)
                      for e in range(num_envs)]
        R = cfg.replay
        self.cutters = [SequenceCutter(R.seq_len, R.overlap, R.burn_in, R.n_step,
                                       R.gamma, R.priority_eta, cfg.model.hidden_size)
                        for _ in range(num_envs)]
        self._obs = np.stack([env.reset() for env in self.envs])
        H = cfg.model.hidden_size
        E = num_envs
        self._h_a = torch.zeros(E, H)
        self._c_a = torch.zeros(E, H)
        self._h_c = torch.zeros(E, H)
        self._c_c = torch.zeros(E, H)
        self.episode_return = np.zeros(E)
        self.episode_returns: List[float] = []
        self.env_steps = 0

    def load_weights(self, actor_sd: dict, critic_sd: dict) -> None:
        self.actor_net.load_state_dict(actor_sd)
        self.critic_net.load_state_dict(critic_sd)

    @torch.no_grad()
    def rollout(self, num_steps: int) -> List[Sequence]:
        out: List[Sequence] = []
        cap = self.cfg.actor.max_env_steps
        if cap and self.env_steps >= cap:
            return out
        E = self.num_envs
        rs = self.cfg.replay.reward_scale
        rep = max(self.cfg.env.action_repeat, 1)
        for _ in range(num_steps):
            snap = torch.stack([self._h_a, self._c_a, self._h_c, self._c_c],
                               dim=1).numpy()  # [E, 4, H]
            obs_t = torch.from_numpy(np.ascontiguousarray(self._obs))
            feat = obs_t if self.spec.pixel else obs_t.float()
            mu, (self._h_a, self._c_a) = self.actor_net.step(feat, (self._h_a, self._c_a))
            a = mu.numpy().copy()
            if not self.eval_mode:
                for e in range(E):
                    a[e] = np.clip(a[e] + self.noise[e](),
                                   -self.spec.action_scale, self.spec.action_scale)
            q, (self._h_c, self._c_c) = self.critic_net.step(
                feat, torch.from_numpy(a), (self._h_c, self._c_c))
            q_np = q.numpy()
            tt = self.cfg.replay.timeout_terminal
            for e, env in enumerate(self.envs):
                r_acc, done, trunc = 0.0, False, False
                for _ in range(rep):
                    next_obs, r, done, trunc = env.step(a[e])
                    r_acc += r
                    if done or trunc:
                        break
                d_eff = done or (trunc and tt)
                self.episode_return[e] += r_acc
                out.extend(self.cutters[e].push(self._obs[e], a[e], r_acc * rs,
                                                d_eff, float(q_np[e]), snap[e]))
                self._obs[e] = next_obs
                if done or trunc:
                    if trunc and not d_eff:
                        out.extend(self.cutters[e].flush())
                    self.episode_returns.append(float(self.episode_return[e]))
                    self.episode_return[e] = 0.0
                    self._obs[e] = env.reset()
                    self.noise[e].reset()
                    for t in (self._h_a, self._c_a, self._h_c, self._c_c):
                        t[e].zero_()
            self.env_steps += E
        return out
