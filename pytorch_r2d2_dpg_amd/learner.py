"""Learner — the GPU-resident training loop.

Per step: sample prioritized sequence batch from the local replay shard →
burn-in (no grad) to rebuild LSTM states at the training-window start →
target-net unroll → fused n-step TD targets (K3) → critic update → actor
update (ascend grad-Q through the critic) → fused Polyak (K4) → priority
write-back (K6) → periodic weight publish / checkpoint.

Data-parallel: each rank runs this identically on its own shard; gradients
are all-reduced as one fused bucket over RCCL/xGMI after each backward
(parallel/dist.py) before the Adam step — DP ranks therefore hold identical
weights at all times; rank 0 alone checkpoints/logs.
"""

from __future__ import annotations

import time
from dataclasses import dataclass
from typing import Dict, Optional

import numpy as np
import torch

from .config import Config
from .models import make_models
from .ops.polyak import PolyakUpdater
from .ops.td import td_targets_and_priorities
from .ops.reference import value_rescale, value_rescale_inv
from .parallel.dist import DistContext
from .replay.buffer import SequenceReplay


@dataclass
class StepStats:
    critic_loss: float
    actor_loss: float
    mean_q: float
    mean_priority: float
    transitions: int


class Learner:
    def __init__(self, cfg: Config, spec, device: str = "cpu",
                 dist: Optional[DistContext] = None, replay: Optional[SequenceReplay] = None):
        self.cfg = cfg
        self.spec = spec
        self.device = torch.device(device)
        self.dist = dist or DistContext.single()
        self.dtype = torch.bfloat16 if cfg.model.dtype == "bf16" else torch.float32

        self.actor, self.critic, self.target_actor, self.target_critic = make_models(
            spec, cfg.model, device=self.device, dtype=self.dtype
        )
        # DP ranks start from identical weights
        self.dist.broadcast_params(list(self.actor.parameters()) + list(self.critic.parameters()))
        with torch.no_grad():
            for t, o in zip(self.target_actor.parameters(), self.actor.parameters()):
                t.copy_(o)
            for t, o in zip(self.target_critic.parameters(), self.critic.parameters()):
                t.copy_(o)

        lc = cfg.learner
        self.opt_actor = torch.optim.Adam(self.actor.parameters(), lr=lc.lr_actor, eps=lc.adam_eps)
        self.opt_critic = torch.optim.Adam(self.critic.parameters(), lr=lc.lr_critic, eps=lc.adam_eps)
        self.polyak_actor = PolyakUpdater(
            list(self.target_actor.parameters()), list(self.actor.parameters()))
        self.polyak_critic = PolyakUpdater(
            list(self.target_critic.parameters()), list(self.critic.parameters()))

        self.replay = replay
        self.step_count = 0

        R = cfg.replay
        self.burn_in = R.burn_in
        self.train_len = cfg.train_len
        self.n_step = R.n_step
        self.gamma = R.gamma
        self.eta = R.priority_eta

    # ------------------------------------------------------------ train step
    def train_step(self, batch: Dict[str, torch.Tensor]) -> StepStats:
        """One learner update on a sampled batch (tensors already on device).

        Batch layout: obs [B, L, ...], action [B, L, A], reward [B, L],
        done [B, L], state [B, 4, H], weight [B]; L = burn_in + T + n_step.
        """
        cfg = self.cfg
        B = batch["obs"].shape[0]
        Tn = self.train_len + self.n_step
        bi = self.burn_in

        # time-major views
        obs = batch["obs"].transpose(0, 1).to(self.dtype)          # [L, B, ...]
        action = batch["action"].transpose(0, 1).to(self.dtype)    # [L, B, A]
        reward = batch["reward"]                                   # [B, L] fp32
        done = batch["done"]                                       # [B, L] fp32
        weight = batch["weight"]                                   # [B] fp32
        st = batch["state"].to(self.dtype)                         # [B, 4, H]
        h_a, c_a = st[:, 0].contiguous(), st[:, 1].contiguous()
        h_c, c_c = st[:, 2].contiguous(), st[:, 3].contiguous()

        # ---- burn-in (no grad): rebuild states at the training-window start
        if bi > 0:
            with torch.no_grad():
                _, (h_a2, c_a2) = self.actor(obs[:bi], (h_a, c_a))
                _, (h_c2, c_c2) = self.critic(obs[:bi], action[:bi], (h_c, c_c))
        else:
            h_a2, c_a2, h_c2, c_c2 = h_a, c_a, h_c, c_c

        # ---- target unroll (no grad) over train window + n-step tail
        with torch.no_grad():
            mu_t, _ = self.target_actor(obs[bi:bi + Tn], (h_a2, c_a2))
            qt, _ = self.target_critic(obs[bi:bi + Tn], mu_t, (h_c2, c_c2))
            # bootstrap value at t+n for each train-window t: [B, T]
            q_boot = qt[self.n_step:self.n_step + self.train_len].transpose(0, 1).float()
            if cfg.learner.value_rescale:
                q_boot = value_rescale_inv(q_boot)

        # ---- critic update on n-step targets
        T = self.train_len
        q_online, _ = self.critic(obs[bi:bi + T], action[bi:bi + T],
                                  (h_c2.detach(), c_c2.detach()))
        q_online_bt = q_online.transpose(0, 1).float()  # [B, T]
        with torch.no_grad():
            y, td_abs, prio = td_targets_and_priorities(
                reward[:, bi:], done[:, bi:], q_boot, q_online_bt.detach(),
                self.gamma, self.n_step, self.eta,
            )
            if cfg.learner.value_rescale:
                y = value_rescale(y)
                td_abs = (y - q_online_bt.detach()).abs()
                prio = self.eta * td_abs.max(1).values + (1 - self.eta) * td_abs.mean(1)
        critic_loss = (weight.unsqueeze(1) * (y - q_online_bt) ** 2).mean()
        self.opt_critic.zero_grad(set_to_none=True)
        critic_loss.backward()
        self.dist.allreduce_grads(self.critic.parameters())
        if cfg.learner.max_grad_norm > 0:
            torch.nn.utils.clip_grad_norm_(self.critic.parameters(), cfg.learner.max_grad_norm)
        self.opt_critic.step()

        # ---- actor update: ascend Q(s, mu(s)) through the critic
        mu, _ = self.actor(obs[bi:bi + T], (h_a2.detach(), c_a2.detach()))
        q_pi, _ = self.critic(obs[bi:bi + T], mu, (h_c2.detach(), c_c2.detach()))
        actor_loss = -q_pi.float().mean()
        self.opt_actor.zero_grad(set_to_none=True)
        self.opt_critic.zero_grad(set_to_none=True)  # discard critic grads from this path
        actor_loss.backward()
        self.dist.allreduce_grads(self.actor.parameters())
        if cfg.learner.max_grad_norm > 0:
            torch.nn.utils.clip_grad_norm_(self.actor.parameters(), cfg.learner.max_grad_norm)
        self.opt_actor.step()
        self.opt_critic.zero_grad(set_to_none=True)

        # ---- Polyak soft target update (K4: one fused launch per net on GPU)
        if self.step_count % cfg.learner.target_update_every == 0:
            self.polyak_actor(cfg.learner.tau)
            self.polyak_critic(cfg.learner.tau)

        # ---- priority write-back (stays on-GPU with the K6 tree kernel)
        if self.replay is not None and batch.get("idx") is not None:
            self.replay.update_priorities(batch["idx"], prio)

        self.step_count += 1
        return StepStats(
            critic_loss=float(critic_loss.detach()),
            actor_loss=float(actor_loss.detach()),
            mean_q=float(q_online_bt.detach().mean()),
            mean_priority=float(prio.mean()),
            transitions=B * T,
        )

    # -------------------------------------------------------------- weights
    def actor_state_for_sync(self) -> Dict[str, torch.Tensor]:
        """fp32 CPU state_dicts actors load (actor + critic for local priorities)."""
        return {
            "actor": {k: v.detach().float().cpu() for k, v in self.actor.state_dict().items()},
            "critic": {k: v.detach().float().cpu() for k, v in self.critic.state_dict().items()},
        }
