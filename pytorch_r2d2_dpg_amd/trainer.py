"""Single-process trainer (BASELINE.json config 1: plumbing, no GPU).

Runs one in-process actor interleaved with the learner — the minimum
end-to-end slice. The multi-process actor-pool path lives
in parallel/actor_pool.py and orchestrator.py.
"""

from __future__ import annotations

import time
from typing import Optional

from .actor import ActorWorker
from .checkpoint import save_checkpoint
from .config import Config
from .learner import Learner
from .replay.buffer import SequenceReplay
from .utils.metrics import MetricsLogger


def train_single_process(cfg: Config, device: str = "cpu",
                         max_steps: Optional[int] = None,
                         env_steps_per_update: int = 40,
                         logger: Optional[MetricsLogger] = None):
    """Returns (learner, actor_worker) after training."""
    actor_w = ActorWorker(cfg, actor_id=0, sigma=cfg.actor.sigma_max)
    spec = actor_w.spec
    R = cfg.replay
    replay = SequenceReplay(
        capacity=R.capacity, stored_len=R.seq_len + R.n_step,
        obs_shape=spec.obs_shape, action_dim=spec.action_dim,
        hidden_size=cfg.model.hidden_size, per_alpha=R.per_alpha,
        per_beta=R.per_beta, device=device, seed=cfg.learner.seed,
    )
    learner = Learner(cfg, spec, device=device, replay=replay)
    logger = logger or MetricsLogger(stdout_every=0)
    steps = max_steps if max_steps is not None else cfg.learner.steps

    # pre-fill
    while len(replay) < max(R.min_fill, cfg.learner.batch_size):
        replay.add(actor_w.rollout(cfg.env.time_limit))

    for step in range(steps):
        replay.add(actor_w.rollout(env_steps_per_update))
        batch = replay.sample(cfg.learner.batch_size)
        stats = learner.train_step(batch)
        if step % cfg.learner.sync_every == 0:
            sd = learner.actor_state_for_sync()
            actor_w.load_weights(sd["actor"], sd["critic"])
        if cfg.learner.checkpoint_every and (step + 1) % cfg.learner.checkpoint_every == 0:
            save_checkpoint(learner, cfg.learner.checkpoint_dir)
        if step % max(cfg.learner.log_every, 1) == 0:
            recent = actor_w.episode_returns[-10:]
            logger.log(step,
                       critic_loss=stats.critic_loss, actor_loss=stats.actor_loss,
                       mean_q=stats.mean_q, mean_priority=stats.mean_priority,
                       replay_size=len(replay),
                       episode_return=float(sum(recent) / len(recent)) if recent else 0.0)
    return learner, actor_w
