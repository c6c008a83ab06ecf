"""Checkpoint save/load.

Preserves the reference family's torch-serialized layout: one dict of
state_dicts {actor, critic, target_actor, target_critic, opt_actor,
opt_critic, step} saved as ``ckpt_{step}.pth`` by rank 0 only. Replay
contents are NOT checkpointed (matches the reference; documented semantic).
Restore on a DP learner = rank-0 load + parameter broadcast.
"""

from __future__ import annotations

import os
from typing import Optional

import torch


def save_checkpoint(learner, directory: str, step: Optional[int] = None) -> str:
    step = learner.step_count if step is None else step
    os.makedirs(directory, exist_ok=True)
    path = os.path.join(directory, f"ckpt_{step}.pth")
    torch.save(
        {
            "actor": learner.actor.state_dict(),
            "critic": learner.critic.state_dict(),
            "target_actor": learner.target_actor.state_dict(),
            "target_critic": learner.target_critic.state_dict(),
            "opt_actor": learner.opt_actor.state_dict(),
            "opt_critic": learner.opt_critic.state_dict(),
            "step": step,
        },
        path,
    )
    return path


def load_checkpoint(learner, path: str) -> int:
    sd = torch.load(path, map_location="cpu", weights_only=False)
    learner.actor.load_state_dict(sd["actor"])
    learner.critic.load_state_dict(sd["critic"])
    learner.target_actor.load_state_dict(sd["target_actor"])
    learner.target_critic.load_state_dict(sd["target_critic"])
    learner.opt_actor.load_state_dict(sd["opt_actor"])
    learner.opt_critic.load_state_dict(sd["opt_critic"])
    learner.step_count = int(sd["step"])
    learner.dist.broadcast_params(
        list(learner.actor.parameters()) + list(learner.critic.parameters()))
    return learner.step_count
