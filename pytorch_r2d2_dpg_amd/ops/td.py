"""n-step TD-target reduction + sequence priorities (kernel K3).

One fused, memory-bound pass on GPU: for each sequence b
and training-window step t,

    folded_t = sum_{k<n} gamma^k r_{t+k}   (truncated at the first done)
    mask_t   = 0 if a done occurs in [t, t+n)
    y_t      = folded_t + gamma^n * mask_t * q'_{t+n}
    delta_t  = y_t - q_t
    p_b      = eta * max_t|delta_t| + (1-eta) * mean_t|delta_t|

Inputs carry the n-step tail: rewards/dones are [B, T+n-ish] (>= T+n-1 used),
bootstrap q' is [B, T] (already the value at t+n from the target-net unroll).
"""

from __future__ import annotations

from typing import Tuple

import torch

from . import hip_ext
from .reference import (fold_nstep_ref, sequence_priority_ref, value_rescale,
                        value_rescale_inv)


def td_targets_and_priorities(
    rewards: torch.Tensor,        # [B, T_ext] raw rewards from window start
    dones: torch.Tensor,          # [B, T_ext] terminal flags (1.0 = terminated)
    bootstrap_q: torch.Tensor,    # [B, T] Q'(s_{t+n}, mu'(s_{t+n}))
    q_online: torch.Tensor,       # [B, T] Q(s_t, a_t), detached
    gamma: float,
    n_step: int,
    eta: float,
    rescale: bool = False,        # R2D2 value rescaling h(.) (flag, default off)
    rescale_eps: float = 1e-3,
    valid_n: torch.Tensor = None,  # [B] int32: real (non-pad) steps from the
                                   # window start, UNCLAMPED (may reach into
                                   # the n-step tail). Steps >= min(valid, T)
                                   # get td_abs=0 and no priority mass; the
                                   # reward fold treats indices >= valid as
                                   # r=0/done=0 so stored pad bytes never
                                   # leak into a real step's target.
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Returns (targets y [B,T], |delta| [B,T], priorities [B])."""
    T = q_online.shape[1]
    if rewards.is_cuda:
        return hip_ext().td_targets(
            rewards.contiguous(), dones.contiguous(), bootstrap_q.contiguous(),
            q_online.contiguous(), float(gamma), int(n_step), float(eta),
            rescale, rescale_eps,
            valid_n.contiguous() if valid_n is not None else None,
        )
    if valid_n is not None:
        # mirror the kernel's reward-extent guard
        t_ext = rewards.shape[1]
        ext_mask = (torch.arange(t_ext, device=rewards.device).unsqueeze(0)
                    < valid_n.to(torch.long).clamp(min=1).unsqueeze(1)).to(rewards.dtype)
        rewards = rewards * ext_mask
        dones = dones * ext_mask
    qb = value_rescale_inv(bootstrap_q, rescale_eps) if rescale else bootstrap_q
    folded, mask = fold_nstep_ref(rewards, dones, gamma, n_step, T)
    y = folded + (gamma ** n_step) * mask * qb
    if rescale:
        y = value_rescale(y, rescale_eps)
    td_abs = (y - q_online).abs()
    if valid_n is not None:
        vn = valid_n.to(torch.long).clamp(1, T)
        vmask = (torch.arange(T, device=td_abs.device).unsqueeze(0)
                 < vn.unsqueeze(1)).to(td_abs.dtype)
        td_abs = td_abs * vmask
        prio = (eta * td_abs.max(dim=1).values
                + (1.0 - eta) * td_abs.sum(dim=1) / vn.to(td_abs.dtype))
    else:
        prio = sequence_priority_ref(td_abs, eta)
    return y, td_abs, prio
