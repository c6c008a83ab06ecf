"""Plain-PyTorch fp32 reference implementations of the hot ops.

These define the semantics that the HIP kernels (ops/csrc/*.hip) must match;
kernel numerics tests compare against THESE on identical inputs (SURVEY.md §4
"Kernel unit"). They also serve as the CPU execution path (actors, CPU CI).

LSTM cell semantics (single fused bias, torch gate order i,f,g,o):
    gates = x @ Wx^T + h @ Wh^T + b
    i,f,g,o = sigmoid/sigmoid/tanh/sigmoid splits of gates
    c' = f*c + i*g ;  h' = o * tanh(c')
"""

from __future__ import annotations

from typing import Tuple

import torch


# ---------------------------------------------------------------- LSTM (K1/K2)
def lstm_seq_ref(
    x: torch.Tensor,      # [T, B, I]
    h0: torch.Tensor,     # [B, H]
    c0: torch.Tensor,     # [B, H]
    wx: torch.Tensor,     # [4H, I]
    wh: torch.Tensor,     # [4H, H]
    b: torch.Tensor,      # [4H]
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Sequence LSTM forward. Returns (h_seq [T,B,H], hT, cT). Differentiable."""
    T = x.shape[0]
    H = h0.shape[1]
    gates_x = x @ wx.t()  # [T, B, 4H]
    h, c = h0, c0
    hs = []
    for t in range(T):
        gates = gates_x[t] + h @ wh.t() + b
        i = torch.sigmoid(gates[:, 0 * H:1 * H])
        f = torch.sigmoid(gates[:, 1 * H:2 * H])
        g = torch.tanh(gates[:, 2 * H:3 * H])
        o = torch.sigmoid(gates[:, 3 * H:4 * H])
        c = f * c + i * g
        h = o * torch.tanh(c)
        hs.append(h)
    return torch.stack(hs, dim=0), h, c


# ------------------------------------------------------- n-step TD target (K3)
def nstep_targets_ref(
    rewards: torch.Tensor,   # [B, T] n-step-FOLDED rewards: sum_k gamma^k r_{t+k}
    bootstrap_q: torch.Tensor,  # [B, T] Q'(s_{t+n}, mu'(s_{t+n})) (already shifted)
    bootstrap_mask: torch.Tensor,  # [B, T] 0 where episode terminated within the n window
    gamma_n: float,
) -> torch.Tensor:
    """y_t = folded_r_t + gamma^n * mask_t * q'_t  (rewards pre-folded by the actor)."""
    return rewards + gamma_n * bootstrap_mask * bootstrap_q


def fold_nstep_ref(
    rewards: torch.Tensor,  # [B, T_ext] raw per-step rewards (T_ext >= T + n - 1)
    dones: torch.Tensor,    # [B, T_ext] 1.0 where the episode terminated at that step
    gamma: float,
    n: int,
    t_out: int,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Fold raw rewards into n-step returns.

    Returns (folded [B, t_out], bootstrap_mask [B, t_out]) where
    folded_t = sum_{k<n} gamma^k r_{t+k} truncated at the first done, and
    bootstrap_mask_t = 0 if a done occurs in [t, t+n).
    """
    B, T_ext = rewards.shape
    folded = torch.zeros(B, t_out, dtype=rewards.dtype, device=rewards.device)
    mask = torch.ones(B, t_out, dtype=rewards.dtype, device=rewards.device)
    for t in range(t_out):
        acc = torch.zeros(B, dtype=rewards.dtype, device=rewards.device)
        alive = torch.ones(B, dtype=rewards.dtype, device=rewards.device)
        g = 1.0
        for k in range(n):
            idx = t + k
            if idx >= T_ext:
                alive = torch.zeros_like(alive)
                break
            acc = acc + g * alive * rewards[:, idx]
            alive = alive * (1.0 - dones[:, idx])
            g *= gamma
        folded[:, t] = acc
        mask[:, t] = alive
    return folded, mask


def td_error_ref(q: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
    return y - q


def sequence_priority_ref(td_abs: torch.Tensor, eta: float) -> torch.Tensor:
    """p = eta*max_t|d| + (1-eta)*mean_t|d| per sequence. td_abs: [B, T]."""
    return eta * td_abs.max(dim=1).values + (1.0 - eta) * td_abs.mean(dim=1)


# ----------------------------------------------------------------- Polyak (K4)
@torch.no_grad()
def polyak_ref(target_params, online_params, tau: float) -> None:
    """theta' <- tau*theta + (1-tau)*theta' over a parameter list, in place."""
    for tp, p in zip(target_params, online_params):
        tp.mul_(1.0 - tau).add_(p.detach().to(tp.dtype), alpha=tau)


# ------------------------------------------------------- value rescaling (opt)
def value_rescale(x: torch.Tensor, eps: float = 1e-3) -> torch.Tensor:
    return torch.sign(x) * (torch.sqrt(torch.abs(x) + 1.0) - 1.0) + eps * x


def value_rescale_inv(x: torch.Tensor, eps: float = 1e-3) -> torch.Tensor:
    # closed-form inverse of value_rescale
    return torch.sign(x) * (
        ((torch.sqrt(1.0 + 4.0 * eps * (torch.abs(x) + 1.0 + eps)) - 1.0) / (2.0 * eps)) ** 2 - 1.0
    )
