"""Actor / Critic networks.

ActorNet : obs -> trunk (MLP, or Conv stack for 84x84 pixels) -> FusedLSTM
           -> Linear -> tanh * action_scale
CriticNet: concat(obs_features, action) -> FusedLSTM -> Linear -> Q scalar

Both expose sequence form ``forward(x, state)`` over [T, B, ...] and a
single-step ``step`` for actor rollouts; recurrent state is (h, c).
Target copies are deep copies updated by Polyak (ops/polyak.py).
"""

from __future__ import annotations

import copy
from typing import Optional

import torch
from torch import nn

from .lstm import FusedLSTM, State


def mlp_trunk(in_dim: int, sizes) -> nn.Sequential:
    layers = []
    d = in_dim
    for s in sizes:
        layers += [nn.Linear(d, s), nn.ReLU(inplace=True)]
        d = s
    return nn.Sequential(*layers)


class ConvTrunk(nn.Module):
    """DQN-style conv stack for 84x84x3 uint8 pixel observations."""

    def __init__(self, channels=(32, 64, 64)):
        super().__init__()
        c1, c2, c3 = channels
        self.net = nn.Sequential(
            nn.Conv2d(3, c1, 8, stride=4), nn.ReLU(inplace=True),
            nn.Conv2d(c1, c2, 4, stride=2), nn.ReLU(inplace=True),
            nn.Conv2d(c2, c3, 3, stride=1), nn.ReLU(inplace=True),
            nn.Flatten(),
        )
        with torch.no_grad():
            self.out_dim = self.net(torch.zeros(1, 3, 84, 84)).shape[1]

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        # x: [N, 84, 84, 3] uint8 (raw) or float (pre-normalized) -> [N, out_dim]
        wdtype = self.net[0].weight.dtype
        if x.dtype == torch.uint8:
            x = x.to(wdtype) / 255.0
        else:
            x = x.to(wdtype)
        return self.net(x.permute(0, 3, 1, 2).contiguous())


class _TrunkEncoder(nn.Module):
    """Flattens [T, B, ...] obs through the trunk to [T, B, F]."""

    def __init__(self, obs_shape, trunk_sizes, conv_channels, pixel: bool):
        super().__init__()
        self.pixel = pixel
        if pixel:
            self.conv = ConvTrunk(conv_channels)
            self.out_dim = self.conv.out_dim
        else:
            in_dim = int(torch.tensor(obs_shape).prod())
            self.mlp = mlp_trunk(in_dim, trunk_sizes)
            self.out_dim = trunk_sizes[-1] if len(trunk_sizes) else in_dim

    def forward(self, obs: torch.Tensor) -> torch.Tensor:
        T, B = obs.shape[0], obs.shape[1]
        flat = obs.reshape(T * B, *obs.shape[2:])
        feat = self.conv(flat) if self.pixel else self.mlp(flat.reshape(T * B, -1))
        return feat.view(T, B, -1)


class ActorNet(nn.Module):
    def __init__(self, obs_shape, action_dim: int, action_scale: float = 1.0,
                 hidden_size: int = 256, trunk_sizes=(256, 256),
                 conv_channels=(32, 64, 64), pixel: bool = False):
        super().__init__()
        self.action_dim = action_dim
        self.action_scale = action_scale
        self.hidden_size = hidden_size
        self.trunk = _TrunkEncoder(obs_shape, trunk_sizes, conv_channels, pixel)
        self.lstm = FusedLSTM(self.trunk.out_dim, hidden_size)
        self.head = nn.Linear(hidden_size, action_dim)
        # DDPG-paper head init (+-3e-3): keeps initial actions near 0 so the
        # tanh does not saturate before the critic is informative (saturated
        # tanh kills the actor gradient permanently)
        with torch.no_grad():
            self.head.weight.uniform_(-3e-3, 3e-3)
            self.head.bias.zero_()

    def zero_state(self, batch: int, device=None, dtype=None) -> State:
        return self.lstm.zero_state(batch, device, dtype)

    def forward(self, obs: torch.Tensor, state: Optional[State] = None):
        """obs: [T, B, ...] -> (action [T, B, A], (h, c))."""
        feat = self.trunk(obs)
        h_seq, st = self.lstm(feat, state)
        a = torch.tanh(self.head(h_seq.reshape(-1, self.hidden_size)))
        return a.view(h_seq.shape[0], h_seq.shape[1], -1) * self.action_scale, st

    def step(self, obs: torch.Tensor, state: Optional[State] = None):
        a, st = self.forward(obs.unsqueeze(0), state)
        return a[0], st


class CriticNet(nn.Module):
    def __init__(self, obs_shape, action_dim: int,
                 hidden_size: int = 256, trunk_sizes=(256, 256),
                 conv_channels=(32, 64, 64), pixel: bool = False):
        super().__init__()
        self.hidden_size = hidden_size
        self.trunk = _TrunkEncoder(obs_shape, trunk_sizes, conv_channels, pixel)
        self.lstm = FusedLSTM(self.trunk.out_dim + action_dim, hidden_size)
        self.head = nn.Linear(hidden_size, 1)
        with torch.no_grad():
            self.head.weight.uniform_(-3e-3, 3e-3)
            self.head.bias.zero_()

    def zero_state(self, batch: int, device=None, dtype=None) -> State:
        return self.lstm.zero_state(batch, device, dtype)

    def forward(self, obs: torch.Tensor, action: torch.Tensor,
                state: Optional[State] = None):
        """obs [T,B,...], action [T,B,A] -> (q [T,B], (h,c))."""
        feat = self.trunk(obs)
        x = torch.cat([feat, action], dim=-1)
        h_seq, st = self.lstm(x, state)
        q = self.head(h_seq.reshape(-1, self.hidden_size))
        return q.view(h_seq.shape[0], h_seq.shape[1]), st

    def step(self, obs: torch.Tensor, action: torch.Tensor,
             state: Optional[State] = None):
        q, st = self.forward(obs.unsqueeze(0), action.unsqueeze(0), state)
        return q[0], st


def make_models(spec, model_cfg, device="cpu", dtype=torch.float32):
    """Build (actor, critic, target_actor, target_critic) for an EnvSpec."""
    kw = dict(hidden_size=model_cfg.hidden_size, trunk_sizes=tuple(model_cfg.trunk_sizes),
              conv_channels=tuple(model_cfg.conv_channels), pixel=spec.pixel)
    actor = ActorNet(spec.obs_shape, spec.action_dim, spec.action_scale, **kw)
    critic = CriticNet(spec.obs_shape, spec.action_dim, **kw)
    actor = actor.to(device=device, dtype=dtype)
    critic = critic.to(device=device, dtype=dtype)
    target_actor = copy.deepcopy(actor).requires_grad_(False)
    target_critic = copy.deepcopy(critic).requires_grad_(False)
    return actor, critic, target_actor, target_critic
