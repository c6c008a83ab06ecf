"""FusedLSTM module — single-layer LSTM over the fused sequence op.

Replaces the reference's ``nn.LSTM`` (the reference's GPU
LSTM math was cuDNN via nn.LSTM — here it is the K1/K2 HIP kernels on GPU and
the fp32 reference on CPU). Weight layout is the torch LSTM convention
(``[4H, I]`` / ``[4H, H]``, gate order i,f,g,o) with a single fused bias.
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
from torch import nn

from ..ops.lstm import lstm_seq

State = Tuple[torch.Tensor, torch.Tensor]


class FusedLSTM(nn.Module):
    def __init__(self, input_size: int, hidden_size: int):
        super().__init__()
        self.input_size = input_size
        self.hidden_size = hidden_size
        self.weight_x = nn.Parameter(torch.empty(4 * hidden_size, input_size))
        self.weight_h = nn.Parameter(torch.empty(4 * hidden_size, hidden_size))
        self.bias = nn.Parameter(torch.zeros(4 * hidden_size))
        stdv = 1.0 / math.sqrt(hidden_size)
        with torch.no_grad():
            self.weight_x.uniform_(-stdv, stdv)
            self.weight_h.uniform_(-stdv, stdv)
            self.bias.uniform_(-stdv, stdv)

    def zero_state(self, batch: int, device=None, dtype=None) -> State:
        p = self.weight_x
        device = device or p.device
        dtype = dtype or p.dtype
        z = torch.zeros(batch, self.hidden_size, device=device, dtype=dtype)
        return z, z.clone()

    def forward(self, x: torch.Tensor, state: Optional[State] = None):
        """x: [T, B, I]; returns (h_seq [T,B,H], (hT, cT))."""
        if state is None:
            state = self.zero_state(x.shape[1], device=x.device, dtype=x.dtype)
        h0, c0 = state
        h_seq, hT, cT = lstm_seq(x, h0, c0, self.weight_x, self.weight_h, self.bias)
        return h_seq, (hT, cT)

    def step(self, x: torch.Tensor, state: Optional[State] = None):
        """x: [B, I] single step; returns (h [B,H], (h, c))."""
        h_seq, st = self.forward(x.unsqueeze(0), state)
        return h_seq[0], st
