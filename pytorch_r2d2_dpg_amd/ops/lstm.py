"""Fused LSTM sequence op (kernels K1/K2 of SURVEY.md §2.3).

MI355X-native split of the work:

* input projection ``gates_x = x @ Wx^T + b`` — one large, fully-parallel
  library GEMM over [T*B, I] (rocBLAS/hipBLASLt via torch.matmul),
* the sequential recurrence — ONE persistent HIP kernel launch per direction
  (``lstm_seq_fwd`` / ``lstm_seq_bwd``): each workgroup owns a tile of batch
  rows, keeps its (h, c) tile in LDS across all T timesteps, streams Wh
  through the XCD's L2, and computes the per-step ``h @ Wh^T`` on MFMA.
  This replaces the reference's per-timestep ``nn.LSTM`` launches (the
  dominant launch-bound cost — SURVEY.md §3.3),
* weight gradients — two large library GEMMs over the stashed activations.

On CPU the differentiable reference (ops/reference.py:lstm_seq_ref) is used
directly; on GPU the HIP extension is mandatory (ops/__init__.py).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from . import hip_ext
from .reference import lstm_seq_ref


class _FusedLSTMSeq(torch.autograd.Function):
    """GPU path. Saves post-activation gates + cell states for backward."""

    @staticmethod
    def forward(ctx, x, h0, c0, wx, wh, b):
        T, B, I = x.shape
        H = h0.shape[1]
        ext = hip_ext()
        # library GEMM: [T*B, I] @ [I, 4H] (+bias)
        gates_x = torch.addmm(b, x.reshape(T * B, I), wx.t()).view(T, B, 4 * H)
        h_seq, c_final, gates_act, c_seq = ext.lstm_seq_fwd(gates_x, h0, c0, wh)
        ctx.save_for_backward(x, h0, c0, wx, wh, h_seq, gates_act, c_seq)
        return h_seq, h_seq[-1], c_final

    @staticmethod
    def backward(ctx, dh_seq, dhT, dcT):
        x, h0, c0, wx, wh, h_seq, gates_act, c_seq = ctx.saved_tensors
        T, B, I = x.shape
        H = h0.shape[1]
        ext = hip_ext()
        dh_seq = dh_seq.contiguous()
        # persistent reverse-time kernel: dgates [T,B,4H], dh0, dc0
        dgates, dh0, dc0 = ext.lstm_seq_bwd(
            dh_seq, dhT.contiguous(), dcT.contiguous(), gates_act, c_seq, c0, wh
        )
        dg_flat = dgates.reshape(T * B, 4 * H)
        # library GEMMs for the dense gradients
        dx = (dg_flat @ wx).view(T, B, I)
        dwx = dg_flat.t() @ x.reshape(T * B, I)
        h_prev = torch.cat([h0.unsqueeze(0), h_seq[:-1]], dim=0).reshape(T * B, H)
        dwh = dg_flat.t() @ h_prev
        db = dg_flat.sum(dim=0)
        return dx, dh0, dc0, dwx, dwh, db


def lstm_seq(
    x: torch.Tensor,
    h0: torch.Tensor,
    c0: torch.Tensor,
    wx: torch.Tensor,
    wh: torch.Tensor,
    b: torch.Tensor,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Run an LSTM over x [T,B,I] from state (h0, c0) [B,H].

    Returns (h_seq [T,B,H], hT [B,H], cT [B,H]).
    """
    if x.is_cuda:
        return _FusedLSTMSeq.apply(x, h0, c0, wx, wh, b)
    return lstm_seq_ref(x, h0, c0, wx, wh, b)
