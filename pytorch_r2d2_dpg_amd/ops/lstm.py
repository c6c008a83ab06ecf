"""Fused LSTM sequence op (kernels K1/K2 of SURVEY.md §2.3).

MI355X-native split of the work:

* input projection ``gates_x = x @ Wx^T + b`` — one large, fully-parallel
  library GEMM over [T*B, I] (rocBLAS/hipBLASLt via torch.addmm),
* the sequential recurrence — ONE persistent HIP kernel launch per direction
  (``lstm_seq_fwd`` / ``lstm_seq_bwd``): each workgroup owns a tile of batch
  rows, keeps its (h, c) tile on-CU across all T timesteps, streams Wh
  through the XCD's L2, and computes the per-step ``h @ Wh^T`` on MFMA.
  This replaces the reference's per-timestep ``nn.LSTM`` launches (the
  dominant launch-bound cost — SURVEY.md §3.3),
* weight gradients — two large library GEMMs over the stashed activations.

On CPU the differentiable reference (ops/reference.py:lstm_seq_ref) is used
directly; on GPU the HIP extension is mandatory (ops/__init__.py).
"""

from __future__ import annotations

from typing import Tuple

import torch

from . import hip_ext
from .reference import lstm_seq_ref


class _FusedLSTMSeq(torch.autograd.Function):
    """GPU path. Saves post-activation gates + cell states for backward.

    Outputs (h_seq, c_final) only; the caller takes hT = h_seq[-1] so its
    gradient folds into dh_seq through the select op (no view-output issues).
    """

    @staticmethod
    def forward(ctx, x, h0, c0, wx, wh, b):
        T, B, I = x.shape
        H = h0.shape[1]
        ext = hip_ext()
        # library GEMM: [T*B, I] @ [I, 4H] (+bias)
        gates_x = torch.addmm(b, x.reshape(T * B, I), wx.t()).view(T, B, 4 * H)
        h_seq, c_final, gates_act, c_seq = ext.lstm_seq_fwd(
            gates_x.contiguous(), h0.contiguous(), c0.contiguous(), wh.contiguous()
        )
        ctx.save_for_backward(x, h0, c0, wx, wh, h_seq, gates_act, c_seq)
        return h_seq, c_final

    @staticmethod
    def backward(ctx, dh_seq, dcT):
        x, h0, c0, wx, wh, h_seq, gates_act, c_seq = ctx.saved_tensors
        T, B, I = x.shape
        H = h0.shape[1]
        ext = hip_ext()
        zero_hT = torch.zeros_like(h0)
        dgates, dh0, dc0 = ext.lstm_seq_bwd(
            dh_seq.contiguous(), zero_hT, dcT.contiguous(), gates_act, c_seq,
            c0.contiguous(), wh.contiguous(),
        )
        dg_flat = dgates.reshape(T * B, 4 * H)
        # library GEMMs for the dense gradients
        dx = (dg_flat @ wx).view(T, B, I)
        dwx = dg_flat.t() @ x.reshape(T * B, I)
        h_prev = torch.cat([h0.unsqueeze(0), h_seq[:-1]], dim=0).reshape(T * B, H)
        dwh = dg_flat.t() @ h_prev
        db = dg_flat.float().sum(dim=0).to(dg_flat.dtype)
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
        if not torch.is_grad_enabled():
            # burn-in / target unrolls: no-stash fast path (no activation
            # stores, nothing saved for backward)
            T, B, I = x.shape
            H = h0.shape[1]
            ext = hip_ext()
            gates_x = torch.addmm(b, x.reshape(T * B, I), wx.t()).view(T, B, 4 * H)
            h_seq, cT, _, _ = ext.lstm_seq_fwd(
                gates_x.contiguous(), h0.contiguous(), c0.contiguous(),
                wh.contiguous(), False)
            return h_seq, h_seq[-1], cT
        h_seq, cT = _FusedLSTMSeq.apply(x, h0, c0, wx, wh, b)
        return h_seq, h_seq[-1], cT
    return lstm_seq_ref(x, h0, c0, wx, wh, b)


def lstm_seq_pair(x0, h00, c00, wx0, wh0, b0, x1, h01, c01, wx1, wh1, b1):
    """Two independent no-grad LSTM unrolls (same B, H) in ONE launch
    (burn-in actor + critic). Falls back to two sequential calls when the
    paired geometry is unsupported. Returns (h_seq0, hT0, cT0, h_seq1, hT1, cT1)."""
    if x0.is_cuda and x0.dtype == torch.bfloat16 and not torch.is_grad_enabled():
        T0, B, _ = x0.shape
        T1 = x1.shape[0]
        H = h00.shape[1]
        ext = hip_ext()
        gx0 = torch.addmm(b0, x0.reshape(T0 * B, -1), wx0.t()).view(T0, B, 4 * H)
        gx1 = torch.addmm(b1, x1.reshape(T1 * B, -1), wx1.t()).view(T1, B, 4 * H)
        try:
            hs0, cf0, _, _, hs1, cf1, _, _ = ext.lstm_seq_fwd_pair(
                gx0.contiguous(), h00.contiguous(), c00.contiguous(), wh0.contiguous(),
                gx1.contiguous(), h01.contiguous(), c01.contiguous(), wh1.contiguous())
            return hs0, hs0[-1], cf0, hs1, hs1[-1], cf1
        except RuntimeError:
            pass
    a = lstm_seq(x0, h00, c00, wx0, wh0, b0)
    c = lstm_seq(x1, h01, c01, wx1, wh1, b1)
    return a[0], a[1], a[2], c[0], c[1], c[2]


class _FusedLSTMSeqPairHalf(torch.autograd.Function):
    """Paired launch where set0 is a NO-GRAD unroll (precomputed gates_x) and
    set1 is grad-tracked with stashed activations. Backward is identical to
    _FusedLSTMSeq's for set1; set0's outputs are marked non-differentiable."""

    @staticmethod
    def forward(ctx, x1, h01, c01, wx1, wh1, b1, gx0, h00, c00, wh0):
        T1, B, I1 = x1.shape
        H = h01.shape[1]
        ext = hip_ext()
        gx1 = torch.addmm(b1, x1.reshape(T1 * B, I1), wx1.t()).view(T1, B, 4 * H)
        hs0, cf0, _, _, hs1, cf1, ga1, cs1 = ext.lstm_seq_fwd_pair(
            gx0.contiguous(), h00.contiguous(), c00.contiguous(), wh0.contiguous(),
            gx1.contiguous(), h01.contiguous(), c01.contiguous(), wh1.contiguous(),
            False, True)
        ctx.save_for_backward(x1, h01, c01, wx1, wh1, hs1, ga1, cs1)
        ctx.mark_non_differentiable(hs0, cf0)
        return hs0, cf0, hs1, cf1

    @staticmethod
    def backward(ctx, dhs0, dcf0, dh_seq, dcT):
        x, h0, c0, wx, wh, h_seq, gates_act, c_seq = ctx.saved_tensors
        T, B, I = x.shape
        H = h0.shape[1]
        ext = hip_ext()
        zero_hT = torch.zeros_like(h0)
        dgates, dh0, dc0 = ext.lstm_seq_bwd(
            dh_seq.contiguous(), zero_hT, dcT.contiguous(), gates_act, c_seq,
            c0.contiguous(), wh.contiguous(),
        )
        dg_flat = dgates.reshape(T * B, 4 * H)
        dx = (dg_flat @ wx).view(T, B, I)
        dwx = dg_flat.t() @ x.reshape(T * B, I)
        h_prev = torch.cat([h0.unsqueeze(0), h_seq[:-1]], dim=0).reshape(T * B, H)
        dwh = dg_flat.t() @ h_prev
        db = dg_flat.float().sum(dim=0).to(dg_flat.dtype)
        return dx, dh0, dc0, dwx, dwh, db, None, None, None, None


def lstm_seq_pair_halfgrad(x0, h00, c00, wx0, wh0, b0, x1, h01, c01, wx1, wh1, b1):
    """Set0 runs WITHOUT grad (e.g. the target-actor unroll), set1 WITH grad
    (the online critic forward), in one paired column-split launch; T0 and T1
    may differ. Falls back to two sequential ``lstm_seq`` calls on CPU or
    unsupported geometry. Returns (h_seq0, hT0, cT0, h_seq1, hT1, cT1)."""
    if (x0.is_cuda and x0.dtype == torch.bfloat16 and torch.is_grad_enabled()
            and x1.requires_grad):
        T0, B, I0 = x0.shape
        H = h00.shape[1]
        with torch.no_grad():
            gx0 = torch.addmm(b0, x0.reshape(T0 * B, I0), wx0.t()).view(T0, B, 4 * H)
        try:
            hs0, cf0, hs1, cf1 = _FusedLSTMSeqPairHalf.apply(
                x1, h01, c01, wx1, wh1, b1,
                gx0, h00.detach(), c00.detach(), wh0)
            return hs0, hs0[-1], cf0, hs1, hs1[-1], cf1
        except RuntimeError:
            pass
    with torch.no_grad():
        a = lstm_seq(x0, h00.detach(), c00.detach(), wx0, wh0, b0)
    c = lstm_seq(x1, h01, c01, wx1, wh1, b1)
    return a[0], a[1], a[2], c[0], c[1], c[2]
