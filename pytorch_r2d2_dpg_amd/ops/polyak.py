"""Fused multi-tensor Polyak soft target update (kernel K4).

theta' <- tau*theta + (1-tau)*theta' over every target parameter in ONE
kernel launch via a device pointer table, instead of the
reference's per-parameter Python loop (2 launches per tensor).
"""

from __future__ import annotations

from typing import List

import torch

from . import hip_ext
from .reference import polyak_ref


class PolyakUpdater:
    """Pre-bakes the pointer/size table for a fixed (target, online) pair."""

    def __init__(self, target_params: List[torch.Tensor], online_params: List[torch.Tensor]):
        self.targets = [p.detach() for p in target_params]
        self.onlines = [p.detach() for p in online_params]
        assert len(self.targets) == len(self.onlines)
        for t, o in zip(self.targets, self.onlines):
            assert t.shape == o.shape and t.dtype == o.dtype, "polyak pair mismatch"
        self._gpu = len(self.targets) > 0 and self.targets[0].is_cuda
        if self._gpu:
            dev = self.targets[0].device
            ptrs, sizes = [], []
            offs = [0]
            for t, o in zip(self.targets, self.onlines):
                assert t.is_contiguous() and o.is_contiguous()
                ptrs.append(t.data_ptr())
                ptrs.append(o.data_ptr())
                sizes.append(t.numel())
                offs.append(offs[-1] + t.numel())
            self.total = offs[-1]
            self.ptr_table = torch.tensor(ptrs, dtype=torch.int64, device=dev)
            self.off_table = torch.tensor(offs, dtype=torch.int64, device=dev)
            self.is_bf16 = self.targets[0].dtype == torch.bfloat16

    @torch.no_grad()
    def __call__(self, tau: float) -> None:
        if self._gpu:
            hip_ext().polyak_multi(self.ptr_table, self.off_table, self.total, float(tau),
                                   bool(self.is_bf16))
        else:
            polyak_ref(self.targets, self.onlines, tau)
