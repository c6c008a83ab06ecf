"""Data-parallel context — RCCL over xGMI (SURVEY.md §2.3 K8, §5 backend).

One process per GPU; ``torch.distributed`` with backend "nccl" (RCCL on
ROCm) on GPU, "gloo" for CPU tests. Gradients are all-reduced as ONE fused
flat bucket per backward: the models here are ~1-10 M params (2-20 MB bf16),
firmly in RCCL's small-message/latency regime on the 7x153 GB/s xGMI mesh,
so a single flattened buffer per step (not per-tensor calls, not DDP's
25 MB bucketing machinery) minimizes collective count.
"""

from __future__ import annotations

import os
from typing import Iterable, List, Optional

import torch
import torch.distributed as dist


class DistContext:
    def __init__(self, rank: int = 0, world_size: int = 1, initialized: bool = False):
        self.rank = rank
        self.world_size = world_size
        self.initialized = initialized
        self._flat_cache: dict = {}

    # ------------------------------------------------------------ lifecycle
    @staticmethod
    def single() -> "DistContext":
        return DistContext(0, 1, False)

    @staticmethod
    def from_env(backend: Optional[str] = None, device: Optional[torch.device] = None) -> "DistContext":
        """Init from torchrun-style env (RANK/WORLD_SIZE/MASTER_ADDR...)."""
        world = int(os.environ.get("WORLD_SIZE", "1"))
        if world <= 1:
            return DistContext.single()
        rank = int(os.environ.get("RANK", "0"))
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        if not dist.is_initialized():
            kw = {}
            if backend == "nccl":
                local = int(os.environ.get("LOCAL_RANK", rank))
                torch.cuda.set_device(local)
                kw["device_id"] = torch.device("cuda", local)
            dist.init_process_group(backend=backend, **kw)
        return DistContext(rank, world, True)

    def barrier(self) -> None:
        if self.initialized:
            dist.barrier()

    # ---------------------------------------------------------- collectives
    def broadcast_params(self, params: List[torch.Tensor]) -> None:
        if not self.initialized:
            return
        with torch.no_grad():
            flat = torch.cat([p.detach().reshape(-1) for p in params])
            dist.broadcast(flat, src=0)
            off = 0
            for p in params:
                n = p.numel()
                p.detach().copy_(flat[off:off + n].view_as(p))
                off += n

    def allreduce_grads(self, params: Iterable[torch.nn.Parameter], key: str = "") -> None:
        """Fused flat-bucket gradient all-reduce (average). No-op at world=1."""
        if not self.initialized:
            return
        plist = [p for p in params if p.grad is not None]
        if not plist:
            return
        ck = key or id(plist[0])
        buf = self._flat_cache.get(ck)
        total = sum(p.grad.numel() for p in plist)
        if buf is None or buf.numel() != total or buf.dtype != plist[0].grad.dtype:
            buf = torch.empty(total, dtype=plist[0].grad.dtype, device=plist[0].grad.device)
            self._flat_cache[ck] = buf
        off = 0
        for p in plist:
            n = p.grad.numel()
            buf[off:off + n].copy_(p.grad.reshape(-1))
            off += n
        dist.all_reduce(buf)
        buf.div_(self.world_size)
        off = 0
        for p in plist:
            n = p.grad.numel()
            p.grad.copy_(buf[off:off + n].view_as(p.grad))
            off += n

    def allreduce_scalar(self, x: float, op: str = "sum") -> float:
        if not self.initialized:
            return x
        t = torch.tensor([x], dtype=torch.float64)
        if dist.get_backend() == "nccl":
            t = t.cuda()
        dist.all_reduce(t, op=dist.ReduceOp.MAX if op == "max" else dist.ReduceOp.SUM)
        return float(t.item())
