"""Structured metrics logging (SURVEY.md §5 observability).

Per-step scalars as jsonl + stdout summaries. The two headline metrics
(learner transitions/sec whole-node, aggregate actor FPS) are first-class.
"""

from __future__ import annotations

import json
import os
import time
from typing import Optional


class MetricsLogger:
    def __init__(self, path: Optional[str] = None, stdout_every: int = 0, rank: int = 0,
                 tensorboard_dir: Optional[str] = None):
        self.path = path
        self.rank = rank
        self.stdout_every = stdout_every
        self._f = None
        self._tb = None
        if path and rank == 0:
            os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
            self._f = open(path, "a", buffering=1)
        if tensorboard_dir and rank == 0:
            try:  # optional: tensorboard is not in the offline wheelhouse
                from torch.utils.tensorboard import SummaryWriter
                self._tb = SummaryWriter(tensorboard_dir)
            except Exception:  # noqa: BLE001
                print("[metrics] tensorboard unavailable; jsonl only")
        self._n = 0

    def log(self, step: int, **scalars) -> None:
        self._n += 1
        rec = {"step": step, "time": time.time(), **scalars}
        if self._f is not None:
            self._f.write(json.dumps(rec) + "\n")
        if self._tb is not None:
            for k, v in scalars.items():
                if isinstance(v, (int, float)):
                    self._tb.add_scalar(k, v, step)
        if self.stdout_every and self.rank == 0 and self._n % self.stdout_every == 0:
            kv = " ".join(f"{k}={v:.4g}" if isinstance(v, float) else f"{k}={v}"
                          for k, v in scalars.items())
            print(f"[step {step}] {kv}", flush=True)

    def close(self) -> None:
        if self._f is not None:
            self._f.close()
            self._f = None
        if self._tb is not None:
            self._tb.close()
            self._tb = None
