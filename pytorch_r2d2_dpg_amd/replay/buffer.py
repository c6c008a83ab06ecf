"""Prioritized sequence replay.

Ring buffer of fixed-length sequences in flat preallocated tensors that live
directly on the learner's device — on GPU this IS the HBM-resident replay
shard (one shard per DP rank; 288 GB HBM3E sizes capacity).

Priorities p^alpha live in a sum-tree. Storage writes and tree updates happen
on the single learner/ingest thread (single-writer, publish-tree-last), which
preserves the reference's concurrency contract by construction (§3.4).

Sampling: stratified inverse-CDF descent. On CPU (and as the v1 GPU path) the
descent runs on the CPU tree; the K5/K6 HIP kernels move it on-device so the
sampled batch gather never leaves HBM.
"""

from __future__ import annotations

from typing import Dict, Sequence as Seq

import numpy as np
import torch

from .sequence import Sequence
from .sumtree_cpu import SumTree


class SequenceReplay:
    def __init__(
        self,
        capacity: int,
        stored_len: int,
        obs_shape,
        action_dim: int,
        hidden_size: int,
        per_alpha: float = 0.9,
        per_beta: float = 0.0,
        device: str = "cpu",
        obs_dtype: torch.dtype = torch.float32,
        seed: int = 0,
    ):
        self.capacity = capacity
        self.stored_len = stored_len
        self.per_alpha = per_alpha
        self.per_beta = per_beta
        self.device = torch.device(device)
        L = stored_len
        self.obs = torch.zeros((capacity, L) + tuple(obs_shape), dtype=obs_dtype, device=self.device)
        self.action = torch.zeros((capacity, L, action_dim), dtype=torch.float32, device=self.device)
        self.reward = torch.zeros((capacity, L), dtype=torch.float32, device=self.device)
        self.done = torch.zeros((capacity, L), dtype=torch.float32, device=self.device)
        self.state = torch.zeros((capacity, 4, hidden_size), dtype=torch.float32, device=self.device)
        self.length = torch.full((capacity,), stored_len, dtype=torch.int32,
                                 device=self.device)
        self.has_partial = False  # sticky: any padded sequence ingested
        self.tree = SumTree(capacity)
        self._next = 0
        self._size = 0
        self._rng = np.random.default_rng(seed)
        self.total_added = 0

    def __len__(self) -> int:
        return self._size

    # ------------------------------------------------------------------ add
    def add(self, seqs: Seq[Sequence]) -> None:
        """Ingest sequences (numpy, from actors). Single-writer only."""
        if not seqs:
            return
        n = len(seqs)
        idx = (self._next + np.arange(n)) % self.capacity
        obs = torch.from_numpy(np.stack([s.obs for s in seqs]))
        act = torch.from_numpy(np.stack([s.action for s in seqs]))
        rew = torch.from_numpy(np.stack([s.reward for s in seqs]))
        done = torch.from_numpy(np.stack([s.done for s in seqs]))
        state = torch.from_numpy(np.stack([s.state for s in seqs]))
        ti = torch.from_numpy(idx)
        self.obs[ti] = obs.to(self.device, self.obs.dtype, non_blocking=True)
        self.action[ti] = act.to(self.device, non_blocking=True)
        self.reward[ti] = rew.to(self.device, non_blocking=True)
        self.done[ti] = done.to(self.device, non_blocking=True)
        self.state[ti] = state.to(self.device, non_blocking=True)
        lengths = np.asarray([s.length for s in seqs], dtype=np.int32)
        self.length[ti] = torch.from_numpy(lengths).to(self.device)
        if (lengths < self.stored_len).any():
            self.has_partial = True
        prios = np.asarray([max(s.priority, 1e-6) for s in seqs], dtype=np.float64)
        # publish-last: tree update makes the slots visible to sampling
        self.tree.update(idx, prios ** self.per_alpha)
        self._next = int((self._next + n) % self.capacity)
        self._size = min(self._size + n, self.capacity)
        self.total_added += n

    # --------------------------------------------------------------- sample
    def sample(self, batch: int) -> Dict[str, torch.Tensor]:
        idx, probs = self.tree.sample(batch, self._rng)
        ti = torch.from_numpy(idx).to(self.device)
        if self.per_beta > 0:
            w = (1.0 / (self._size * np.maximum(probs, 1e-12))) ** self.per_beta
            w = w / w.max()
        else:
            w = np.ones(batch)
        return {
            "idx": idx,
            "obs": self.obs[ti],
            "action": self.action[ti],
            "reward": self.reward[ti],
            "done": self.done[ti],
            "state": self.state[ti],
            "weight": torch.from_numpy(w.astype(np.float32)).to(self.device),
            "length": self.length[ti] if self.has_partial else None,
        }

    # ------------------------------------------------------- priority write
    def update_priorities(self, idx, priorities) -> None:
        if torch.is_tensor(idx):
            idx = idx.cpu().numpy()
        if torch.is_tensor(priorities):
            priorities = priorities.detach().cpu().numpy()
        p = np.maximum(np.asarray(priorities, dtype=np.float64), 1e-6)
        self.tree.update(np.asarray(idx, dtype=np.int64), p ** self.per_alpha)
