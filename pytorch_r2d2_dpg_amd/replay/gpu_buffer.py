"""GPU-resident prioritized sequence replay shard.

Storage tensors live in the rank's HBM; the sum-tree over p^alpha lives
beside them and is sampled/updated by the K5/K6 HIP kernels, so a sampled
batch never leaves the GPU (the reference's per-step H2D batch copy is
eliminated — only actor ingest crosses the host boundary, asynchronously).

On CPU (CI, gloo multi-process tests) the same class runs a vectorized torch
implementation of the identical tree layout — the semantics under test are
shared, the kernels have their own GPU numerics tests vs the CPU oracle.

Observations can be stored fp8 e4m3 (K7) to shrink shard bytes (pixels
especially); decode happens inside the fused gather.
"""

from __future__ import annotations

from typing import Dict

import numpy as np
import torch

from .. import ops as _ops


def _pow2(n: int) -> int:
    c = 1
    while c < n:
        c *= 2
    return c


class GpuSequenceReplay:
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
        fp8_obs: bool = False,
        store_dtype: torch.dtype = torch.float32,  # action/state (and float obs)
        seed: int = 0,
    ):
        self.capacity = capacity
        self.cap2 = _pow2(capacity)
        self.stored_len = stored_len
        self.per_alpha = per_alpha
        self.per_beta = per_beta
        self.device = torch.device(device)
        self.is_cuda = self.device.type == "cuda"
        # fp8 compresses FLOAT observations 4->1 byte; uint8 pixels are
        # already 1 byte and e4m3 would both lose precision vs 8-bit and
        # skip the trunk's /255 normalization — ignore the flag for them
        self.fp8_obs = fp8_obs and self.is_cuda and obs_dtype != torch.uint8
        L = stored_len
        if self.fp8_obs:
            store_obs_dtype = torch.uint8
        elif obs_dtype == torch.uint8:
            store_obs_dtype = torch.uint8
        else:
            store_obs_dtype = store_dtype
        self.obs = torch.zeros((capacity, L) + tuple(obs_shape), dtype=store_obs_dtype,
                               device=self.device)
        self.action = torch.zeros((capacity, L, action_dim), dtype=store_dtype,
                                  device=self.device)
        self.reward = torch.zeros((capacity, L), dtype=torch.float32, device=self.device)
        self.done = torch.zeros((capacity, L), dtype=torch.float32, device=self.device)
        self.state = torch.zeros((capacity, 4, hidden_size), dtype=store_dtype,
                                 device=self.device)
        self.length = torch.full((capacity,), stored_len, dtype=torch.int32,
                                 device=self.device)
        # host-visible sticky flag: False until a padded (length < stored_len)
        # sequence is ingested — lets sample() skip returning lengths and the
        # learner skip the loss-mask entirely on full-length workloads (bench)
        self.has_partial = False
        self.tree = torch.zeros(2 * self.cap2, dtype=torch.float32, device=self.device)
        self._next = 0
        self._size = 0
        self._seed = seed
        self._step = 0
        self.total_added = 0

    def __len__(self) -> int:
        return self._size

    # ------------------------------------------------------------- tree ops
    def _tree_update(self, idx: torch.Tensor, prio_alpha: torch.Tensor) -> None:
        if self.is_cuda:
            _ops.hip_ext().sumtree_update(self.tree, self.cap2, idx, prio_alpha)
            return
        # CPU torch path (same level-synchronous recompute)
        pos = idx + self.cap2
        self.tree[pos] = prio_alpha
        depth = self.cap2.bit_length() - 1
        for _ in range(depth):
            pos = torch.unique(pos // 2)
            self.tree[pos] = self.tree[2 * pos] + self.tree[2 * pos + 1]

    def _tree_sample(self, batch: int):
        if self.is_cuda:
            idx, prob = _ops.hip_ext().sumtree_sample(
                self.tree, self.cap2, batch, self._seed, self._step)
            self._step += 1
            return idx, prob
        total = float(self.tree[1])
        if total <= 0:
            raise ValueError("sample from empty replay")
        g = np.random.default_rng(self._seed * 1_000_003 + self._step)
        self._step += 1
        u = torch.from_numpy(((np.arange(batch) + g.random(batch)) * (total / batch))
                             .astype(np.float32))
        pos = torch.ones(batch, dtype=torch.int64)
        while int(pos[0]) < self.cap2:
            left = 2 * pos
            lmass = self.tree[left]
            # drift guard (mirrors sumtree.hip): fp error accumulating in
            # "u -= lmass" can walk into a zero-mass right subtree spanning
            # unwritten slots in [size, cap2) — fall back left there
            go_right = (u >= lmass) & (self.tree[left + 1] > 0)
            u = torch.where(go_right, u - lmass, u)
            pos = left + go_right.long()
        idx = (pos - self.cap2).clamp_(max=max(self._size - 1, 0))
        prob = self.tree[pos] / total
        return idx, prob

    # ------------------------------------------------------------------ add
    @torch.no_grad()
    def add_batch(self, obs, action, reward, done, state, priority,
                  length=None, any_partial: bool = None) -> None:
        """Ingest a batch of sequences (torch tensors, any device).

        ``length`` [n] int = real (non-pad) steps per sequence (None = all
        full). ``any_partial`` lets the caller pass a host-side precomputed
        "any length < stored_len" so no device sync is needed; if omitted it
        is derived from ``length`` (syncing when that tensor is on GPU).
        """
        n = obs.shape[0]
        if n == 0:
            return
        idx = torch.arange(self._next, self._next + n, dtype=torch.int64) % self.capacity
        didx = idx.to(self.device)
        if self.fp8_obs:
            flat = obs.to(self.device, torch.float32, non_blocking=True).reshape(n, -1).contiguous()
            enc = torch.empty(flat.shape, dtype=torch.uint8, device=self.device)
            _ops.hip_ext().fp8_encode(flat, enc, 1.0)
            self.obs[didx] = enc.view(self.obs[didx].shape)
        else:
            self.obs[didx] = obs.to(self.device, self.obs.dtype, non_blocking=True)
        self.action[didx] = action.to(self.device, self.action.dtype, non_blocking=True)
        self.reward[didx] = reward.to(self.device, torch.float32, non_blocking=True)
        self.done[didx] = done.to(self.device, torch.float32, non_blocking=True)
        self.state[didx] = state.to(self.device, self.state.dtype, non_blocking=True)
        if length is not None:
            if not torch.is_tensor(length):
                length = torch.as_tensor(np.asarray(length))
            if any_partial is None:
                src = length.cpu() if length.is_cuda else length
                any_partial = bool((src < self.stored_len).any())
            self.length[didx] = length.to(self.device, torch.int32, non_blocking=True)
            if any_partial:
                self.has_partial = True
        else:
            self.length[didx] = self.stored_len
        if not torch.is_tensor(priority):
            priority = torch.as_tensor(priority)
        p = priority.to(self.device, torch.float32).clamp_min(1e-6) ** self.per_alpha
        # publish-last: the tree update makes the slots visible to sampling
        self._tree_update(didx, p)
        self._next = int((self._next + n) % self.capacity)
        self._size = min(self._size + n, self.capacity)
        self.total_added += n

    # --------------------------------------------------------------- sample
    @torch.no_grad()
    def sample(self, batch: int) -> Dict[str, torch.Tensor]:
        idx, prob = self._tree_sample(batch)
        if self.fp8_obs:
            obs = _ops.hip_ext().fp8_gather_decode(
                self.obs.view(self.capacity, -1), idx, torch.float32, 1.0)
            obs = obs.view((batch,) + self.obs.shape[1:])
        else:
            obs = self.obs.index_select(0, idx)
        if self.per_beta > 0:
            w = (1.0 / (self._size * prob.clamp_min(1e-12))) ** self.per_beta
            w = w / w.max()
        else:
            w = torch.ones(batch, dtype=torch.float32, device=self.device)
        return {
            "idx": idx,
            "obs": obs,
            "action": self.action.index_select(0, idx),
            "reward": self.reward.index_select(0, idx),
            "done": self.done.index_select(0, idx),
            "state": self.state.index_select(0, idx),
            "weight": w,
            # only materialized once a padded sequence exists in the shard —
            # the learner's masked-loss path keys off this being non-None
            "length": (self.length.index_select(0, idx)
                       if self.has_partial else None),
        }

    # ------------------------------------------------------- priority write
    @torch.no_grad()
    def update_priorities(self, idx, priorities) -> None:
        if not torch.is_tensor(idx):
            idx = torch.as_tensor(np.asarray(idx), dtype=torch.int64)
        if not torch.is_tensor(priorities):
            priorities = torch.as_tensor(np.asarray(priorities), dtype=torch.float32)
        idx = idx.to(self.device, torch.int64)
        p = priorities.to(self.device, torch.float32).clamp_min(1e-6) ** self.per_alpha
        self._tree_update(idx, p)
