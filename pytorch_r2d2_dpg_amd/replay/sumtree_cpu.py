"""CPU sum-tree for prioritized replay.

Binary indexed complete tree over priorities p_i^alpha, vectorized with
numpy: O(log n) batched update, stratified inverse-CDF batched sampling.
The GPU shard uses the K5/K6 HIP kernels with the same 1-indexed flat-array
layout (tree[1] = root, leaves at [capacity, 2*capacity)), so this class is
also the oracle for their tests.
"""

from __future__ import annotations

from typing import Tuple

import numpy as np


class SumTree:
    def __init__(self, capacity: int):
        # round up to a power of two so the leaf layer is contiguous
        cap = 1
        while cap < capacity:
            cap *= 2
        self.capacity = cap
        self.n_slots = capacity
        self.tree = np.zeros(2 * cap, dtype=np.float64)

    @property
    def total(self) -> float:
        return float(self.tree[1])

    def update(self, idx: np.ndarray, priority: np.ndarray) -> None:
        """Set leaves `idx` (slot indices) to `priority` and re-propagate."""
        idx = np.asarray(idx, dtype=np.int64)
        priority = np.asarray(priority, dtype=np.float64)
        pos = idx + self.capacity
        self.tree[pos] = priority
        # level-by-level parent recomputation (deterministic, no atomics —
        # mirrors the K6 kernel's segmented re-reduction)
        parents = np.unique(pos // 2)
        while parents.size and parents[0] >= 1:
            self.tree[parents] = self.tree[2 * parents] + self.tree[2 * parents + 1]
            if parents[0] == 1:
                break
            parents = np.unique(parents // 2)

    def sample(self, batch: int, rng: np.random.Generator) -> Tuple[np.ndarray, np.ndarray]:
        """Stratified inverse-CDF sampling.

        Returns (slot_idx [batch], probabilities [batch]) where probability is
        the leaf mass / total (for optional IS weights).
        """
        total = self.tree[1]
        if total <= 0:
            raise ValueError("sample from empty sum-tree")
        seg = total / batch
        u = (np.arange(batch) + rng.random(batch)) * seg
        pos = np.ones(batch, dtype=np.int64)
        while pos[0] < self.capacity:  # complete tree: all walks same depth
            left = 2 * pos
            lmass = self.tree[left]
            go_right = u >= lmass
            u = np.where(go_right, u - lmass, u)
            pos = left + go_right
        idx = pos - self.capacity
        probs = self.tree[pos] / total
        return idx, probs

    def leaf(self, idx) -> np.ndarray:
        return self.tree[np.asarray(idx, dtype=np.int64) + self.capacity]
