"""Shared-memory sequence ring: the zero-pickle actor->learner data plane.

Each actor process owns one ring of K fixed-size sequence slots in shared
memory (torch shared tensors). The actor writes a slot and bumps its write
counter; the learner-side ingest thread batch-copies the new slots straight
into the pinned staging buffers (plain memcpy — the mp.Queue pickling of
~10 KB numpy arrays was the ingest bottleneck at ~50k aggregate actor FPS).

Single-producer / single-consumer per ring: the write counter (mp.Value with
lock) is the publish point; the consumer updates the read counter so the
producer can apply backpressure (drop-newest, matching the old queue-full
behaviour).
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import numpy as np
import torch
import torch.multiprocessing as mp


class SeqRing:
    def __init__(self, ctx, slots: int, stored_len: int, obs_shape, obs_dtype,
                 action_dim: int, hidden_size: int):
        self.slots = slots
        L = stored_len
        t_obs_dtype = torch.uint8 if obs_dtype == np.uint8 else torch.float32

        def shared(shape, dtype=torch.float32):
            t = torch.zeros(shape, dtype=dtype)
            t.share_memory_()
            return t

        self.obs = shared((slots, L) + tuple(obs_shape), t_obs_dtype)
        self.act = shared((slots, L, action_dim))
        self.rew = shared((slots, L))
        self.done = shared((slots, L))
        self.state = shared((slots, 4, hidden_size))
        self.prio = shared((slots,))
        self.length = shared((slots,), torch.int32)  # real (non-pad) steps
        self.wr = ctx.Value("l", 0)   # written-count (publish point)
        self.rd = ctx.Value("l", 0)   # consumed-count (backpressure)
        self.dropped = ctx.Value("l", 0)

    # ------------------------------------------------------------- producer
    def push(self, seq) -> bool:
        """Write one sequence (actor side). Returns False if full (dropped)."""
        with self.wr.get_lock():
            wr = self.wr.value
        with self.rd.get_lock():
            rd = self.rd.value
        if wr - rd >= self.slots:
            with self.dropped.get_lock():
                self.dropped.value += 1
            return False
        s = wr % self.slots
        self.obs[s].copy_(torch.from_numpy(np.ascontiguousarray(seq.obs)))
        self.act[s].copy_(torch.from_numpy(seq.action))
        self.rew[s].copy_(torch.from_numpy(seq.reward))
        self.done[s].copy_(torch.from_numpy(seq.done))
        self.state[s].copy_(torch.from_numpy(seq.state))
        self.prio[s] = float(seq.priority)
        self.length[s] = int(seq.length)
        with self.wr.get_lock():
            self.wr.value = wr + 1    # publish AFTER the slot is fully written
        return True

    # ------------------------------------------------------------- consumer
    def drain_ranges(self, max_slots: int = 0) -> List[Tuple[int, int]]:
        """Claim unread slots; returns up to two contiguous [lo, hi) ranges of
        slot indices (the caller copies them, then calls `commit(n)`)."""
        with self.wr.get_lock():
            wr = self.wr.value
        rd = self.rd.value  # consumer-owned; no other writer
        n = wr - rd
        if max_slots:
            n = min(n, max_slots)
        if n <= 0:
            return []
        lo = rd % self.slots
        first = min(n, self.slots - lo)
        out = [(lo, lo + first)]
        if n > first:
            out.append((0, n - first))
        self._pending = n
        return out

    def commit(self) -> None:
        with self.rd.get_lock():
            self.rd.value += self._pending
        self._pending = 0
