"""Pinned-host staging ring for actor->GPU replay ingest.

Actors produce numpy sequences on CPU; the learner-side ingest packs them
into a ring of pinned host buffers and issues the H2D copy on a dedicated
side stream (hipMemcpyAsync under torch), so the bulk bytes overlap the
learner's compute. The scatter into the shard + sum-tree publish then run on
the learner's stream after an event wait, preserving the single-writer /
publish-last ordering (§3.4) without ever blocking on the copy.
"""

from __future__ import annotations

from typing import List

import numpy as np
import torch

from .gpu_buffer import GpuSequenceReplay
from .sequence import Sequence


class StagingRing:
    def __init__(self, replay: GpuSequenceReplay, slots: int = 4, slot_seqs: int = 128):
        self.replay = replay
        self.device = replay.device
        self.is_cuda = replay.is_cuda
        self.slot_seqs = slot_seqs
        self.slots = slots
        L = replay.stored_len
        obs_shape = tuple(replay.obs.shape[2:])
        A = replay.action.shape[2]
        H = replay.state.shape[2]
        pin = self.is_cuda

        def mk(shape, dtype=torch.float32):
            return [torch.empty((slot_seqs,) + shape, dtype=dtype, pin_memory=pin)
                    for _ in range(slots)]

        # obs staged in fp32 (fp8 encode happens on-GPU at ingest, K7)
        self._obs = mk((L,) + obs_shape,
                       torch.float32 if replay.fp8_obs else replay.obs.dtype)
        self._act = mk((L, A))
        self._rew = mk((L,))
        self._done = mk((L,))
        self._state = mk((4, H))
        self._prio = mk(())
        self._len = mk((), torch.int32)  # real (non-pad) steps per sequence
        self._events = [torch.cuda.Event() if pin else None for _ in range(slots)]
        self._busy = [False] * slots
        self._cursor = 0
        self.stream = torch.cuda.Stream() if pin else None

    def _next_slot(self) -> int:
        s = self._cursor
        if self._busy[s]:
            self._events[s].synchronize()  # ring full: wait for oldest copy
            self._busy[s] = False
        self._cursor = (s + 1) % self.slots
        return s

    @torch.no_grad()
    def ingest_from_rings(self, rings, max_per_ring: int = 0) -> int:
        """Copy unread slots from ALL shared-memory SeqRings into pinned
        staging, AGGREGATED into full slots before each dispatch — one
        dispatch is ~15 CUDA enqueues, and thousands of tiny dispatches/s
        from the ingest thread contend with the learner on the HIP runtime
        lock (plain memcpy with dtype cast; no pickling).

        max_per_ring bounds the slots taken per ring per call (0 = all):
        the synchronous CPU ingest uses it so a slow learner is not starved
        by draining the actors' whole backlog every step — full rings then
        throttle the actors through drop-newest backpressure."""
        total = 0
        s = None
        fill = 0
        for ring in rings:
            ranges = ring.drain_ranges(max_per_ring)
            for lo, hi in ranges:
                off = lo
                while off < hi:
                    if s is None:
                        s = self._next_slot()
                        fill = 0
                    n = min(hi - off, self.slot_seqs - fill)
                    self._obs[s][fill:fill + n].copy_(ring.obs[off:off + n])
                    self._act[s][fill:fill + n].copy_(ring.act[off:off + n])
                    self._rew[s][fill:fill + n].copy_(ring.rew[off:off + n])
                    self._done[s][fill:fill + n].copy_(ring.done[off:off + n])
                    self._state[s][fill:fill + n].copy_(ring.state[off:off + n])
                    self._prio[s][fill:fill + n].copy_(ring.prio[off:off + n])
                    self._len[s][fill:fill + n].copy_(ring.length[off:off + n])
                    fill += n
                    off += n
                    if fill == self.slot_seqs:
                        total += self._dispatch(s, fill)
                        s = None
            if ranges:
                ring.commit()
        if s is not None and fill:
            total += self._dispatch(s, fill)
        return total

    @torch.no_grad()
    def ingest_arrays(self, obs, act, rew, done, state, prio, length=None) -> int:
        """Stage stacked numpy arrays (the fast path from poll_arrays)."""
        total = 0
        n_all = obs.shape[0]
        L = self.replay.stored_len
        for off in range(0, n_all, self.slot_seqs):
            n = min(self.slot_seqs, n_all - off)
            s = self._next_slot()
            self._obs[s][:n].copy_(torch.from_numpy(obs[off:off + n]).to(self._obs[s].dtype))
            self._act[s][:n].copy_(torch.from_numpy(act[off:off + n]))
            self._rew[s][:n].copy_(torch.from_numpy(rew[off:off + n]))
            self._done[s][:n].copy_(torch.from_numpy(done[off:off + n]))
            self._state[s][:n].copy_(torch.from_numpy(state[off:off + n]))
            self._prio[s][:n].copy_(torch.from_numpy(prio[off:off + n]))
            if length is None:
                self._len[s][:n].fill_(L)
            else:
                self._len[s][:n].copy_(torch.from_numpy(length[off:off + n]).to(torch.int32))
            total += self._dispatch(s, n)
        return total

    @torch.no_grad()
    def ingest(self, seqs: List[Sequence]) -> int:
        """Stage and enqueue sequences into the replay shard. Returns count."""
        total = 0
        for off in range(0, len(seqs), self.slot_seqs):
            chunk = seqs[off:off + self.slot_seqs]
            n = len(chunk)
            s = self._next_slot()
            obs_t = self._obs[s][:n]
            obs_np = np.stack([q.obs for q in chunk])
            obs_t.copy_(torch.from_numpy(obs_np).to(obs_t.dtype))
            self._act[s][:n].copy_(torch.from_numpy(np.stack([q.action for q in chunk])))
            self._rew[s][:n].copy_(torch.from_numpy(np.stack([q.reward for q in chunk])))
            self._done[s][:n].copy_(torch.from_numpy(np.stack([q.done for q in chunk])))
            self._state[s][:n].copy_(torch.from_numpy(np.stack([q.state for q in chunk])))
            self._prio[s][:n].copy_(torch.tensor([q.priority for q in chunk]))
            self._len[s][:n].copy_(torch.tensor([q.length for q in chunk],
                                                dtype=torch.int32))
            total += self._dispatch(s, n)
        return total

    @torch.no_grad()
    def _dispatch(self, s: int, n: int) -> int:
        # host-side partial check (pinned buffer, pre-copy): lets the replay
        # skip the loss-mask path entirely until a padded sequence exists
        any_partial = bool((self._len[s][:n] < self.replay.stored_len).any())
        if self.is_cuda:
            # bulk H2D on the side stream, scatter on the learner stream
            with torch.cuda.stream(self.stream):
                dev = [t[:n].to(self.device, non_blocking=True)
                       for t in (self._obs[s], self._act[s], self._rew[s],
                                 self._done[s], self._state[s], self._prio[s],
                                 self._len[s])]
                self._events[s].record(self.stream)
            self._busy[s] = True
            cur = torch.cuda.current_stream()
            cur.wait_event(self._events[s])
            # dev tensors are ALLOCATED on the side stream but CONSUMED by the
            # scatter on this stream: without record_stream the allocator may
            # recycle their memory for the next slot's H2D before the pending
            # scatter reads it (observed as NaN training under the threaded
            # ingest — silent replay corruption)
            for t in dev:
                t.record_stream(cur)
            self.replay.add_batch(*dev[:6], length=dev[6], any_partial=any_partial)
        else:
            self.replay.add_batch(self._obs[s][:n], self._act[s][:n],
                                  self._rew[s][:n], self._done[s][:n],
                                  self._state[s][:n], self._prio[s][:n],
                                  length=self._len[s][:n], any_partial=any_partial)
        return n
