"""Actor process pool (SURVEY.md §3.1/§5 failure detection).

Spawns N CPU actor processes per learner rank (each optionally vectorized
over K envs). Each actor runs the rollout loop (actor.py), writes cut
sequences into its zero-pickle shared-memory ring (seq_ring.py; episode
returns go through a light queue), and pulls fresh weights from a
shared-memory flat parameter buffer (version-stamped).

Elasticity: actors are stateless beyond their env, so the pool heartbeats
each actor (env-step watermark) and respawns dead/stalled ones with a fresh
seed. A test-only fault-injection hook (`crash_at`) kills actor i at step s
to exercise that path. Learner ranks are NOT elastic (fixed RCCL world).
"""

from __future__ import annotations

import queue as queue_mod
import time
from typing import Dict, List, Optional, Tuple

import numpy as np
import torch
import torch.multiprocessing as mp

from ..actor import ActorWorker, VectorActorWorker
from ..config import Config
from ..replay.sequence import Sequence
from .seq_ring import SeqRing


# ---------------------------------------------------------- sd <-> flat buf
def flat_size(sd: Dict[str, torch.Tensor]) -> int:
    return sum(v.numel() for v in sd.values())


def pack_sd(sd: Dict[str, torch.Tensor], buf: torch.Tensor) -> None:
    off = 0
    for k in sorted(sd):
        v = sd[k]
        n = v.numel()
        buf[off:off + n].copy_(v.detach().float().reshape(-1))
        off += n


def unpack_sd(template: Dict[str, torch.Tensor], buf: torch.Tensor) -> Dict[str, torch.Tensor]:
    out = {}
    off = 0
    for k in sorted(template):
        v = template[k]
        n = v.numel()
        out[k] = buf[off:off + n].view(v.shape).clone()
        off += n
    return out


# -------------------------------------------------------------- actor proc
def _actor_main(actor_id: int, slot: int, cfg_dict: dict, seq_queue, weight_buf,
                weight_ver, stop_event, heartbeat, crash_at: int, chunk: int = 64,
                eval_mode: bool = False, ring=None):
    # actor_id is GLOBAL (seeds, sigma ladder); slot is the LOCAL pool index
    # (heartbeat array) — on learner rank r>0 they differ
    cfg = Config.from_dict(cfg_dict)
    torch.set_num_threads(max(cfg.actor.torch_threads, 1))
    E = max(cfg.actor.envs_per_actor, 1)
    if E > 1 and not eval_mode:
        worker = VectorActorWorker(cfg, actor_id=actor_id, num_envs=E)
    else:
        worker = ActorWorker(cfg, actor_id=actor_id, eval_mode=eval_mode)
    a_tpl = worker.actor_net.state_dict()
    c_tpl = worker.critic_net.state_dict()
    na = flat_size(a_tpl)
    last_ver = -1
    ep_reported = 0
    while not stop_event.is_set():
        ver = int(weight_ver.value)
        if ver != last_ver and ver > 0 and ver % 2 == 0:
            a_sd = unpack_sd(a_tpl, weight_buf[:na])
            c_sd = unpack_sd(c_tpl, weight_buf[na:])
            if int(weight_ver.value) == ver:  # seqlock: no publish overlapped
                worker.load_weights(a_sd, c_sd)
                last_ver = ver
        seqs = worker.rollout(chunk)
        if crash_at and worker.env_steps >= crash_at:
            raise RuntimeError(f"fault injection: actor {actor_id} crash at {crash_at}")
        if eval_mode:
            seqs = []  # noise-free eval actors report episode returns only
        for sq in seqs:
            # zero-pickle data plane: write straight into the shared-memory
            # ring (drop-newest on backpressure, as the old queue did)
            ring.push(sq)
        new_eps = worker.episode_returns[ep_reported:]
        if new_eps:
            try:
                seq_queue.put(("eval_episodes" if eval_mode else "episodes",
                               actor_id, new_eps), timeout=0.1)
                ep_reported = len(worker.episode_returns)
            except queue_mod.Full:
                pass
        heartbeat[slot] = worker.env_steps


class ActorPool:
    def __init__(self, cfg: Config, num_actors: int, base_actor_id: int = 0,
                 crash_at: Optional[Dict[int, int]] = None, chunk: int = 64,
                 queue_size: int = 512, eval_actors: int = 0):
        self.cfg = cfg
        self.num_actors = num_actors + eval_actors
        self._eval_from = num_actors  # pool indices >= this run noise-free
        self.base_actor_id = base_actor_id
        self.crash_at = crash_at or {}
        self.chunk = chunk
        self._ctx = mp.get_context("spawn")
        self.seq_queue = self._ctx.Queue(maxsize=queue_size)
        self.stop_event = self._ctx.Event()
        self.heartbeat = self._ctx.Array("l", self.num_actors)
        self.weight_ver = self._ctx.Value("l", 0)
        # shared flat weight buffer sized from a probe worker
        probe = ActorWorker(cfg, actor_id=0)
        self._a_tpl = probe.actor_net.state_dict()
        self._c_tpl = probe.critic_net.state_dict()
        self._na = flat_size(self._a_tpl)
        self._nc = flat_size(self._c_tpl)
        self.weight_buf = torch.zeros(self._na + self._nc, dtype=torch.float32)
        self.weight_buf.share_memory_()
        # per-actor shared-memory sequence rings (zero-pickle data plane)
        spec = probe.spec
        L = cfg.replay.seq_len + cfg.replay.n_step
        slot_bytes = (int(np.prod(spec.obs_shape)) * (1 if spec.pixel else 4)
                      + spec.action_dim * 4 + 8) * L + cfg.model.hidden_size * 16
        ring_slots = int(min(256, max(16, (32 << 20) // max(slot_bytes, 1))))
        self.rings = [SeqRing(self._ctx, ring_slots, L, spec.obs_shape,
                              np.uint8 if spec.pixel else np.float32,
                              spec.action_dim, cfg.model.hidden_size)
                      for _ in range(self.num_actors)]
        del probe
        self.procs: List[Optional[mp.Process]] = [None] * self.num_actors
        self._last_beat = [0] * self.num_actors
        self._last_beat_time = [time.time()] * self.num_actors
        self.respawn_count = 0
        self.episode_returns: List[Tuple[int, float]] = []
        self.eval_returns: List[Tuple[int, float]] = []
        self.total_sequences = 0

    # ------------------------------------------------------------ lifecycle
    def start(self) -> None:
        for i in range(self.num_actors):
            self._spawn(i)

    def _spawn(self, i: int) -> None:
        p = self._ctx.Process(
            target=_actor_main,
            args=(self.base_actor_id + i, i, self.cfg.to_dict(), self.seq_queue,
                  self.weight_buf, self.weight_ver, self.stop_event, self.heartbeat,
                  self.crash_at.get(self.base_actor_id + i, 0), self.chunk,
                  i >= self._eval_from, self.rings[i]),
            daemon=True,
        )
        p.start()
        self.procs[i] = p
        self._last_beat[i] = -1
        self._last_beat_time[i] = time.time()

    def stop(self) -> None:
        self.stop_event.set()
        deadline = time.time() + 5
        for p in self.procs:
            if p is not None:
                p.join(timeout=max(0.1, deadline - time.time()))
                if p.is_alive():
                    p.terminate()

    # ------------------------------------------------------------- data i/o
    def drain_episodes(self, max_msgs: int = 256) -> None:
        """Drain the (light) episode-return queue."""
        for _ in range(max_msgs):
            try:
                payload = self.seq_queue.get_nowait()
            except queue_mod.Empty:
                break
            tag, aid, eps = payload
            if tag == "eval_episodes":
                self.eval_returns.extend((aid, r) for r in eps)
            else:
                self.episode_returns.extend((aid, r) for r in eps)

    def poll_arrays(self, max_batches: int = 64) -> List[tuple]:
        """Drain the shared-memory rings as stacked numpy arrays (copies)."""
        self.drain_episodes()
        out: List[tuple] = []
        for ring in self.rings:
            ranges = ring.drain_ranges()
            for lo, hi in ranges:
                out.append((ring.obs[lo:hi].numpy().copy(),
                            ring.act[lo:hi].numpy().copy(),
                            ring.rew[lo:hi].numpy().copy(),
                            ring.done[lo:hi].numpy().copy(),
                            ring.state[lo:hi].numpy().copy(),
                            ring.prio[lo:hi].numpy().copy(),
                            ring.length[lo:hi].numpy().copy()))
                self.total_sequences += hi - lo
            if ranges:
                ring.commit()
        return out

    def poll(self, max_batches: int = 64) -> List[Sequence]:
        """Drain available sequences as objects (tests / small runs)."""
        out: List[Sequence] = []
        for batch in self.poll_arrays(max_batches):
            obs, act, rew, done, state, prio, length = batch
            for i in range(obs.shape[0]):
                out.append(Sequence(obs[i], act[i], rew[i], done[i], state[i],
                                    float(prio[i]), int(length[i])))
        return out

    def ingest_rings(self, staging, max_per_ring: int = 0) -> int:
        """Zero-pickle fast path: memcpy unread ring slots straight into the
        pinned staging buffers. Called from the learner-side ingest thread
        (unbounded) or synchronously between CPU learner steps (bounded)."""
        return staging.ingest_from_rings(self.rings, max_per_ring)

    def publish_weights(self, actor_sd: Dict[str, torch.Tensor],
                        critic_sd: Dict[str, torch.Tensor]) -> None:
        # seqlock: odd version = pack in progress; actors accept only even
        # versions that are unchanged across their read, so a read that
        # overlaps a publish is detected and retried (no torn weights)
        with self.weight_ver.get_lock():
            self.weight_ver.value += 1  # -> odd
        pack_sd(actor_sd, self.weight_buf[:self._na])
        pack_sd(critic_sd, self.weight_buf[self._na:])
        with self.weight_ver.get_lock():
            self.weight_ver.value += 1  # -> even (stable)

    # ------------------------------------------------- failure detection
    def watchdog(self, stall_seconds: float = 60.0) -> int:
        """Respawn dead or stalled actors; returns number respawned."""
        n = 0
        now = time.time()
        for i, p in enumerate(self.procs):
            beat = self.heartbeat[i]
            if beat != self._last_beat[i]:
                self._last_beat[i] = beat
                self._last_beat_time[i] = now
            dead = p is None or not p.is_alive()
            stalled = (now - self._last_beat_time[i]) > stall_seconds
            if dead or stalled:
                if p is not None and p.is_alive():
                    p.terminate()
                self.crash_at.pop(self.base_actor_id + i, None)  # don't re-crash
                self._spawn(i)
                self.respawn_count += 1
                n += 1
        return n

    def aggregate_env_steps(self) -> int:
        return int(sum(self.heartbeat))

    def aggregate_drops(self) -> int:
        """Total sequences dropped to ring backpressure (drop-newest)."""
        return int(sum(r.dropped.value for r in self.rings))
