"""Actor-side sequence cutter + local-priority computation.

Maintains the rolling transition window and the recurrent-state snapshots;
every ``stride = seq_len - overlap`` steps emits a fixed-length sequence of
``stored_len = seq_len + n_step`` transitions (the n-step tail rides along so
the learner's fused K3 kernel can fold raw rewards on GPU), carrying the
(actor, critic) LSTM states snapshotted at the sequence's first step and an
initial priority from the actor's own step-wise Q values:

    delta_t = fold_n(r)_t + gamma^n * mask_t * q_{t+n} - q_t
    p0      = eta * max|delta| + (1-eta) * mean|delta|

At episode end the window is flushed (right-padded to fixed length by
repeating the last observation with zero reward and zero action) and
reset. Pad semantics depend on WHY the episode ended:

* **Termination** (env terminal, or time limit with
  ``replay.timeout_terminal``): pads are the ABSORBING state — done=1,
  r=0, and they COUNT as real steps (``length`` = stored length), so the
  learner trains Q(absorbing, ·) → 0. This is the standard absorbing-state
  formalism and it is load-bearing: supervising the zero anchor directly
  grounds the critic's value scale orders of magnitude faster than letting
  it propagate only through bootstrapped folds (measured on the Pendulum
  smoke config — without it the critic sinks toward r/(1−γ) for thousands
  of steps while the policy stays degenerate).
* **Truncation** (time limit, PEB): pads keep done=0 and are EXCLUDED
  from the critic/actor losses and the priority reduction via
  ``Sequence.length``; the n-step fold bootstraps through the repeated
  final observation — i.e. from the target net's value of the last real
  state — instead of wrongly treating truncation as termination.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List

import numpy as np


@dataclass
class Sequence:
    obs: np.ndarray        # [L, *obs_shape]
    action: np.ndarray     # [L, A]
    reward: np.ndarray     # [L]
    done: np.ndarray       # [L] 1.0 where episode terminated at that step
    state: np.ndarray      # [4, H]: h_actor, c_actor, h_critic, c_critic at t=0
    priority: float
    length: int            # number of real (non-pad) steps


class SequenceCutter:
    def __init__(self, seq_len: int, overlap: int, burn_in: int, n_step: int,
                 gamma: float, eta: float, hidden_size: int):
        assert 0 <= overlap < seq_len
        self.seq_len = seq_len
        self.stride = seq_len - overlap
        self.burn_in = burn_in
        self.n_step = n_step
        self.stored_len = seq_len + n_step
        self.gamma = gamma
        self.eta = eta
        self.hidden_size = hidden_size
        self._reset_window()

    def _reset_window(self):
        self.obs: List[np.ndarray] = []
        self.act: List[np.ndarray] = []
        self.rew: List[float] = []
        self.done: List[float] = []
        self.q: List[float] = []
        self.states: List[np.ndarray] = []  # [4, H] snapshot BEFORE each step

    def push(self, obs, action, reward, done, q_value, state_snapshot) -> List[Sequence]:
        """Add one transition; returns 0..1 emitted sequences."""
        self.obs.append(np.asarray(obs))
        self.act.append(np.asarray(action, dtype=np.float32))
        self.rew.append(float(reward))
        self.done.append(1.0 if done else 0.0)
        self.q.append(float(q_value))
        self.states.append(np.asarray(state_snapshot, dtype=np.float32))
        out: List[Sequence] = []
        if len(self.obs) >= self.stored_len:
            out.append(self._emit(self.stored_len))
            # slide by stride: keep the overlap + tail for the next sequence
            k = self.stride
            self.obs = self.obs[k:]
            self.act = self.act[k:]
            self.rew = self.rew[k:]
            self.done = self.done[k:]
            self.q = self.q[k:]
            self.states = self.states[k:]
        if done:
            seqs = self.flush(terminated=True)
            out.extend(seqs)
        return out

    def flush(self, terminated: bool = False) -> List[Sequence]:
        """Emit the remaining window (padded) at episode end, then reset.

        ``terminated`` selects absorbing-state pads (trained, done=1) vs
        truncation pads (masked, done=0) — module docstring."""
        out: List[Sequence] = []
        # only emit if there is at least one step beyond burn-in to train on
        if len(self.obs) > self.burn_in + 1:
            out.append(self._emit(len(self.obs), absorbing=terminated))
        self._reset_window()
        return out

    def _emit(self, length: int, absorbing: bool = False) -> Sequence:
        L = self.stored_len
        obs_arr = np.zeros((L,) + self.obs[0].shape, dtype=self.obs[0].dtype)
        act_arr = np.zeros((L, self.act[0].shape[0]), dtype=np.float32)
        rew_arr = np.zeros(L, dtype=np.float32)
        # pad semantics (module docstring): terminated -> absorbing state
        # (done=1, counts as real -> the learner trains Q(absorb)->0);
        # truncated -> done=0, masked, bootstraps through the repeated obs
        done_arr = np.full(L, 1.0 if absorbing else 0.0, dtype=np.float32)
        n = min(length, L)
        obs_arr[:n] = np.stack(self.obs[:n])
        act_arr[:n] = np.stack(self.act[:n])
        rew_arr[:n] = np.asarray(self.rew[:n], dtype=np.float32)
        done_arr[:n] = np.asarray(self.done[:n], dtype=np.float32)
        if n < L:
            obs_arr[n:] = obs_arr[n - 1]  # repeat last obs into the pad
        prio = self._initial_priority(rew_arr, done_arr, np.asarray(self.q[:n], dtype=np.float32), n)
        return Sequence(obs_arr, act_arr, rew_arr, done_arr,
                        self.states[0].copy(), float(prio),
                        L if absorbing else n)

    def _initial_priority(self, rew, done, q, n_real) -> float:
        """n-step TD error against the actor's own step-wise Q estimates.

        Mirrors the learner's K3 semantics on the padded window: pad rewards
        are zero, the alive-mask evolves from the REAL done flags only, and
        the bootstrap beyond the real tail reads the last real Q (the
        repeated-obs pad) — so truncated episodes bootstrap instead of being
        treated as terminated."""
        t_lo = self.burn_in
        t_hi = max(t_lo + 1, min(self.seq_len, n_real))
        deltas = []
        for t in range(t_lo, t_hi):
            acc, g, alive = 0.0, 1.0, 1.0
            for k in range(self.n_step):
                if t + k >= n_real:
                    break  # pad rewards are zero, pad done is 0
                acc += g * alive * rew[t + k]
                alive *= 1.0 - done[t + k]
                g *= self.gamma
            boot = q[t + self.n_step] if (t + self.n_step) < n_real else q[n_real - 1]
            y = acc + (self.gamma ** self.n_step) * alive * boot
            deltas.append(abs(y - q[t]) if t < n_real else 0.0)
        d = np.asarray(deltas, dtype=np.float64)
        if d.size == 0:
            return 1.0
        return self.eta * d.max() + (1.0 - self.eta) * d.mean()
