"""Unit tests: sum-tree vs cumsum oracle, n-step fold vs brute force,
sequence cutter boundary cases, priority eta-mix (SURVEY.md §4 Unit)."""

import numpy as np
import pytest
import torch

from pytorch_r2d2_dpg_amd.ops.reference import (
    fold_nstep_ref, sequence_priority_ref, value_rescale, value_rescale_inv,
)
from pytorch_r2d2_dpg_amd.replay.buffer import SequenceReplay
from pytorch_r2d2_dpg_amd.replay.sequence import Sequence, SequenceCutter
from pytorch_r2d2_dpg_amd.replay.sumtree_cpu import SumTree


# ------------------------------------------------------------------ sum-tree
def test_sumtree_total_matches_leaf_sum():
    rng = np.random.default_rng(0)
    st = SumTree(1000)
    idx = rng.integers(0, 1000, size=500)
    pri = rng.random(500) + 0.01
    st.update(idx, pri)
    # duplicates: last write wins per slot; emulate with dict
    want = np.zeros(1024)
    for i, p in zip(idx, pri):
        want[i] = p
    assert np.isclose(st.total, want.sum())
    assert np.allclose(st.leaf(np.arange(1000)), want[:1000])


def test_sumtree_sampling_distribution():
    rng = np.random.default_rng(1)
    st = SumTree(64)
    pri = rng.random(64) + 0.05
    st.update(np.arange(64), pri)
    counts = np.zeros(64)
    n_rounds = 400
    for _ in range(n_rounds):
        idx, probs = st.sample(64, rng)
        np.add.at(counts, idx, 1)
        # probs must equal leaf/total
        assert np.allclose(probs, pri[idx] / pri.sum())
    freq = counts / counts.sum()
    expect = pri / pri.sum()
    # chi-square-ish closeness
    assert np.abs(freq - expect).max() < 0.02


def test_sumtree_update_overwrites():
    st = SumTree(8)
    st.update(np.array([0, 1]), np.array([1.0, 2.0]))
    st.update(np.array([1]), np.array([5.0]))
    assert np.isclose(st.total, 6.0)


# --------------------------------------------------------------- n-step fold
def test_fold_nstep_matches_bruteforce():
    rng = np.random.default_rng(2)
    B, T_ext, n, gamma, T = 7, 20, 5, 0.97, 12
    r = torch.tensor(rng.standard_normal((B, T_ext)), dtype=torch.float32)
    d = torch.zeros(B, T_ext)
    d[2, 8] = 1.0
    d[5, 0] = 1.0
    folded, mask = fold_nstep_ref(r, d, gamma, n, T)
    for b in range(B):
        for t in range(T):
            acc, g, alive = 0.0, 1.0, 1.0
            for k in range(n):
                if t + k >= T_ext:
                    alive = 0.0
                    break
                acc += g * alive * r[b, t + k].item()
                alive *= 1.0 - d[b, t + k].item()
                g *= gamma
            assert abs(folded[b, t].item() - acc) < 1e-5, (b, t)
            assert abs(mask[b, t].item() - alive) < 1e-6, (b, t)


def test_priority_eta_mix():
    td = torch.tensor([[1.0, 2.0, 3.0], [0.5, 0.5, 0.5]])
    p = sequence_priority_ref(td, eta=0.9)
    assert np.isclose(p[0].item(), 0.9 * 3.0 + 0.1 * 2.0)
    assert np.isclose(p[1].item(), 0.5)


def test_value_rescale_roundtrip():
    x = torch.linspace(-50, 50, 101, dtype=torch.float64)
    y = value_rescale_inv(value_rescale(x))
    assert torch.allclose(x, y, atol=1e-6)


# ----------------------------------------------------------- sequence cutter
def _mk_cutter(seq_len=8, overlap=4, burn_in=2, n_step=2, H=4):
    return SequenceCutter(seq_len, overlap, burn_in, n_step, 0.99, 0.9, H)


def test_cutter_emits_at_stride():
    c = _mk_cutter()
    emitted = []
    for t in range(40):
        emitted += c.push(np.full(3, t, dtype=np.float32), np.zeros(2, np.float32),
                          0.1, False, 0.0, np.zeros((4, 4), np.float32))
    # stored_len = 10, stride = 4: first at t=9, then every 4 steps
    assert len(emitted) == 1 + (40 - 10) // 4
    s0 = emitted[0]
    assert s0.obs.shape == (10, 3)
    assert s0.length == 10
    # overlap check: second sequence starts stride=4 steps later
    assert emitted[1].obs[0, 0] == 4.0
    assert s0.obs[0, 0] == 0.0


def test_cutter_flush_on_done_pads():
    """TERMINATED flush: pads are the absorbing state — done=1, trained
    (length = stored_len so the learner does NOT mask them; the Q->0 anchor
    at absorbing states grounds the critic's value scale)."""
    c = _mk_cutter()
    emitted = []
    for t in range(7):
        done = t == 6
        emitted += c.push(np.full(3, t, dtype=np.float32), np.zeros(2, np.float32),
                          1.0, done, 0.0, np.zeros((4, 4), np.float32))
    assert len(emitted) == 1  # flushed partial
    s = emitted[0]
    assert s.length == 10     # absorbing pads count as real steps
    assert s.obs.shape[0] == 10
    assert s.done[6] == 1.0   # the REAL terminal flag
    assert s.done[7] == 1.0 and s.done[9] == 1.0  # absorbing pads
    assert (s.reward[7:] == 0).all()
    assert s.obs[8, 0] == 6.0  # pad repeats last obs
    # window reset after done
    assert len(c.obs) == 0


def test_cutter_flush_on_truncation_masks_pads():
    """TRUNCATED flush (PEB): pads keep done=0 and are excluded from the
    losses via length < stored_len; the learner bootstraps through the
    repeated final observation."""
    c = _mk_cutter()
    emitted = []
    for t in range(7):
        emitted += c.push(np.full(3, t, dtype=np.float32), np.zeros(2, np.float32),
                          1.0, False, 0.0, np.zeros((4, 4), np.float32))
    emitted += c.flush(terminated=False)
    assert len(emitted) == 1
    s = emitted[0]
    assert s.length == 7      # real steps only -> learner masks the pads
    assert (s.done == 0).all()
    assert s.obs[8, 0] == 6.0
    assert len(c.obs) == 0


def test_cutter_short_episode_below_burnin_dropped():
    c = _mk_cutter()
    emitted = []
    for t in range(3):
        emitted += c.push(np.full(3, t, np.float32), np.zeros(2, np.float32),
                          1.0, t == 2, 0.0, np.zeros((4, 4), np.float32))
    assert emitted == []


# ---------------------------------------------------------------- replay buf
def _mk_seq(L=10, obs_dim=3, A=2, H=4, prio=1.0, fill=0.0):
    return Sequence(
        obs=np.full((L, obs_dim), fill, np.float32),
        action=np.zeros((L, A), np.float32),
        reward=np.zeros(L, np.float32),
        done=np.zeros(L, np.float32),
        state=np.zeros((4, H), np.float32),
        priority=prio, length=L,
    )


def test_replay_add_sample_update():
    rep = SequenceReplay(capacity=32, stored_len=10, obs_shape=(3,), action_dim=2,
                         hidden_size=4, per_alpha=1.0, seed=0)
    rep.add([_mk_seq(prio=1.0, fill=i) for i in range(8)])
    assert len(rep) == 8
    batch = rep.sample(16)
    assert batch["obs"].shape == (16, 10, 3)
    assert batch["state"].shape == (16, 4, 4)
    assert set(batch["idx"]) <= set(range(8))
    # priority update: zero out everything but slot 3
    rep.update_priorities(np.arange(8), np.array([1e-6] * 3 + [100.0] + [1e-6] * 4))
    b2 = rep.sample(16)
    assert (b2["idx"] == 3).mean() > 0.9


def test_replay_ring_wraps():
    rep = SequenceReplay(capacity=4, stored_len=10, obs_shape=(3,), action_dim=2,
                         hidden_size=4, seed=0)
    for i in range(6):
        rep.add([_mk_seq(prio=1.0, fill=float(i))])
    assert len(rep) == 4
    # slots 0,1 overwritten by 4,5
    assert rep.obs[0, 0, 0].item() == 4.0
    assert rep.obs[1, 0, 0].item() == 5.0


# ------------------------------------------------------------- shm seq ring
def test_seq_ring_push_drain_commit():
    import multiprocessing as mp
    from pytorch_r2d2_dpg_amd.parallel.seq_ring import SeqRing

    ctx = mp.get_context("spawn")
    ring = SeqRing(ctx, slots=4, stored_len=6, obs_shape=(3,), obs_dtype=np.float32,
                   action_dim=2, hidden_size=8)
    for i in range(3):
        assert ring.push(_mk_seq(L=6, obs_dim=3, A=2, H=8, prio=1.0 + i, fill=float(i)))
    ranges = ring.drain_ranges()
    assert ranges == [(0, 3)]
    assert ring.obs[1, 0, 0].item() == 1.0
    ring.commit()
    # wrap-around: push 4 more -> ranges split in two
    for i in range(4):
        assert ring.push(_mk_seq(L=6, obs_dim=3, A=2, H=8, fill=10.0 + i))
    ranges = ring.drain_ranges()
    assert ranges == [(3, 4), (0, 3)]
    assert ring.obs[3, 0, 0].item() == 10.0
    assert ring.obs[0, 0, 0].item() == 11.0
    ring.commit()


def test_seq_ring_backpressure_drops_newest():
    import multiprocessing as mp
    from pytorch_r2d2_dpg_amd.parallel.seq_ring import SeqRing

    ctx = mp.get_context("spawn")
    ring = SeqRing(ctx, slots=2, stored_len=6, obs_shape=(3,), obs_dtype=np.float32,
                   action_dim=2, hidden_size=8)
    assert ring.push(_mk_seq(L=6, obs_dim=3, A=2, H=8))
    assert ring.push(_mk_seq(L=6, obs_dim=3, A=2, H=8))
    assert not ring.push(_mk_seq(L=6, obs_dim=3, A=2, H=8))  # full -> dropped
    assert ring.dropped.value == 1
    ring.drain_ranges()
    ring.commit()
    assert ring.push(_mk_seq(L=6, obs_dim=3, A=2, H=8))  # space again
