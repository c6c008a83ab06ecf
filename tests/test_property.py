"""Property-based tests (hypothesis) for the edge-case-prone data-plane
components: sequence cutter invariants over arbitrary episode structures,
the n-step fold vs brute force on random inputs, and sum-tree consistency
under random update/sample interleavings (SURVEY.md §4 Unit)."""

import numpy as np
import torch
from hypothesis import given, settings, strategies as st

from pytorch_r2d2_dpg_amd.ops.reference import fold_nstep_ref
from pytorch_r2d2_dpg_amd.replay.sequence import SequenceCutter
from pytorch_r2d2_dpg_amd.replay.sumtree_cpu import SumTree

SEQ, OVL, BURN, NSTEP = 8, 4, 2, 2
STORED = SEQ + NSTEP  # 10


@settings(max_examples=60, deadline=None)
@given(st.lists(st.booleans(), min_size=1, max_size=100))
def test_cutter_invariants_any_episode_structure(dones):
    c = SequenceCutter(SEQ, OVL, BURN, NSTEP, 0.99, 0.9, 4)
    emitted = []
    for t, d in enumerate(dones):
        emitted += c.push(np.full(3, float(t), np.float32),
                          np.zeros(2, np.float32), 0.5, d, 0.0,
                          np.full((4, 4), float(t), np.float32))
    emitted += c.flush(terminated=False)
    for s in emitted:
        # fixed storage shape, real length bounded
        assert s.obs.shape == (STORED, 3)
        assert BURN + 1 < s.length <= STORED
        t0 = s.obs[0, 0]
        # state snapshot taken at the sequence's first step
        assert s.state[0, 0] == t0
        # real region = consecutive env steps up to the first repeat; the
        # pad repeats the last real obs. For TERMINATED flushes the pads
        # are absorbing steps (done=1, length = STORED); for truncation
        # flushes they are masked (done=0, length < STORED).
        n_real = 1
        while n_real < STORED and s.obs[n_real, 0] == t0 + n_real:
            n_real += 1
        for k in range(n_real, STORED):
            assert s.obs[k, 0] == s.obs[n_real - 1, 0]
        if s.length == STORED and n_real < STORED:  # absorbing pads
            assert (s.done[n_real:] == 1.0).all()
            assert (s.reward[n_real:] == 0.0).all()
        elif s.length < STORED:                      # truncation pads
            assert s.length == n_real
            assert (s.done[n_real:] == 0.0).all()
        # the done flags form a (possibly empty) SUFFIX: once an episode
        # terminates nothing after it is alive
        d = s.done.astype(bool)
        if d.any():
            first = int(np.argmax(d))
            assert d[first:].all()
        assert np.isfinite(s.priority) and s.priority >= 0.0


@settings(max_examples=40, deadline=None)
@given(st.integers(1, 6), st.integers(2, 24), st.data())
def test_fold_nstep_random_vs_bruteforce(B, T_ext, data):
    n = data.draw(st.integers(1, min(6, T_ext)))
    T = data.draw(st.integers(1, T_ext))
    rng = np.random.default_rng(data.draw(st.integers(0, 10_000)))
    r = torch.tensor(rng.standard_normal((B, T_ext)), dtype=torch.float32)
    d = torch.tensor((rng.random((B, T_ext)) < 0.15).astype(np.float32))
    folded, mask = fold_nstep_ref(r, d, 0.97, n, T)
    for b in range(B):
        for t in range(T):
            acc, g, alive = 0.0, 1.0, 1.0
            for k in range(n):
                if t + k >= T_ext:
                    alive = 0.0
                    break
                acc += g * alive * float(r[b, t + k])
                alive *= 1.0 - float(d[b, t + k])
                g *= 0.97
            assert abs(float(folded[b, t]) - acc) < 1e-4
            assert abs(float(mask[b, t]) - alive) < 1e-6


@settings(max_examples=40, deadline=None)
@given(st.integers(2, 200), st.data())
def test_sumtree_consistency_under_interleaving(cap, data):
    st_tree = SumTree(cap)
    rng = np.random.default_rng(data.draw(st.integers(0, 10_000)))
    for _ in range(data.draw(st.integers(1, 8))):
        n = data.draw(st.integers(1, cap))
        idx = rng.choice(cap, size=n, replace=False)
        pri = rng.random(n) + 1e-3
        st_tree.update(idx, pri)
        # root equals the sum of all leaves after every batched update
        assert np.isclose(st_tree.total,
                          st_tree.tree[st_tree.capacity:
                                       st_tree.capacity + cap].sum())
        if st_tree.total > 0:
            sample_idx, probs = st_tree.sample(data.draw(st.integers(1, 32)), rng)
            # sampled leaves must carry nonzero mass; probs match leaf/total
            leaf = st_tree.leaf(sample_idx)
            assert (leaf > 0).all()
            assert np.allclose(probs, leaf / st_tree.total)
