"""Learner-step, checkpoint round-trip, and config tests (CPU)."""

import os

import numpy as np
import pytest
import torch

from pytorch_r2d2_dpg_amd.actor import ActorWorker
from pytorch_r2d2_dpg_amd.checkpoint import load_checkpoint, save_checkpoint
from pytorch_r2d2_dpg_amd.config import Config
from pytorch_r2d2_dpg_amd.learner import Learner
from pytorch_r2d2_dpg_amd.replay.buffer import SequenceReplay


def small_cfg():
    cfg = Config()
    cfg.env.name = "pendulum"
    cfg.env.time_limit = 50
    cfg.model.hidden_size = 16
    cfg.model.trunk_sizes = (16,)
    cfg.replay.seq_len = 12
    cfg.replay.burn_in = 4
    cfg.replay.overlap = 6
    cfg.replay.n_step = 3
    cfg.replay.capacity = 128
    cfg.replay.min_fill = 4
    cfg.learner.batch_size = 8
    return cfg


def _fill_and_learner(cfg, steps=60):
    w = ActorWorker(cfg, actor_id=0, sigma=0.3)
    rep = SequenceReplay(cfg.replay.capacity, cfg.replay.seq_len + cfg.replay.n_step,
                         w.spec.obs_shape, w.spec.action_dim, cfg.model.hidden_size,
                         per_alpha=cfg.replay.per_alpha, seed=0)
    while len(rep) < cfg.replay.min_fill:
        rep.add(w.rollout(steps))
    lrn = Learner(cfg, w.spec, device="cpu", replay=rep)
    return w, rep, lrn


def test_train_step_runs_and_updates():
    cfg = small_cfg()
    w, rep, lrn = _fill_and_learner(cfg)
    p0 = [p.clone() for p in lrn.actor.parameters()]
    t0 = [p.clone() for p in lrn.target_actor.parameters()]
    batch = rep.sample(cfg.learner.batch_size)
    stats = lrn.train_step(batch)
    assert np.isfinite(stats.critic_loss) and np.isfinite(stats.actor_loss)
    assert stats.transitions == cfg.learner.batch_size * (cfg.replay.seq_len - cfg.replay.burn_in)
    changed = any(not torch.equal(a, b) for a, b in zip(p0, lrn.actor.parameters()))
    assert changed, "actor params did not update"
    t_changed = any(not torch.equal(a, b) for a, b in zip(t0, lrn.target_actor.parameters()))
    assert t_changed, "polyak did not move targets"


def test_priority_writeback_changes_tree():
    cfg = small_cfg()
    w, rep, lrn = _fill_and_learner(cfg)
    before = rep.tree.total
    batch = rep.sample(cfg.learner.batch_size)
    lrn.train_step(batch)
    after = rep.tree.total
    assert before != after  # priorities recomputed for the sampled slots


def test_fixed_seed_determinism():
    def run():
        torch.manual_seed(7)
        cfg = small_cfg()
        w, rep, lrn = _fill_and_learner(cfg)
        losses = []
        for _ in range(3):
            batch = rep.sample(cfg.learner.batch_size)
            losses.append(lrn.train_step(batch).critic_loss)
        return losses

    assert run() == run()


def test_checkpoint_roundtrip(tmp_path):
    cfg = small_cfg()
    w, rep, lrn = _fill_and_learner(cfg)
    batch = rep.sample(cfg.learner.batch_size)
    lrn.train_step(batch)
    path = save_checkpoint(lrn, str(tmp_path))
    assert os.path.basename(path) == "ckpt_1.pth"
    sd = torch.load(path, map_location="cpu", weights_only=False)
    assert set(sd) == {"actor", "critic", "target_actor", "target_critic",
                       "opt_actor", "opt_critic", "step"}
    lrn2 = Learner(cfg, w.spec, device="cpu", replay=rep)
    step = load_checkpoint(lrn2, path)
    assert step == 1
    for a, b in zip(lrn.actor.parameters(), lrn2.actor.parameters()):
        assert torch.equal(a, b)
    for a, b in zip(lrn.target_critic.parameters(), lrn2.target_critic.parameters()):
        assert torch.equal(a, b)


def test_value_rescale_flag_runs():
    cfg = small_cfg()
    cfg.learner.value_rescale = True
    w, rep, lrn = _fill_and_learner(cfg)
    stats = lrn.train_step(rep.sample(cfg.learner.batch_size))
    assert np.isfinite(stats.critic_loss)


def test_config_toml_roundtrip(tmp_path):
    p = tmp_path / "c.toml"
    p.write_text("[model]\nhidden_size = 64\n[replay]\nseq_len = 120\nn_step = 5\n")
    cfg = Config.from_toml(str(p))
    assert cfg.model.hidden_size == 64
    assert cfg.replay.seq_len == 120
    cfg.apply_overrides(["learner.batch_size=32", "model.dtype=bf16"])
    assert cfg.learner.batch_size == 32
    assert cfg.model.dtype == "bf16"


def test_manual_decomposition_matches_module_forward():
    """learner._step_core computes the target-actor and online-critic
    forwards manually (trunk -> paired LSTM -> head) so the two unrolls can
    share one px2 launch; this invariant keeps that decomposition in
    lock-step with ActorNet.forward / CriticNet.forward."""
    from pytorch_r2d2_dpg_amd.envs import make_env
    from pytorch_r2d2_dpg_amd.models import make_models
    from pytorch_r2d2_dpg_amd.ops.lstm import lstm_seq, lstm_seq_pair_halfgrad

    spec = make_env("pendulum").spec
    torch.manual_seed(0)

    class _M:
        hidden_size, trunk_sizes, conv_channels = 64, (64,), (32, 64, 64)
    actor, critic, ta, tc = make_models(spec, _M)
    T, B, H = 7, 5, 64
    obs = torch.randn(T, B, *spec.obs_shape)
    act = torch.randn(T, B, spec.action_dim).clamp_(-1, 1)
    h = torch.randn(B, H) * 0.1
    c = torch.randn(B, H) * 0.1

    # actor: module forward vs manual head math
    mu_mod, _ = ta(obs, (h, c))
    feat = ta.trunk(obs)
    hs, _, _ = lstm_seq(feat, h, c, ta.lstm.weight_x, ta.lstm.weight_h, ta.lstm.bias)
    mu_man = (torch.tanh(ta.head(hs.reshape(-1, H))).view(T, B, -1)
              * ta.action_scale)
    assert torch.equal(mu_mod, mu_man)

    # critic: module forward vs manual (cat -> lstm -> head)
    q_mod, _ = critic(obs, act, (h, c))
    x = torch.cat([critic.trunk(obs), act], dim=-1)
    hs_c, _, _ = lstm_seq(x, h, c, critic.lstm.weight_x, critic.lstm.weight_h,
                          critic.lstm.bias)
    q_man = critic.head(hs_c.reshape(-1, H)).view(T, B)
    assert torch.equal(q_mod, q_man)

    # CPU fallback of the half-grad pair == two plain lstm_seq calls,
    # gradients included
    feat2 = feat.detach()
    x1 = x.detach().requires_grad_(True)
    hs0, hT0, cT0, hs1, hT1, cT1 = lstm_seq_pair_halfgrad(
        feat2, h, c, ta.lstm.weight_x, ta.lstm.weight_h, ta.lstm.bias,
        x1, h, c, critic.lstm.weight_x, critic.lstm.weight_h, critic.lstm.bias)
    assert torch.equal(hs0, hs)
    hs1.square().mean().backward()
    x2 = x.detach().requires_grad_(True)
    r, _, _ = lstm_seq(x2, h, c, critic.lstm.weight_x, critic.lstm.weight_h,
                       critic.lstm.bias)
    assert torch.equal(hs1, r)
    r.square().mean().backward()
    assert torch.allclose(x1.grad, x2.grad)


def test_truncation_pad_steps_cannot_influence_update():
    """Masked-loss contract: for a TRUNCATED (PEB) sequence, nothing in the
    pad region may influence the parameter update — scrambling pad rewards
    and actions must yield the identical step (the repeated pad OBS is the
    bootstrap source and is kept identical here)."""
    from pytorch_r2d2_dpg_amd.envs import make_env
    torch.manual_seed(0)
    cfg = Config()
    cfg.env.name = "synthetic"
    cfg.model.hidden_size = 32
    cfg.model.trunk_sizes = (32,)
    cfg.replay.seq_len = 12
    cfg.replay.burn_in = 4
    cfg.replay.n_step = 3
    cfg.learner.batch_size = 6
    spec = make_env(cfg.env.name).spec
    L = cfg.replay.seq_len + cfg.replay.n_step
    B, H = cfg.learner.batch_size, cfg.model.hidden_size
    g = torch.Generator().manual_seed(7)
    length = torch.tensor([L, 9, L, 11, 7, L], dtype=torch.int32)
    base = {
        "obs": torch.randn(B, L, *spec.obs_shape, generator=g),
        "action": torch.randn(B, L, spec.action_dim, generator=g),
        "reward": torch.rand(B, L, generator=g),
        "done": torch.zeros(B, L),
        "state": torch.randn(B, 4, H, generator=g) * 0.05,
        "weight": torch.ones(B),
        "idx": None,
        "length": length,
    }
    # repeat the last real obs into the pad (as the cutter does) and zero
    # pad rewards so the two runs share the bootstrap inputs
    for b in range(B):
        n = int(length[b])
        if n < L:
            base["obs"][b, n:] = base["obs"][b, n - 1]
            base["reward"][b, n:] = 0.0

    results = []
    for scramble in (False, True):
        batch = {k: (v.clone() if torch.is_tensor(v) else v) for k, v in base.items()}
        if scramble:
            gs = torch.Generator().manual_seed(99)
            for b in range(B):
                n = int(length[b])
                if n < L:
                    batch["reward"][b, n:] = torch.randn(L - n, generator=gs) * 50
                    batch["action"][b, n:] = torch.randn(L - n, spec.action_dim,
                                                         generator=gs) * 3
        torch.manual_seed(3)
        lrn = Learner(cfg, spec, device="cpu")
        lrn.train_step(batch)
        results.append(torch.cat([p.detach().reshape(-1)
                                  for p in list(lrn.actor.parameters())
                                  + list(lrn.critic.parameters())]))
    assert torch.equal(results[0], results[1]), \
        "pad rewards/actions leaked into the update"
