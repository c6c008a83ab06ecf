import numpy as np

from pytorch_r2d2_dpg_amd.envs import make_env, SYNTHETIC_SPECS


def test_pendulum_dynamics():
    env = make_env("pendulum", seed=0)
    obs = env.reset()
    assert obs.shape == (3,)
    total = 0.0
    for _ in range(200):
        obs, r, done, trunc = env.step(np.array([0.0], np.float32))
        assert r <= 0.0
        total += r
    assert trunc
    assert not done


def test_pendulum_seeded_reproducible():
    e1, e2 = make_env("pendulum", seed=5), make_env("pendulum", seed=5)
    o1, o2 = e1.reset(), e2.reset()
    assert np.allclose(o1, o2)


def test_synthetic_shapes():
    for name, spec in SYNTHETIC_SPECS.items():
        env = make_env(name, seed=1)
        obs = env.reset()
        assert obs.shape == spec.obs_shape, name
        obs, r, done, trunc = env.step(np.zeros(spec.action_dim, np.float32))
        assert obs.shape == spec.obs_shape
        assert 0.0 <= r <= 1.0


def test_synthetic_pixel_uint8():
    env = make_env("quadruped-run-pixels")
    obs = env.reset()
    assert obs.dtype == np.uint8 and obs.shape == (84, 84, 3)


def test_action_repeat_accumulates_reward():
    from pytorch_r2d2_dpg_amd.actor import ActorWorker
    from pytorch_r2d2_dpg_amd.config import Config

    cfg = Config()
    cfg.env.name = "pendulum"
    cfg.env.time_limit = 40
    cfg.env.action_repeat = 4
    cfg.model.hidden_size = 16
    cfg.model.trunk_sizes = (16,)
    cfg.replay.seq_len = 8
    cfg.replay.burn_in = 2
    cfg.replay.overlap = 4
    cfg.replay.n_step = 2
    w = ActorWorker(cfg, actor_id=0, sigma=0.2)
    w.rollout(10)  # 10 policy steps = 40 env frames = one episode
    assert w.env_steps == 10
    assert len(w.episode_returns) >= 1  # time limit hit within repeats


def test_max_env_steps_caps_rollout():
    from pytorch_r2d2_dpg_amd.actor import ActorWorker
    from pytorch_r2d2_dpg_amd.config import Config

    cfg = Config()
    cfg.env.name = "pendulum"
    cfg.model.hidden_size = 16
    cfg.model.trunk_sizes = (16,)
    cfg.replay.seq_len = 8
    cfg.replay.burn_in = 2
    cfg.replay.overlap = 4
    cfg.replay.n_step = 2
    cfg.actor.max_env_steps = 15
    w = ActorWorker(cfg, actor_id=0, sigma=0.2)
    w.rollout(30)
    assert w.env_steps <= 30
    w.rollout(30)
    more = w.env_steps
    w.rollout(30)
    assert w.env_steps == more  # capped: no further progress


def test_dmc_adapter_with_stubbed_suite(monkeypatch):
    """The optional DM-Control adapter: flattening, spec derivation and the
    termination/truncation split from ts.discount (exercised against a stub
    suite — the real dm_control is absent in this sandbox)."""
    import sys
    import types

    class _TS:
        def __init__(self, obs, reward, discount, last):
            self.observation = obs
            self.reward = reward
            self.discount = discount
            self._last = last

        def last(self):
            return self._last

    class _ASpec:
        shape = (2,)
        maximum = np.array([1.0, 1.0])

    class _FakeEnv:
        def __init__(self):
            self.t = 0

        def action_spec(self):
            return _ASpec()

        def observation_spec(self):
            class _S:  # shaped entries: (2,) vector + scalar
                pass
            v = _S(); v.shape = (2,)
            s = _S(); s.shape = ()
            return {"pos": v, "height": s}

        def reset(self):
            self.t = 0
            return _TS({"pos": np.array([1.0, 2.0]), "height": 3.0}, None, 1.0, False)

        def step(self, a):
            self.t += 1
            if self.t >= 3:  # terminate (discount 0)
                return _TS({"pos": np.zeros(2), "height": 0.0}, 1.0, 0.0, True)
            return _TS({"pos": np.ones(2), "height": 1.0}, 0.5, 1.0, False)

    suite = types.ModuleType("dm_control.suite")
    suite.load = lambda d, t, task_kwargs=None: _FakeEnv()
    pkg = types.ModuleType("dm_control")
    pkg.suite = suite
    monkeypatch.setitem(sys.modules, "dm_control", pkg)
    monkeypatch.setitem(sys.modules, "dm_control.suite", suite)

    env = make_env("walker-walk", seed=3)
    assert env.spec.obs_shape == (3,) and env.spec.action_dim == 2
    obs = env.reset()
    # sorted key order: height then pos
    assert np.allclose(obs, [3.0, 1.0, 2.0])
    obs, r, done, trunc = env.step(np.zeros(2, np.float32))
    assert r == 0.5 and not done and not trunc
    env.step(np.zeros(2, np.float32))
    obs, r, done, trunc = env.step(np.zeros(2, np.float32))
    assert done and r == 1.0  # discount 0 at last() -> true termination


def test_make_env_unknown_dmc_raises_clear_error():
    import pytest as _pytest
    with _pytest.raises(ValueError, match="dm_control is not installed"):
        make_env("walker-walk")


def test_ou_noise_temporal_correlation_and_reset():
    from pytorch_r2d2_dpg_amd.utils.noise import OUNoise, make_noise
    n = OUNoise(4, sigma=0.3, seed=0)
    xs = np.stack([n() for _ in range(200)])
    # OU is temporally correlated: successive-step correlation well above iid
    a, b = xs[:-1].reshape(-1), xs[1:].reshape(-1)
    corr = np.corrcoef(a, b)[0, 1]
    assert corr > 0.8
    n.reset()
    assert np.allclose(n._x, 0.0)
    assert isinstance(make_noise("ou", 2, 0.1), OUNoise)


def test_actor_with_ou_noise_rolls_out():
    from pytorch_r2d2_dpg_amd.actor import ActorWorker
    from pytorch_r2d2_dpg_amd.config import Config

    cfg = Config()
    cfg.env.name = "pendulum"
    cfg.actor.noise = "ou"
    cfg.model.hidden_size = 16
    cfg.model.trunk_sizes = (16,)
    cfg.replay.seq_len = 8
    cfg.replay.burn_in = 2
    cfg.replay.overlap = 4
    cfg.replay.n_step = 2
    w = ActorWorker(cfg, actor_id=0, sigma=0.3)
    seqs = w.rollout(30)
    assert w.env_steps == 30 and seqs
