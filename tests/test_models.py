"""Model + reference-LSTM tests (CPU)."""

import numpy as np
import pytest
import torch

from pytorch_r2d2_dpg_amd.config import Config
from pytorch_r2d2_dpg_amd.envs import make_env
from pytorch_r2d2_dpg_amd.models import ActorNet, CriticNet, FusedLSTM, make_models
from pytorch_r2d2_dpg_amd.ops.reference import lstm_seq_ref, polyak_ref


def test_lstm_ref_matches_torch_lstmcell():
    torch.manual_seed(0)
    T, B, I, H = 6, 4, 5, 8
    cell = torch.nn.LSTMCell(I, H)
    x = torch.randn(T, B, I)
    h0, c0 = torch.randn(B, H), torch.randn(B, H)
    # our single fused bias b = b_ih + b_hh
    wx, wh = cell.weight_ih, cell.weight_hh
    b = cell.bias_ih + cell.bias_hh
    h_seq, hT, cT = lstm_seq_ref(x, h0, c0, wx, wh, b)
    h, c = h0, c0
    for t in range(T):
        h, c = cell(x[t], (h, c))
        assert torch.allclose(h_seq[t], h, atol=1e-5), t
    assert torch.allclose(hT, h, atol=1e-5)
    assert torch.allclose(cT, c, atol=1e-5)


def test_lstm_ref_gradcheck():
    torch.manual_seed(1)
    T, B, I, H = 3, 2, 3, 4
    args = [torch.randn(T, B, I, dtype=torch.double, requires_grad=True),
            torch.randn(B, H, dtype=torch.double, requires_grad=True),
            torch.randn(B, H, dtype=torch.double, requires_grad=True),
            torch.randn(4 * H, I, dtype=torch.double, requires_grad=True),
            torch.randn(4 * H, H, dtype=torch.double, requires_grad=True),
            torch.randn(4 * H, dtype=torch.double, requires_grad=True)]
    assert torch.autograd.gradcheck(lambda *a: lstm_seq_ref(*a)[0], args, eps=1e-6, atol=1e-4)


def test_fused_lstm_module_step_vs_seq():
    torch.manual_seed(2)
    m = FusedLSTM(5, 8)
    x = torch.randn(4, 3, 5)
    h_seq, (hT, cT) = m(x)
    st = None
    for t in range(4):
        h, st = m.step(x[t], st)
        assert torch.allclose(h, h_seq[t], atol=1e-6)
    assert torch.allclose(st[0], hT)


def test_actor_critic_shapes_state_obs():
    torch.manual_seed(3)
    a = ActorNet((17,), 6, action_scale=1.0, hidden_size=32, trunk_sizes=(32,))
    c = CriticNet((17,), 6, hidden_size=32, trunk_sizes=(32,))
    obs = torch.randn(5, 4, 17)
    act, st = a(obs)
    assert act.shape == (5, 4, 6)
    assert act.abs().max() <= 1.0
    q, _ = c(obs, act)
    assert q.shape == (5, 4)


def test_actor_critic_pixel_trunk():
    torch.manual_seed(4)
    a = ActorNet((84, 84, 3), 12, hidden_size=16, pixel=True)
    obs = torch.randint(0, 255, (2, 3, 84, 84, 3), dtype=torch.uint8)
    act, _ = a(obs)
    assert act.shape == (2, 3, 12)


def test_polyak_ref():
    t = [torch.zeros(3), torch.ones(2)]
    o = [torch.ones(3), torch.zeros(2)]
    polyak_ref(t, o, tau=0.1)
    assert torch.allclose(t[0], torch.full((3,), 0.1))
    assert torch.allclose(t[1], torch.full((2,), 0.9))


def test_make_models_targets_are_copies():
    cfg = Config()
    cfg.model.hidden_size = 16
    cfg.model.trunk_sizes = (16,)
    env = make_env("pendulum")
    a, c, ta, tc = make_models(env.spec, cfg.model)
    for p, tp in zip(a.parameters(), ta.parameters()):
        assert torch.equal(p, tp)
        assert not tp.requires_grad
