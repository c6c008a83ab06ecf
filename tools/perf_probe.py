"""Perf probe: where does the learner step spend its time? (GPU only)

1. Isolated fused-LSTM fwd/bwd timing (K1/K2) at the flagship shape.
2. Phase-level timing of the learner step with cuda events.
3. torch.profiler table of the top ops over a few steps.

Run: python tools/perf_probe.py [--hidden 512 --batch 256]
"""

import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from pytorch_r2d2_dpg_amd.config import Config
from pytorch_r2d2_dpg_amd.envs import make_env
from pytorch_r2d2_dpg_amd.learner import Learner
from pytorch_r2d2_dpg_amd.ops.lstm import lstm_seq


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--hidden", type=int, default=512)
    p.add_argument("--batch", type=int, default=256)
    p.add_argument("--T", type=int, default=40)
    p.add_argument("--dtype", default="bf16")
    args = p.parse_args()
    dt = torch.bfloat16 if args.dtype == "bf16" else torch.float32
    H, B, T = args.hidden, args.batch, args.T
    I = H

    dev = "cuda"
    x = torch.randn(T, B, I, dtype=dt, device=dev)
    h0 = torch.randn(B, H, dtype=dt, device=dev)
    c0 = torch.randn(B, H, dtype=dt, device=dev)
    wx = (torch.randn(4 * H, I, dtype=dt, device=dev) / I ** 0.5)
    wh = (torch.randn(4 * H, H, dtype=dt, device=dev) / H ** 0.5)
    b = torch.zeros(4 * H, dtype=dt, device=dev)

    # ---- isolated kernel timings
    from pytorch_r2d2_dpg_amd.ops import hip_ext
    ext = hip_ext()
    gates_x = torch.addmm(b, x.reshape(T * B, I), wx.t()).view(T, B, 4 * H).contiguous()
    t_fwd_k = timeit(lambda: ext.lstm_seq_fwd(gates_x, h0, c0, wh))
    print(f"K1 lstm fwd kernel only      : {t_fwd_k:8.3f} ms  (T={T} B={B} H={H} {args.dtype})")

    h_seq, c_final, gates_act, c_seq = ext.lstm_seq_fwd(gates_x, h0, c0, wh)
    dh = torch.randn_like(h_seq)
    whT = wh.contiguous()  # lstm_seq_bwd takes the untransposed weight now
    zero = torch.zeros_like(h0)
    t_bwd_k = timeit(lambda: ext.lstm_seq_bwd(dh, zero, zero, gates_act, c_seq, c0, whT))
    print(f"K2 lstm bwd kernel only      : {t_bwd_k:8.3f} ms")

    t_gx = timeit(lambda: torch.addmm(b, x.reshape(T * B, I), wx.t()))
    print(f"gates_x GEMM (rocBLAS)       : {t_gx:8.3f} ms")

    xg = x.clone().requires_grad_(True)
    def full_bwd():
        hs, hT, cT = lstm_seq(xg, h0, c0, wx, wh, b)
        hs.square().mean().backward()
    t_full = timeit(full_bwd, iters=10)
    print(f"lstm_seq fwd+bwd (autograd)  : {t_full:8.3f} ms")

    # ---- learner phase breakdown
    cfg = Config()
    cfg.env.name = "humanoid-run"
    cfg.model.hidden_size = H
    cfg.model.trunk_sizes = (H, H)
    cfg.model.dtype = args.dtype
    cfg.replay.seq_len = 80
    cfg.replay.burn_in = 40
    cfg.learner.batch_size = B
    spec = make_env(cfg.env.name).spec
    lrn = Learner(cfg, spec, device=dev)
    lrn.collect_stats = False
    L = cfg.replay.seq_len + cfg.replay.n_step
    g = torch.Generator().manual_seed(0)
    batch = {
        "obs": torch.randn(B, L, *spec.obs_shape, generator=g).to(dev),
        "action": torch.randn(B, L, spec.action_dim, generator=g).clamp_(-1, 1).to(dev),
        "reward": torch.rand(B, L, generator=g).to(dev),
        "done": torch.zeros(B, L).to(dev),
        "state": torch.zeros(B, 4, H).to(dev),
        "weight": torch.ones(B).to(dev),
        "idx": None,
    }
    t_step = timeit(lambda: lrn.train_step(batch), iters=10, warmup=3)
    print(f"full learner step (eager)    : {t_step:8.3f} ms")

    # ---- profiler table
    from torch.profiler import profile, ProfilerActivity
    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA]) as prof:
        for _ in range(3):
            lrn.train_step(batch)
        torch.cuda.synchronize()
    print(prof.key_averages().table(sort_by="self_cuda_time_total", row_limit=25))


if __name__ == "__main__":
    main()
