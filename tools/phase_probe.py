"""Learner phase breakdown via the hipEvent phase timers.

Prints per-phase ms (burn_in / target / critic / actor_opt) averaged over a
few steps at the flagship config. Run on a GPU box:

    R2D2_PHASE_TIMERS=1 python tools/phase_probe.py [--batch 256 --hidden 512]

(The env var must be set before the Learner is constructed; the script sets
it itself if missing.)
"""

import argparse
import os
import sys

os.environ.setdefault("R2D2_PHASE_TIMERS", "1")

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from pytorch_r2d2_dpg_amd.config import Config
from pytorch_r2d2_dpg_amd.envs import make_env
from pytorch_r2d2_dpg_amd.learner import Learner


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--hidden", type=int, default=512)
    p.add_argument("--batch", type=int, default=256)
    p.add_argument("--seq-len", type=int, default=80)
    p.add_argument("--burn-in", type=int, default=40)
    p.add_argument("--iters", type=int, default=10)
    args = p.parse_args()
    H, B = args.hidden, args.batch

    cfg = Config()
    cfg.env.name = "humanoid-run"
    cfg.model.hidden_size = H
    cfg.model.trunk_sizes = (H, H)
    cfg.model.dtype = "bf16"
    cfg.replay.seq_len = args.seq_len
    cfg.replay.burn_in = args.burn_in
    cfg.learner.batch_size = B
    spec = make_env(cfg.env.name).spec
    lrn = Learner(cfg, spec, device="cuda")
    lrn.collect_stats = False

    L = cfg.replay.seq_len + cfg.replay.n_step
    g = torch.Generator().manual_seed(0)
    batch = {
        "obs": torch.randn(B, L, *spec.obs_shape, generator=g).cuda(),
        "action": torch.randn(B, L, spec.action_dim, generator=g).clamp_(-1, 1).cuda(),
        "reward": torch.rand(B, L, generator=g).cuda(),
        "done": torch.zeros(B, L).cuda(),
        "state": torch.zeros(B, 4, H).cuda(),
        "weight": torch.ones(B).cuda(),
        "idx": None,
    }
    for _ in range(5):
        lrn.train_step(batch)
    acc = {}
    for _ in range(args.iters):
        lrn.train_step(batch)
        for k, v in lrn.phase_times_ms().items():
            acc[k] = acc.get(k, 0.0) + v
    total = 0.0
    for k, v in acc.items():
        print(f"{k:10s}: {v / args.iters:7.3f} ms")
        total += v / args.iters
    print(f"{'sum':10s}: {total:7.3f} ms  (B={B} H={H} seq={args.seq_len}/{args.burn_in})")


if __name__ == "__main__":
    main()
