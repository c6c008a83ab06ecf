"""Hyperparameter sweep for the Pendulum learning smoke (CPU, parallel).

Variants run in separate processes; each reports the mean of the last 10
episode returns after N learner steps. Picks the config for
configs/pendulum_cpu.toml and the CI learning test.
"""

import itertools
import multiprocessing as mp
import sys


def run_variant(args):
    (rs, sigma, wd, lra, steps, seed) = args
    import os, sys
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    import numpy as np
    import torch
    torch.manual_seed(seed)
    torch.set_num_threads(2)
    from pytorch_r2d2_dpg_amd.config import Config
    from pytorch_r2d2_dpg_amd.trainer import train_single_process

    cfg = Config.from_toml("configs/pendulum_cpu.toml")
    cfg.replay.reward_scale = rs
    cfg.actor.sigma_max = sigma
    cfg.learner.critic_weight_decay = wd
    cfg.learner.lr_actor = lra
    cfg.learner.checkpoint_every = 0
    cfg.learner.sync_every = 20
    cfg.learner.log_every = 10 ** 9
    try:
        lrn, w = train_single_process(cfg, max_steps=steps, env_steps_per_update=40)
        rets = w.episode_returns[-10:]
        return args, float(np.mean(rets)) if rets else float("nan")
    except Exception as e:  # noqa: BLE001
        return args, f"ERR {e!r}"


if __name__ == "__main__":
    steps = int(sys.argv[1]) if len(sys.argv) > 1 else 1500
    grid = list(itertools.product(
        [0.05, 0.1],       # reward_scale
        [0.3, 0.5],        # sigma_max
        [0.0],             # critic_weight_decay
        [3e-4],            # lr_actor
        [steps], [0, 1],
    ))
    with mp.get_context("spawn").Pool(8) as pool:
        for args, ret in pool.imap_unordered(run_variant, grid):
            rs, sg, wd, lra = args[:4]
            print(f"rs={rs:<4} sigma={sg:<4} wd={wd:<6} lra={lra:<7} -> ret {ret}",
                  flush=True)
