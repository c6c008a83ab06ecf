"""CLI.

    python -m pytorch_r2d2_dpg_amd.cli train --config configs/humanoid_8gpu.toml \\
        [--set learner.batch_size=128 model.dtype=bf16 ...]
    python -m pytorch_r2d2_dpg_amd.cli train-single --config configs/pendulum_cpu.toml
    python -m pytorch_r2d2_dpg_amd.cli eval --config ... --checkpoint ckpt.pth

Multi-GPU: launch `train` under torchrun (one rank per GPU over RCCL):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \\
        --master-addr 127.0.0.1 -m pytorch_r2d2_dpg_amd.cli train --config ...
"""

from __future__ import annotations

import argparse
import sys

import numpy as np
import torch

from .config import Config


def _load_cfg(args) -> Config:
    cfg = Config.from_toml(args.config) if args.config else Config()
    if args.set:
        cfg.apply_overrides(args.set)
    return cfg


def cmd_train(args) -> int:
    from .orchestrator import run_training
    cfg = _load_cfg(args)
    run_training(cfg, resume=args.resume, log_path=args.log,
                 max_steps=args.steps if args.steps > 0 else None)
    return 0


def cmd_train_single(args) -> int:
    from .trainer import train_single_process
    from .utils.metrics import MetricsLogger
    cfg = _load_cfg(args)
    logger = MetricsLogger(args.log, stdout_every=1)
    train_single_process(cfg, device=args.device,
                         max_steps=args.steps if args.steps > 0 else None,
                         logger=logger)
    return 0


def cmd_eval(args) -> int:
    from .actor import ActorWorker
    cfg = _load_cfg(args)
    worker = ActorWorker(cfg, actor_id=0, eval_mode=True)
    if args.checkpoint:
        sd = torch.load(args.checkpoint, map_location="cpu", weights_only=False)
        worker.actor_net.load_state_dict(sd["actor"])
        worker.critic_net.load_state_dict(sd["critic"])
    for ep in range(args.episodes):
        worker.rollout(cfg.env.time_limit)
    rets = worker.episode_returns[:args.episodes]
    print(f"episodes={len(rets)} mean_return={np.mean(rets):.2f} "
          f"min={np.min(rets):.2f} max={np.max(rets):.2f}")
    return 0


def main(argv=None) -> int:
    p = argparse.ArgumentParser(prog="r2d2dpg")
    sub = p.add_subparsers(dest="cmd", required=True)

    def common(sp):
        sp.add_argument("--config", type=str, default="")
        sp.add_argument("--set", nargs="*", default=[],
                        help="section.key=value overrides")
        sp.add_argument("--log", type=str, default="")
        sp.add_argument("--steps", type=int, default=0)

    sp = sub.add_parser("train", help="distributed actor/learner training")
    common(sp)
    sp.add_argument("--resume", type=str, default="")
    sp.set_defaults(fn=cmd_train)

    sp = sub.add_parser("train-single", help="single-process training (config 1)")
    common(sp)
    sp.add_argument("--device", type=str, default="cpu")
    sp.set_defaults(fn=cmd_train_single)

    sp = sub.add_parser("eval", help="noise-free evaluation rollouts")
    common(sp)
    sp.add_argument("--checkpoint", type=str, default="")
    sp.add_argument("--episodes", type=int, default=10)
    sp.set_defaults(fn=cmd_eval)

    args = p.parse_args(argv)
    return args.fn(args)


if __name__ == "__main__":
    sys.exit(main())
