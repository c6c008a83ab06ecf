"""Flagship benchmark — learner transitions/sec, humanoid-run (BASELINE.json).

Measures the full learner step on N GPUs of one node (one rank per GPU over
RCCL): prioritized sample from the GPU-resident replay shard -> burn-in ->
target unroll -> fused n-step TD targets (K3) -> critic+actor updates through
the fused HIP LSTM (K1/K2) with fused flat-bucket gradient all-reduce ->
fused Polyak (K4) -> priority write-back. Replay is pre-filled with synthetic
humanoid-run-shaped trajectories (obs 67, act 21) and weights are random-init
(no datasets/checkpoints exist offline; BASELINE.json prescribes synthetic).

value = whole-job replayed transitions/sec = n_gpus * batch * train_len / t_step
(max step time over ranks). Weak scaling: per-GPU batch fixed as N grows.

Launch (driver contract):
  python bench.py --gpus N --steps K --warmup W
  torchrun --nnodes=1 --nproc-per-node N --master-addr 127.0.0.1 bench.py ...
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from pytorch_r2d2_dpg_amd.config import Config
from pytorch_r2d2_dpg_amd.envs import make_env
from pytorch_r2d2_dpg_amd.learner import Learner
from pytorch_r2d2_dpg_amd.parallel.dist import DistContext
from pytorch_r2d2_dpg_amd.replay.gpu_buffer import GpuSequenceReplay


def flagship_config(args) -> Config:
    cfg = Config()
    cfg.env.name = args.env
    if args.env == "quadruped-run-pixels":
        # BASELINE.json config 4 (configs/quadruped_pixel.toml): Conv+LSTM
        # encoder, uint8 pixel obs resident in the HBM shard
        cfg.env.pixel = True
        cfg.model.hidden_size = args.hidden
        cfg.model.trunk_sizes = (args.hidden,)
    else:
        cfg.model.hidden_size = args.hidden
        cfg.model.trunk_sizes = (args.hidden, args.hidden)
    cfg.model.dtype = args.dtype
    cfg.replay.seq_len = args.seq_len
    cfg.replay.burn_in = args.burn_in
    cfg.replay.n_step = 5
    cfg.replay.gamma = 0.997
    cfg.replay.capacity = args.replay_capacity
    cfg.learner.batch_size = args.batch
    cfg.learner.seed = 1234
    return cfg


def fill_replay_synthetic(replay: GpuSequenceReplay, spec, cfg: Config, n_seqs: int,
                          seed: int) -> None:
    """Pre-fill the GPU shard with random synthetic trajectories, in chunks."""
    g = torch.Generator().manual_seed(seed)
    L = cfg.replay.seq_len + cfg.replay.n_step
    H = cfg.model.hidden_size
    chunk = 256
    done = 0
    while done < n_seqs:
        n = min(chunk, n_seqs - done)
        if spec.pixel:
            obs = torch.randint(0, 256, (n, L) + tuple(spec.obs_shape),
                                dtype=torch.uint8, generator=g)
        else:
            obs = torch.randn(n, L, *spec.obs_shape, generator=g)
        act = torch.randn(n, L, spec.action_dim, generator=g).clamp_(-1, 1)
        rew = torch.rand(n, L, generator=g)
        dn = torch.zeros(n, L)
        st = torch.randn(n, 4, H, generator=g) * 0.05
        prio = torch.rand(n, generator=g) + 0.5
        replay.add_batch(obs, act, rew, dn, st, prio)
        done += n


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    # defaults give a multi-second timed region (~2.5 s at 5.8 ms/step) so
    # run-to-run variance is observable, while still finishing in well under
    # a minute including warmup/fill
    p.add_argument("--steps", type=int, default=400)
    p.add_argument("--warmup", type=int, default=40)
    p.add_argument("--env", type=str, default="humanoid-run",
                   choices=["humanoid-run", "cheetah-run", "quadruped-run-pixels"],
                   help="humanoid-run = flagship (BASELINE config 3); "
                        "quadruped-run-pixels = Conv+LSTM pixel config 4")
    p.add_argument("--batch", type=int, default=None,
                   help="sequences per rank per step (default: 256; pixel: 32 "
                        "to match configs/quadruped_pixel.toml)")
    p.add_argument("--seq-len", type=int, default=80)
    p.add_argument("--burn-in", type=int, default=40)
    p.add_argument("--hidden", type=int, default=None,
                   help="LSTM hidden (default: 512; pixel: 256)")
    p.add_argument("--dtype", type=str, default="bf16", choices=["bf16", "fp32"])
    p.add_argument("--replay-capacity", type=int, default=None)
    p.add_argument("--fp8-obs", action="store_true",
                   help="store observations fp8 e4m3 in the replay shard (K7)")
    p.add_argument("--device", type=str, default="")
    p.add_argument("--graph", action="store_true", default=False,
                   help="capture the learner step in a hipGraph (default off: "
                        "capture serializes the multi-stream overlap)")
    p.add_argument("--no-graph", dest="graph", action="store_false")
    args = p.parse_args()
    pixel = args.env == "quadruped-run-pixels"
    cheetah = args.env == "cheetah-run"
    # per-env defaults follow the BASELINE.json configs: humanoid (config 3)
    # B=256/H=512, cheetah (config 2) B=64/H=256, pixels (config 4) B=32/H=256
    if args.batch is None:
        args.batch = 32 if pixel else (64 if cheetah else 256)
    if args.hidden is None:
        args.hidden = 256 if (pixel or cheetah) else 512
    if args.replay_capacity is None:
        args.replay_capacity = 2048 if pixel else 8192

    world = int(os.environ.get("WORLD_SIZE", "1"))
    dist = DistContext.from_env()
    rank = dist.rank
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    if args.device:
        device = args.device
    elif torch.cuda.is_available():
        device = f"cuda:{local_rank}"
        torch.cuda.set_device(device)
    else:
        device = "cpu"

    cfg = flagship_config(args)
    torch.manual_seed(cfg.learner.seed + rank)
    spec = make_env(cfg.env.name).spec

    replay = GpuSequenceReplay(
        capacity=cfg.replay.capacity, stored_len=cfg.replay.seq_len + cfg.replay.n_step,
        obs_shape=spec.obs_shape, action_dim=spec.action_dim,
        hidden_size=cfg.model.hidden_size, per_alpha=cfg.replay.per_alpha,
        device=device, seed=cfg.learner.seed + 17 * rank,
        obs_dtype=torch.uint8 if spec.pixel else torch.float32,
        store_dtype=torch.bfloat16 if args.dtype == "bf16" else torch.float32,
        fp8_obs=args.fp8_obs,
    )
    fill_replay_synthetic(replay, spec, cfg, min(2048, cfg.replay.capacity),
                          seed=cfg.learner.seed + 31 * rank)
    lrn = Learner(cfg, spec, device=device, dist=dist, replay=replay)

    B, T = cfg.learner.batch_size, cfg.train_len
    lrn.collect_stats = False  # no per-step host syncs in the timed loop

    def one_step():
        batch = replay.sample(B)
        lrn.train_step(batch)

    # ---- hipGraph capture of the learner update (launch-bound workload)
    graphed = False
    if args.graph and device.startswith("cuda"):
        try:
            lrn.capture_graph(replay.sample(B))
            graphed = True
        except Exception as e:  # noqa: BLE001
            print(f"[bench] graph capture failed ({e!r}); running eager",
                  file=sys.stderr, flush=True)

    # ---- warmup
    for _ in range(args.warmup):
        one_step()

    # ---- timed region: barrier + sync on both sides, max over ranks
    if device != "cpu":
        torch.cuda.synchronize()
    dist.barrier()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_step()
    if device != "cpu":
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    dist.barrier()
    elapsed = dist.allreduce_scalar(elapsed, op="max")

    # safety net: the column-split kernels flag bounded-spin timeouts here
    if device.startswith("cuda"):
        from pytorch_r2d2_dpg_amd.ops import hip_ext
        px_err = hip_ext().lstm_px_error()
        if px_err:
            print(f"[bench] WARNING: px spin-timeout flagged (code {px_err}) — "
                  "numerics of some steps are unreliable", file=sys.stderr, flush=True)

    ms_per_step = elapsed / args.steps * 1000.0
    n_gpus = world if world > 1 else args.gpus
    value = n_gpus * B * T / (elapsed / args.steps)

    if pixel:
        model_desc = ("R2D2-DPG quadruped-run-pixels (84x84x3 uint8, act 12), "
                      "Conv(32,64,64)+LSTM H=" + str(args.hidden))
    else:
        model_desc = ("R2D2-DPG " + args.env + " (obs "
                      + str(spec.obs_shape[0]) + ", act " + str(spec.action_dim)
                      + "), LSTM H=" + str(args.hidden) + " trunk 2x" + str(args.hidden))
    if rank == 0:
        out = {
            "metric": "learner transitions/sec (whole node), " + args.env,
            "value": round(value, 1),
            "unit": "transitions/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": args.dtype,
            "data": "synthetic",
            "config": {
                "model": model_desc,
                "global_batch": B * n_gpus,
                "seq_len": args.seq_len,
                "burn_in": args.burn_in,
                "train_len": T,
                "n_step": cfg.replay.n_step,
                "parallelism": f"dp{n_gpus}",
                "replay": "GPU shard, prioritized (K5/K6), per-rank"
                          + (", fp8 obs (K7)" if args.fp8_obs else ""),
                "hipgraph": graphed,
            },
        }
        print(json.dumps(out), flush=True)
    return 0


if __name__ == "__main__":
    sys.exit(main())
