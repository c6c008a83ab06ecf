"""Aggregate actor-FPS measurement at configured scale.

Runs the FULL orchestrated path (actor processes + shm rings + pinned
staging + learner training concurrently) for a bounded number of learner
steps, then prints one JSON summary line: steady-state aggregate actor FPS,
learner transitions/s, ring-drop count, respawns.

Usage (GPU box; also runs on CPU):
  python tools/actor_fps_probe.py --config configs/cheetah_1gpu.toml \
      --steps 400 [--set actor.num_actors=64 ...]
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from pytorch_r2d2_dpg_amd.config import Config  # noqa: E402
from pytorch_r2d2_dpg_amd.orchestrator import run_training  # noqa: E402


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--config", required=True)
    p.add_argument("--steps", type=int, default=400)
    p.add_argument("--log", default="gpurun_out/actor_fps.jsonl")
    p.add_argument("--set", nargs="*", default=[], dest="overrides")
    args = p.parse_args()

    cfg = Config.from_toml(args.config)
    cfg.apply_overrides(args.overrides)
    cfg.learner.checkpoint_every = 0
    cfg.learner.log_every = 20
    os.makedirs(os.path.dirname(args.log) or ".", exist_ok=True)

    t0 = time.time()
    run_training(cfg, log_path=args.log, max_steps=args.steps)
    wall = time.time() - t0

    # steady state = the second half of the logged windows
    rows = [json.loads(l) for l in open(args.log)]
    rows = [r for r in rows if "actor_fps" in r]
    half = rows[len(rows) // 2:]
    if not half:
        print(json.dumps({"error": "no log rows"}))
        return 1
    fps = [r["actor_fps"] for r in half]
    tps = [r["learner_transitions_per_s"] for r in half]
    out = {
        "config": os.path.basename(args.config),
        "num_actors": cfg.actor.num_actors,
        "envs_per_actor": cfg.actor.envs_per_actor,
        "total_envs": cfg.actor.num_actors * max(cfg.actor.envs_per_actor, 1),
        "steps": args.steps,
        "wall_s": round(wall, 1),
        "actor_fps_mean": round(sum(fps) / len(fps), 1),
        "actor_fps_max": round(max(fps), 1),
        "learner_tps_mean": round(sum(tps) / len(tps), 1),
        "ring_drops": half[-1].get("ring_drops"),
        "actor_respawns": half[-1].get("actor_respawns"),
        "px_err": half[-1].get("px_err"),
        "replay_added": half[-1].get("replay_added"),
    }
    print(json.dumps(out), flush=True)
    return 0


if __name__ == "__main__":
    sys.exit(main())
