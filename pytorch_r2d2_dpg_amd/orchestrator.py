"""Node orchestrator.

Wires the full distributed run on one node: k learner ranks (one per GPU,
launched via torchrun / torch.distributed.run), each owning a GPU replay
shard and a slice of the CPU actor pool. Ingest is drained between learner
steps (single-writer contract); weights are published to actors through the
shared-memory buffer every sync_every steps; rank 0 checkpoints and logs.
"""

from __future__ import annotations

import os
import threading
import time
from typing import Optional

import numpy as np
import torch

from .checkpoint import load_checkpoint, save_checkpoint
from .config import Config
from .envs import make_env
from .learner import Learner
from .parallel.actor_pool import ActorPool
from .parallel.dist import DistContext
from .replay.gpu_buffer import GpuSequenceReplay
from .replay.staging import StagingRing
from .utils.metrics import MetricsLogger


# px spin-timeout surveillance lives in the LEARNER now (it polls the error
# word every R2D2_PX_CHECK_EVERY steps and raises on corruption unless
# R2D2_PX_ERR_WARN=1); the orchestrator just logs the accumulated counter.


def run_training(cfg: Config, device: Optional[str] = None, resume: Optional[str] = None,
                 log_path: Optional[str] = None, max_steps: Optional[int] = None):
    dist = DistContext.from_env()
    rank, world = dist.rank, dist.world_size
    if device is None:
        if torch.cuda.is_available():
            local = int(os.environ.get("LOCAL_RANK", rank))
            device = f"cuda:{local}"
            torch.cuda.set_device(device)
        else:
            device = "cpu"

    spec = make_env(cfg.env.name).spec
    R = cfg.replay
    replay = GpuSequenceReplay(
        capacity=R.capacity, stored_len=R.seq_len + R.n_step,
        obs_shape=spec.obs_shape, action_dim=spec.action_dim,
        hidden_size=cfg.model.hidden_size, per_alpha=R.per_alpha, per_beta=R.per_beta,
        device=device, fp8_obs=R.fp8_obs, seed=cfg.learner.seed + 1000 * rank,
        obs_dtype=torch.uint8 if spec.pixel else torch.float32,
        store_dtype=torch.bfloat16 if cfg.model.dtype == "bf16" else torch.float32,
    )
    learner = Learner(cfg, spec, device=device, dist=dist, replay=replay)
    if resume:
        load_checkpoint(learner, resume)

    actors_per_rank = max(1, cfg.actor.num_actors // world)
    if torch.device(device).type == "cpu":
        # leave cores for the actor processes: a full-width intra-op thread
        # pool spin-waits on preempted workers and slows train_step ~80x
        # (measured 2.45s -> 0.031s/step at 8 cores with 2 actors)
        torch.set_num_threads(max(2, (os.cpu_count() or 8) // world
                              - actors_per_rank - 1))
    pool = ActorPool(cfg, actors_per_rank, base_actor_id=rank * actors_per_rank,
                     eval_actors=cfg.actor.eval_actors if rank == 0 else 0)
    pool.start()
    logger = MetricsLogger(log_path, stdout_every=1, rank=rank)

    staging = StagingRing(replay)

    # ---- background ingest thread: drains stacked-array payloads and
    # stages them into the shard off the learner's critical path. Storage
    # writes + tree publish enqueue on the learner's stream, so sampling
    # (same stream) never sees a torn slot (§3.4 single-writer contract:
    # this thread is the only writer).
    stop_ingest = threading.Event()
    # The threaded ingest is only safe on GPU, where every add_batch op
    # enqueues on the learner's HIP stream (stream-serialized with its
    # sampling — §3.4 single-writer contract). On CPU there is no stream:
    # a background-thread writer races the learner's index_select and the
    # sum-tree recompute, silently corrupting sampled batches (observed as
    # an orchestrated CPU pendulum run that never learns while the same
    # data plane driven synchronously does), so CPU always ingests
    # synchronously between learner steps.
    sync_ingest = bool(os.environ.get("R2D2_SYNC_INGEST")) or not replay.is_cuda

    def ingest_once():
        pool.drain_episodes()
        # bounded: a slow (CPU) learner must not drain the actors' whole
        # backlog each step — full rings throttle actors via drop-newest
        pool.ingest_rings(staging, max_per_ring=64)

    def ingest_loop():
        if replay.is_cuda:
            torch.cuda.set_device(replay.device)
        while not stop_ingest.is_set():
            pool.drain_episodes()
            pool.ingest_rings(staging)
            time.sleep(0.02)  # pace: batch ~dozens of sequences per pass

    ingest_thread = None
    if not sync_ingest:
        ingest_thread = threading.Thread(target=ingest_loop, daemon=True)
        ingest_thread.start()

    steps = max_steps if max_steps is not None else cfg.learner.steps
    try:
        # ---- warm the replay
        while len(replay) < max(R.min_fill, cfg.learner.batch_size):
            if sync_ingest:
                ingest_once()
            pool.watchdog()
            time.sleep(0.05)

        t_last = time.time()
        trans_acc = 0
        env_steps_last = pool.aggregate_env_steps()
        for step in range(learner.step_count, steps):
            if sync_ingest:
                ingest_once()
            batch = replay.sample(cfg.learner.batch_size)
            stats = learner.train_step(batch)
            trans_acc += stats.transitions
            if step % cfg.learner.sync_every == 0:
                sd = learner.actor_state_for_sync()
                pool.publish_weights(sd["actor"], sd["critic"])
                pool.watchdog()
            if rank == 0 and cfg.learner.checkpoint_every and \
                    (step + 1) % cfg.learner.checkpoint_every == 0:
                save_checkpoint(learner, cfg.learner.checkpoint_dir)
            if step % max(cfg.learner.log_every, 1) == 0:
                now = time.time()
                dt = max(now - t_last, 1e-9)
                env_steps = pool.aggregate_env_steps()
                recent = [r for _, r in pool.episode_returns[-20:]]
                logger.log(
                    step,
                    critic_loss=stats.critic_loss, actor_loss=stats.actor_loss,
                    mean_q=stats.mean_q,
                    learner_transitions_per_s=trans_acc / dt,
                    actor_fps=(env_steps - env_steps_last) / dt,
                    replay_size=len(replay), replay_added=replay.total_added,
                    episode_return=float(np.mean(recent)) if recent else 0.0,
                    eval_return=float(np.mean([r for _, r in pool.eval_returns[-10:]]))
                    if pool.eval_returns else 0.0,
                    actor_respawns=pool.respawn_count,
                    ring_drops=pool.aggregate_drops(),
                    px_err=learner.px_err_total,
                )
                t_last, trans_acc, env_steps_last = now, 0, env_steps
    finally:
        stop_ingest.set()
        if ingest_thread is not None:
            ingest_thread.join(timeout=5)
        pool.stop()
        logger.close()
    return learner
