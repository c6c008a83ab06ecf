"""pytorch-r2d2-dpg_amd — MI355X-native R2D2-DPG distributed RL engine.

A from-scratch rebuild of the capability set of ``jinbeizame007/pytorch-r2d2-DPG``
(see SURVEY.md — the reference tree was unavailable, so SURVEY.md §2 is the
behavioural spec): distributed actor/learner split, prioritized sequence replay
with stored LSTM hidden state and burn-in, and a deterministic actor-critic
(DDPG-style) update — re-designed MI355X-first:

* learner is data-parallel across the GPUs of one node (`torch.distributed`
  over RCCL/xGMI, one rank per GPU),
* the replay buffer is sharded into HBM3E per GPU with an on-GPU sum-tree,
* the hot ops (fused LSTM cell fwd/bwd, n-step TD-target reduction, Polyak
  soft update, sum-tree sample/priority-write, fp8 replay pack) are
  hand-written CDNA4 HIP kernels (gfx950, MFMA, LDS) — no Triton, no
  CUDA-compat shims, no multi-backend dispatch,
* CPU actor rollouts stream into GPU replay via a pinned staging ring and
  `hipMemcpyAsync` on a side stream.
"""

__version__ = "0.1.0"
