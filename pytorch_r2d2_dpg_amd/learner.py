"""Learner — the GPU-resident training loop.

Per step: sample prioritized sequence batch from the local replay shard →
burn-in (no grad) to rebuild LSTM states at the training-window start →
target-net unroll → fused n-step TD targets (K3) → critic update → actor
update (ascend grad-Q through the critic) → fused Polyak (K4) → priority
write-back (K6) → periodic weight publish / checkpoint.

Padded tail steps of episode-flushed sequences (``batch["length"]`` <
stored length) are masked out of the critic loss, the actor loss and the
K3 priority reduction; time-limit truncation bootstraps through the
repeated final observation (replay/sequence.py).

Data-parallel: each rank runs this identically on its own shard; gradients
are all-reduced as one fused bucket over RCCL/xGMI after each backward
(parallel/dist.py) before the Adam step — DP ranks therefore hold identical
weights at all times; rank 0 alone checkpoints/logs. Stream-ordering
contracts for the multi-stream execution are documented in docs/STREAMS.md.

Execution (GPU): independent unrolls share paired (grid.z=2) column-split
launches — the two burn-ins, the target-actor unroll with the online
critic forward, and (bf16) the two critic-LSTM backwards; the actor chain
(through the PRE-update critic) runs on a side HIP stream overlapped with
the target/critic chain at every batch size (the round-1 B/seq gates are
gone — the CU-budget px launcher removed the residency livelocks); at
world>1 the critic+actor gradients go out as ONE fused RCCL bucket on the
main stream. ``capture_graph()`` can record the whole update into a
hipGraph, but eager + stream overlap measures faster at these sizes
(steady-state device occupancy is already ~92%), so it is not the default.

The step is ONE code path (`_step_core`): shared unpack → shared burn-in →
a pluggable middle (`_mid_pair_bwd` = manual paired-K2 backward, default on
GPU; `_mid_autograd` = plain autograd, CPU + fallback) → shared tail
(all-reduce, clip, optimizer steps, Polyak, priority write-back). The two
middles are parameter-equivalent (test_pair_bwd_step_matches_default).
"""

from __future__ import annotations

import contextlib
import os
from dataclasses import dataclass
from typing import Dict, Optional

import torch

from .config import Config
from .models import make_models
from .ops.polyak import PolyakUpdater
from .ops.lstm import _FusedLSTMSeq, lstm_seq_pair, lstm_seq_pair_halfgrad
from .ops.td import td_targets_and_priorities
from .parallel.dist import DistContext


def _null_ctx():
    return contextlib.nullcontext()


@dataclass
class StepStats:
    critic_loss: float
    actor_loss: float
    mean_q: float
    mean_priority: float
    transitions: int


class Learner:
    def __init__(self, cfg: Config, spec, device: str = "cpu",
                 dist: Optional[DistContext] = None, replay=None):
        self.cfg = cfg
        self.spec = spec
        self.device = torch.device(device)
        self.dist = dist or DistContext.single()
        self.dtype = torch.bfloat16 if cfg.model.dtype == "bf16" else torch.float32
        if self.device.type == "cuda":
            from .ops import enable_tunableop
            enable_tunableop()  # shipped gfx950 GEMM solution table (+4%)

        self.actor, self.critic, self.target_actor, self.target_critic = make_models(
            spec, cfg.model, device=self.device, dtype=self.dtype
        )
        # DP ranks start from identical weights
        self.dist.broadcast_params(list(self.actor.parameters()) + list(self.critic.parameters()))
        with torch.no_grad():
            for t, o in zip(self.target_actor.parameters(), self.actor.parameters()):
                t.copy_(o)
            for t, o in zip(self.target_critic.parameters(), self.critic.parameters()):
                t.copy_(o)

        lc = cfg.learner
        fused_ok = self.device.type == "cuda"
        try:
            self.opt_actor = torch.optim.Adam(self.actor.parameters(), lr=lc.lr_actor,
                                              eps=lc.adam_eps, fused=fused_ok)
            self.opt_critic = torch.optim.Adam(self.critic.parameters(), lr=lc.lr_critic,
                                               eps=lc.adam_eps, fused=fused_ok,
                                               weight_decay=lc.critic_weight_decay)
        except (RuntimeError, ValueError):  # fused unsupported on this build
            self.opt_actor = torch.optim.Adam(self.actor.parameters(), lr=lc.lr_actor,
                                              eps=lc.adam_eps)
            self.opt_critic = torch.optim.Adam(self.critic.parameters(), lr=lc.lr_critic,
                                               eps=lc.adam_eps,
                                               weight_decay=lc.critic_weight_decay)
        self.polyak_actor = PolyakUpdater(
            list(self.target_actor.parameters()), list(self.actor.parameters()))
        self.polyak_critic = PolyakUpdater(
            list(self.target_critic.parameters()), list(self.critic.parameters()))

        self.replay = replay
        self.step_count = 0
        self.collect_stats = True

        R = cfg.replay
        self.burn_in = R.burn_in
        self.train_len = cfg.train_len
        self.n_step = R.n_step
        self.gamma = R.gamma
        self.eta = R.priority_eta

        self._graph = None
        self._static: Optional[Dict[str, torch.Tensor]] = None
        self._graph_stats: Optional[Dict[str, torch.Tensor]] = None

        # stream-overlap machinery (GPU only): the actor chain runs on a side
        # stream concurrently with the target/critic chain
        self._actor_params = list(self.actor.parameters())
        # Stream overlap + pairing is the default at EVERY batch size now.
        # Round 1 gated it to B <= 256 and train_len <= 48 after observing
        # CU-residency crawls/livelocks between concurrent column-split
        # grids; both failure modes are gone under the round-2 launcher
        # (CU-budget accounting with drain-on-over-subscription,
        # bindings.cpp px_budget_*) — measured: seq-120 B=256 +42% and
        # B=512/seq-120 +31% from lifting the gates, zero px spin
        # timeouts. R2D2_NO_OVERLAP=1 forces the sequential chains.
        overlap_ok = (self.device.type == "cuda"
                      and not os.environ.get("R2D2_NO_OVERLAP"))
        if overlap_ok:
            self._side_stream = torch.cuda.Stream()
            self._ev_ba = torch.cuda.Event()   # burn-in done (main -> side)
            self._ev_ab = torch.cuda.Event()   # side chain done (side -> main)
        else:
            self._side_stream = None
            self._ev_ba = self._ev_ab = None
        # Paired backward (default on with overlap + bf16; R2D2_NO_PAIR_BWD=1
        # opts out): the critic-loss backward and the actor-chain critic
        # backward run as ONE paired (grid.z=2) K2 launch with manual
        # gradient assembly — two concurrent solo bwd grids contend for CU
        # residency instead (+3% measured; parameter-equivalent to the
        # autograd path, test_pair_bwd_step_matches_default).
        _H = cfg.model.hidden_size
        self._pair_bwd = (overlap_ok and self.dtype == torch.bfloat16
                          and _H >= 64 and (_H & (_H - 1)) == 0  # px geometry
                          and not os.environ.get("R2D2_NO_PAIR_BWD"))
        if self._pair_bwd:
            self._ev_bw = torch.cuda.Event()
            self._ev_ab2 = torch.cuda.Event()

        # Column-split spin-timeout surveillance: a px kernel that
        # times out its bounded spin flags a device error word and PROCEEDS
        # with stale data — poll the word every N steps (one tiny D2H read)
        # and FAIL LOUDLY instead of training on corrupted numerics.
        # R2D2_PX_ERR_WARN=1 downgrades to a warning + counter.
        self._px_check_every = int(os.environ.get("R2D2_PX_CHECK_EVERY", "50"))
        self.px_err_total = 0

        # optional hipEvent phase timers (SURVEY §5 tracing): R2D2_PHASE_TIMERS=1
        self._phase_events = None
        if self.device.type == "cuda" and os.environ.get("R2D2_PHASE_TIMERS"):
            self._phase_events = {k: (torch.cuda.Event(enable_timing=True),
                                      torch.cuda.Event(enable_timing=True))
                                  for k in ("burn_in", "target", "critic", "actor_opt")}

    def phase_times_ms(self) -> Dict[str, float]:
        """Elapsed ms per learner phase for the LAST step (synchronizes)."""
        if not self._phase_events:
            return {}
        torch.cuda.synchronize()
        return {k: s.elapsed_time(e) for k, (s, e) in self._phase_events.items()}

    def _pe(self, key: str, which: int):
        if self._phase_events:
            self._phase_events[key][which].record()

    # --------------------------------------------------------- shared pieces
    def _unpack(self, batch: Dict[str, torch.Tensor]):
        """Batch unpack + pad-mask construction, shared by both middles.

        Returns (obs [L,B,...], action [L,B,A], reward/done [B,L], weight [B],
        h_a, c_a, h_c, c_c [B,H], vmask [B,T] fp32 or None, valid [B] i32 or
        None). vmask/valid are None when the batch has no padded sequences
        (bench fast path — replay returns length=None then)."""
        obs = batch["obs"].transpose(0, 1)                         # [L, B, ...]
        if obs.dtype != torch.uint8:  # pixel obs stay uint8 (trunk normalizes)
            obs = obs.to(self.dtype)
        action = batch["action"].transpose(0, 1).to(self.dtype)    # [L, B, A]
        reward = batch["reward"]                                   # [B, L] fp32
        done = batch["done"]                                       # [B, L] fp32
        weight = batch["weight"]                                   # [B] fp32
        st = batch["state"].to(self.dtype)                         # [B, 4, H]
        h_a, c_a = st[:, 0].contiguous(), st[:, 1].contiguous()
        h_c, c_c = st[:, 2].contiguous(), st[:, 3].contiguous()
        length = batch.get("length")
        if length is not None:
            T = self.train_len
            # valid = real steps from the window start, UNCLAMPED into the
            # n-step tail (K3 masks reward reads beyond it); the loss mask
            # clamps to the training window
            valid = (length.to(torch.int32) - self.burn_in).clamp(1, T + self.n_step)
            vmask = (torch.arange(T, device=obs.device).unsqueeze(0)
                     < valid.to(torch.int64).unsqueeze(1)).to(torch.float32)
        else:
            valid = vmask = None
        return obs, action, reward, done, weight, h_a, c_a, h_c, c_c, vmask, valid

    def _burn_in(self, obs, action, h_a, c_a, h_c, c_c):
        """BOTH no-grad burn-in unrolls in ONE paired column-split launch
        (grid.z = 2) — a single grid co-dispatches uniformly, so the two
        halves genuinely parallelize (two px launches on separate streams
        contend for CU residency instead)."""
        bi = self.burn_in
        if bi == 0:
            return h_a, c_a, h_c, c_c
        with torch.no_grad():
            feat_a = self.actor.trunk(obs[:bi])
            feat_c = torch.cat([self.critic.trunk(obs[:bi]), action[:bi]], dim=-1)
            la, lc = self.actor.lstm, self.critic.lstm
            _, h_a2, c_a2, _, h_c2, c_c2 = lstm_seq_pair(
                feat_a, h_a, c_a, la.weight_x, la.weight_h, la.bias,
                feat_c, h_c, c_c, lc.weight_x, lc.weight_h, lc.bias)
        return h_a2, c_a2, h_c2, c_c2

    def _critic_loss(self, y, q_online_bt, weight, vmask):
        """Weighted MSE over REAL steps only (pad steps carry no loss)."""
        if vmask is None:
            return (weight.unsqueeze(1) * (y - q_online_bt) ** 2).mean()
        return ((weight.unsqueeze(1) * vmask * (y - q_online_bt) ** 2).sum()
                / vmask.sum())

    def _actor_loss(self, q_pi, vmask):
        """-mean Q over REAL steps (q_pi is [T, B]; vmask is [B, T])."""
        if vmask is None:
            return -q_pi.float().mean()
        m_tb = vmask.t()
        return -(q_pi.float() * m_tb).sum() / m_tb.sum()

    def _bump_weight_version(self):
        """Invalidate the HIP extension's fragment-major weight-pack cache
        after the optimizer/Polyak writes (ops/csrc/bindings.cpp)."""
        if self.device.type == "cuda":
            from .ops import hip_ext
            hip_ext().bump_weight_version()

    # ---------------------------------------------------------- core update
    def _step_core(self, batch: Dict[str, torch.Tensor],
                   update_priorities: bool = True,
                   zero_to_none: bool = True) -> Dict[str, torch.Tensor]:
        """The full device-side update. No host syncs, no Python floats.

        The actor update evaluates Q through the PRE-update critic (standard
        DDPG variant), which makes the actor chain independent of the critic
        optimizer — on GPU the two chains run on separate HIP streams (the
        LSTM unroll kernels occupy ~16 CUs each, so overlapping independent
        unrolls is nearly free parallelism). The CPU path runs the identical
        math on one stream.
        """
        cfg = self.cfg
        (obs, action, reward, done, weight,
         h_a, c_a, h_c, c_c, vmask, valid) = self._unpack(batch)
        overlap = self.device.type == "cuda" and self._side_stream is not None
        main = torch.cuda.current_stream() if overlap else None
        s1 = self._side_stream
        if overlap:
            s1.wait_stream(main)

        self._pe("burn_in", 0)
        h_a2, c_a2, h_c2, c_c2 = self._burn_in(obs, action, h_a, c_a, h_c, c_c)
        if overlap:
            self._ev_ba.record(main)

        mid = self._mid_pair_bwd if getattr(self, "_pair_bwd", False) \
            else self._mid_autograd
        critic_loss, actor_loss, actor_grads, prio, q_online_bt, join_ev = mid(
            obs, action, reward, done, weight, h_a2, c_a2, h_c2, c_c2,
            vmask, valid, overlap, main, s1, zero_to_none)

        # ---- shared tail: all-reduce -> clip -> join -> optimizer steps ----
        cparams = list(self.critic.parameters())
        fuse_buckets = self.dist.world_size > 1

        def _fill_actor_grads():
            for p, g in zip(self._actor_params, actor_grads):
                if overlap:
                    # g was allocated on the side stream but is consumed by
                    # the all-reduce/optimizer on main: mark the cross-stream
                    # use so the caching allocator does not recycle g for a
                    # later side-stream allocation while main's reads are in
                    # flight
                    g.record_stream(main)
                if p.grad is None:
                    p.grad = g.clone() if not zero_to_none else g
                else:
                    p.grad.copy_(g)

        if fuse_buckets:
            # DP: ONE fused collective per step. The grad bytes are ~2-20 MB
            # bf16 — firmly RCCL's latency regime on the xGMI mesh — so one
            # bucket for critic+actor beats two back-to-back collectives.
            # Requires the actor grads first -> join the side stream now.
            if join_ev is not None:
                main.wait_event(join_ev)
            _fill_actor_grads()
            self.dist.allreduce_grads(cparams + self._actor_params, key="all")
            if cfg.learner.max_grad_norm > 0:
                torch.nn.utils.clip_grad_norm_(cparams, cfg.learner.max_grad_norm)
            self.opt_critic.step()
            self._pe("critic", 1)
            self._pe("actor_opt", 0)
            if cfg.learner.max_grad_norm > 0:
                torch.nn.utils.clip_grad_norm_(self._actor_params,
                                               cfg.learner.max_grad_norm)
            self.opt_actor.step()
        else:
            if cfg.learner.max_grad_norm > 0:
                torch.nn.utils.clip_grad_norm_(cparams, cfg.learner.max_grad_norm)
            # the side stream reads critic params: join before the optimizer
            # writes
            if join_ev is not None:
                main.wait_event(join_ev)
            self.opt_critic.step()
            self._pe("critic", 1)
            self._pe("actor_opt", 0)
            _fill_actor_grads()
            if cfg.learner.max_grad_norm > 0:
                torch.nn.utils.clip_grad_norm_(self._actor_params,
                                               cfg.learner.max_grad_norm)
            self.opt_actor.step()

        # ---- Polyak soft target update (K4: one fused launch per net on GPU)
        # (branch is Python-side: under graph capture target_update_every == 1)
        if self.step_count % cfg.learner.target_update_every == 0:
            self.polyak_actor(cfg.learner.tau)
            self.polyak_critic(cfg.learner.tau)
        self._bump_weight_version()

        self._pe("actor_opt", 1)
        # ---- priority write-back (stays on-GPU with the K6 tree kernel)
        if update_priorities and self.replay is not None and batch.get("idx") is not None:
            self.replay.update_priorities(batch["idx"], prio)

        return {
            "critic_loss": critic_loss.detach(),
            "actor_loss": actor_loss.detach(),
            "mean_q": q_online_bt.detach().mean(),
            "mean_priority": prio.mean(),
        }

    # -------------------------------------------- middle: autograd backward
    def _mid_autograd(self, obs, action, reward, done, weight,
                      h_a2, c_a2, h_c2, c_c2, vmask, valid,
                      overlap, main, s1, zero_to_none):
        """Plain-autograd middle (CPU path + GPU fallback): actor chain on
        the side stream via autograd.grad; critic loss via .backward()."""
        cfg = self.cfg
        bi, T = self.burn_in, self.train_len
        Tn = T + self.n_step

        with torch.cuda.stream(s1) if overlap else _null_ctx():
            if overlap:
                s1.wait_event(self._ev_ba)
            # actor chain through the PRE-update critic
            mu, _ = self.actor(obs[bi:bi + T], (h_a2.detach(), c_a2.detach()))
            q_pi, _ = self.critic(obs[bi:bi + T], mu, (h_c2.detach(), c_c2.detach()))
            actor_loss = self._actor_loss(q_pi, vmask)
            actor_grads = torch.autograd.grad(actor_loss, self._actor_params)
            if overlap:
                self._ev_ab.record(s1)

        self._pe("burn_in", 1)
        self._pe("target", 0)
        # ---- target-actor unroll + critic online fwd in ONE paired
        # column-split launch (the two are independent: the online critic
        # reads the REPLAY action, not the target action), then the
        # target-critic unroll which consumes mu_t (main stream; balances
        # the side stream's actor-chain length)
        ta = self.target_actor
        with torch.no_grad():
            feat_ta = ta.trunk(obs[bi:bi + Tn])
        feat_cf = self.critic.trunk(obs[bi:bi + T])
        x_cf = torch.cat([feat_cf, action[bi:bi + T]], dim=-1)
        hs_ta, _, _, hs_cf, _, _ = lstm_seq_pair_halfgrad(
            feat_ta, h_a2, c_a2,
            ta.lstm.weight_x, ta.lstm.weight_h, ta.lstm.bias,
            x_cf, h_c2.detach(), c_c2.detach(),
            self.critic.lstm.weight_x, self.critic.lstm.weight_h,
            self.critic.lstm.bias)
        Hs = ta.hidden_size
        with torch.no_grad():
            mu_t = (torch.tanh(ta.head(hs_ta.reshape(-1, Hs)))
                    .view(Tn, -1, ta.action_dim) * ta.action_scale)
            qt, _ = self.target_critic(obs[bi:bi + Tn], mu_t, (h_c2, c_c2))
            q_boot = qt[self.n_step:self.n_step + T].transpose(0, 1).float()
        q_online = self.critic.head(hs_cf.reshape(-1, Hs)).view(T, -1)
        q_online_bt = q_online.transpose(0, 1).float()  # [B, T]
        with torch.no_grad():
            y, td_abs, prio = td_targets_and_priorities(
                reward[:, bi:], done[:, bi:], q_boot, q_online_bt.detach(),
                self.gamma, self.n_step, self.eta,
                rescale=cfg.learner.value_rescale, valid_n=valid,
            )
        self._pe("target", 1)
        self._pe("critic", 0)
        critic_loss = self._critic_loss(y, q_online_bt, weight, vmask)
        self.opt_critic.zero_grad(set_to_none=zero_to_none)
        critic_loss.backward()
        return (critic_loss, actor_loss, actor_grads, prio, q_online_bt,
                self._ev_ab if overlap else None)

    # ------------------------------------------ middle: paired-K2 backward
    def _mid_pair_bwd(self, obs, action, reward, done, weight,
                      h_a2, c_a2, h_c2, c_c2, vmask, valid,
                      overlap, main, s1, zero_to_none):
        """Default GPU middle at bf16 + overlap (R2D2_NO_PAIR_BWD=1 falls
        back to autograd): identical math, but the critic-loss backward and
        the actor-chain critic backward run as ONE paired (grid.z=2) K2
        launch with the surrounding dense gradients (gates GEMMs, trunk,
        heads) assembled manually — two concurrent solo bwd grids contend
        for CU residency instead. Parameter-equivalence is locked by
        test_pair_bwd_step_matches_default."""
        from .ops import hip_ext
        cfg = self.cfg
        ext = hip_ext()
        bi, T = self.burn_in, self.train_len
        Tn = T + self.n_step
        H = self.actor.hidden_size

        lc = self.critic.lstm
        wx_c, wh_c, b_c = lc.weight_x, lc.weight_h, lc.bias
        obs_w = obs[bi:bi + T]
        c_c2d = c_c2.detach().contiguous()
        w_head = self.critic.head.weight          # [1, H]
        b_head = self.critic.head.bias

        # ---- side stream: actor chain FORWARD only + the q_pi loss seed.
        # The critic unroll here is a RAW stash forward (its weight grads
        # are discarded by design — pre-update-critic DDPG), so only dmu is
        # needed from its backward, which the paired K2 below provides. The
        # loss seed d(actor_loss)/d(hs_q) is ANALYTIC (-m/Σm ⊗ w_head): no
        # autograd graph touches the critic head here, which keeps the head
        # params' AccumulateGrad nodes on the main stream (docs/STREAMS.md).
        with torch.cuda.stream(s1):
            s1.wait_event(self._ev_ba)
            la = self.actor.lstm
            feat_a = self.actor.trunk(obs_w)
            hs_a, _ = _FusedLSTMSeq.apply(feat_a, h_a2.detach(), c_a2.detach(),
                                          la.weight_x, la.weight_h, la.bias)
            mu = (torch.tanh(self.actor.head(hs_a.reshape(-1, H)))
                  .view(T, -1, self.actor.action_dim) * self.actor.action_scale)
            with torch.no_grad():
                feat_qc = self.critic.trunk(obs_w)
                x_q = torch.cat([feat_qc, mu.detach()], dim=-1)
                B_ = x_q.shape[1]
                gx_q = torch.addmm(b_c, x_q.reshape(T * B_, -1),
                                   wx_c.t()).view(T, -1, 4 * H).contiguous()
                hs_q, _, ga_q, cs_q = ext.lstm_seq_fwd(
                    gx_q, h_c2.detach().contiguous(), c_c2d, wh_c.contiguous())
                q_pi = (hs_q.reshape(-1, H) @ w_head.t() + b_head).view(T, B_)
                actor_loss = self._actor_loss(q_pi, vmask)
                if vmask is None:
                    dhs_q = (w_head.mul(-1.0 / (T * B_)).to(hs_q.dtype)
                             .view(1, 1, H).expand(T, B_, H).contiguous())
                else:
                    m_tb = vmask.t()
                    dhs_q = ((m_tb / -m_tb.sum()).unsqueeze(-1).to(hs_q.dtype)
                             * w_head.view(1, 1, H)).contiguous()
            self._ev_ab.record(s1)

        self._pe("burn_in", 1)
        self._pe("target", 0)
        # ---- main: TA + CF (raw stash) paired fwd, TC, K3, critic loss
        ta = self.target_actor
        feat_cf = self.critic.trunk(obs_w)            # grad (trunk params)
        x_cf = torch.cat([feat_cf, action[bi:bi + T]], dim=-1)
        with torch.no_grad():
            feat_ta = ta.trunk(obs[bi:bi + Tn])
            gx_ta = torch.addmm(ta.lstm.bias,
                                feat_ta.reshape(Tn * feat_ta.shape[1], -1),
                                ta.lstm.weight_x.t()).view(Tn, -1, 4 * H).contiguous()
            gx_cf = torch.addmm(b_c, x_cf.detach().reshape(T * x_cf.shape[1], -1),
                                wx_c.t()).view(T, -1, 4 * H).contiguous()
            hs_ta, _, _, _, hs_cf, _, ga_cf, cs_cf = ext.lstm_seq_fwd_pair(
                gx_ta, h_a2.contiguous(), c_a2.contiguous(),
                ta.lstm.weight_h.contiguous(),
                gx_cf, h_c2.detach().contiguous(), c_c2d, wh_c.contiguous(),
                False, True)
            mu_t = (torch.tanh(ta.head(hs_ta.reshape(-1, H)))
                    .view(Tn, -1, ta.action_dim) * ta.action_scale)
            qt, _ = self.target_critic(obs[bi:bi + Tn], mu_t, (h_c2, c_c2))
            q_boot = qt[self.n_step:self.n_step + T].transpose(0, 1).float()
        hs_cf_leaf = hs_cf.detach().requires_grad_(True)
        q_online = self.critic.head(hs_cf_leaf.reshape(-1, H)).view(T, -1)
        q_online_bt = q_online.transpose(0, 1).float()
        with torch.no_grad():
            y, td_abs, prio = td_targets_and_priorities(
                reward[:, bi:], done[:, bi:], q_boot, q_online_bt.detach(),
                self.gamma, self.n_step, self.eta,
                rescale=cfg.learner.value_rescale, valid_n=valid,
            )
        self._pe("target", 1)
        self._pe("critic", 0)
        critic_loss = self._critic_loss(y, q_online_bt, weight, vmask)
        head_p = [self.critic.head.weight, self.critic.head.bias]
        dhs_cf, dw_head, db_head = torch.autograd.grad(
            critic_loss, [hs_cf_leaf] + head_p)

        # ---- joint paired K2: both critic-LSTM backwards in one launch
        main.wait_event(self._ev_ab)
        for t_ in (dhs_q, ga_q, cs_q):
            t_.record_stream(main)
        zero = torch.zeros_like(h_c2)
        try:
            dg_cf, _, _, dg_q, _, _ = ext.lstm_seq_bwd_pair(
                dhs_cf.contiguous(), zero, zero, ga_cf, cs_cf, c_c2d, wh_c,
                dhs_q.contiguous(), zero, zero, ga_q, cs_q, c_c2d, wh_c)
        except RuntimeError:  # paired geometry unsupported: two solo K2 calls
            dg_cf, _, _ = ext.lstm_seq_bwd(
                dhs_cf.contiguous(), zero, zero, ga_cf, cs_cf, c_c2d, wh_c)
            dg_q, _, _ = ext.lstm_seq_bwd(
                dhs_q.contiguous(), zero, zero, ga_q, cs_q, c_c2d, wh_c)
        self._ev_bw.record(main)

        # ---- side: finish the actor chain (dmu -> actor grads), overlapping
        # the critic dense-gradient GEMMs below
        dg_q.record_stream(s1)
        with torch.cuda.stream(s1):
            s1.wait_event(self._ev_bw)
            F_q = feat_qc.shape[-1]
            dx_q = (dg_q.reshape(-1, 4 * H) @ wx_c).view(T, -1, x_q.shape[-1])
            dmu = dx_q[..., F_q:].contiguous()
            actor_grads = torch.autograd.grad(mu, self._actor_params,
                                              grad_outputs=dmu)
            self._ev_ab2.record(s1)

        # ---- main: critic dense grads (same math as _FusedLSTMSeq.backward)
        B_ = x_cf.shape[1]
        dgf = dg_cf.reshape(T * B_, 4 * H)
        dwx_c = dgf.t() @ x_cf.detach().reshape(T * B_, -1)
        h_prev = torch.cat([h_c2.detach().unsqueeze(0), hs_cf[:-1]],
                           dim=0).reshape(T * B_, H)
        dwh_c = dgf.t() @ h_prev
        db_c = dgf.float().sum(dim=0).to(dgf.dtype)
        dx_cf = (dgf @ wx_c).view(T, B_, -1)
        F_cf = feat_cf.shape[-1]
        trunk_params = list(self.critic.trunk.parameters())
        trunk_grads = torch.autograd.grad(
            feat_cf, trunk_params, grad_outputs=dx_cf[..., :F_cf].contiguous()
        ) if trunk_params else []
        gmap = {id(p): g for p, g in zip(trunk_params, trunk_grads)}
        gmap[id(wx_c)] = dwx_c
        gmap[id(wh_c)] = dwh_c
        gmap[id(b_c)] = db_c
        gmap[id(head_p[0])] = dw_head
        gmap[id(head_p[1])] = db_head
        self.opt_critic.zero_grad(set_to_none=zero_to_none)
        for p in self.critic.parameters():
            g = gmap.get(id(p))
            if g is None:  # a None grad would be silently SKIPPED by the
                raise RuntimeError(  # all-reduce — fail loudly instead
                    "pair-bwd path: critic parameter without an assembled "
                    "gradient (model structure changed?)")
            if p.grad is None:
                p.grad = g if zero_to_none else g.clone()
            else:
                p.grad.copy_(g)
        return (critic_loss, actor_loss, actor_grads, prio, q_online_bt,
                self._ev_ab2)

    # ------------------------------------------------------------ graph cap
    def capture_graph(self, example_batch: Dict[str, torch.Tensor]) -> None:
        """Record the whole update into a hipGraph with static input buffers.

        Requires: GPU device, fixed batch shape, target_update_every == 1
        (the Polyak branch is baked into the recording).
        """
        assert self.device.type == "cuda", "graph capture needs a GPU"
        assert self.cfg.learner.target_update_every == 1
        # the paired-backward path uses record_stream (allocator hints),
        # which is not legal inside graph capture — capture the standard
        # autograd path instead (identical math). Restored on FAILURE so an
        # aborted capture doesn't silently change the eager path selection.
        pair_saved = self._pair_bwd
        self._pair_bwd = False
        try:
            keys = ("obs", "action", "reward", "done", "state", "weight", "idx")
            self._static = {k: example_batch[k].clone() for k in keys}
            # capturable Adam: state must live on device, step as device tensor
            for opt in (self.opt_actor, self.opt_critic):
                for gparam in opt.param_groups:
                    gparam["capturable"] = True
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                for _ in range(3):  # warmup allocations + optimizer state
                    self._step_core(self._static, zero_to_none=False)
            torch.cuda.current_stream().wait_stream(side)
            torch.cuda.synchronize()
            self._graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self._graph):
                self._graph_stats = self._step_core(self._static, zero_to_none=False)
        except Exception:
            self._pair_bwd = pair_saved
            self._graph = None
            raise

    # ------------------------------------------------------------ train step
    def train_step(self, batch: Dict[str, torch.Tensor]) -> StepStats:
        if self._graph is not None:
            for k, dst in self._static.items():
                src = batch[k]
                if src is not None:
                    dst.copy_(src, non_blocking=True)
            self._graph.replay()
            stats = self._graph_stats
        else:
            stats = self._step_core(batch)
        self.step_count += 1
        self._check_px_errors()
        B = batch["obs"].shape[0]
        if self.collect_stats:
            return StepStats(
                critic_loss=float(stats["critic_loss"]),
                actor_loss=float(stats["actor_loss"]),
                mean_q=float(stats["mean_q"]),
                mean_priority=float(stats["mean_priority"]),
                transitions=B * self.train_len,
            )
        return StepStats(0.0, 0.0, 0.0, 0.0, B * self.train_len)

    def _check_px_errors(self):
        """Poll the column-split spin-timeout error word every
        ``_px_check_every`` steps and fail loudly on corruption (the kernels
        proceed with stale data after a timed-out spin; silently training on
        that is the failure mode ADVICE r1 flagged)."""
        if (self.device.type != "cuda" or not self._px_check_every
                or self.step_count % self._px_check_every != 0):
            return
        from .ops import hip_ext
        err = int(hip_ext().lstm_px_error())
        if err:
            self.px_err_total += err
            msg = (f"column-split LSTM spin timeout flagged (code {err}) within "
                   f"the last {self._px_check_every} learner steps — numerics "
                   "of the affected steps are unreliable")
            if os.environ.get("R2D2_PX_ERR_WARN"):
                import warnings
                warnings.warn(msg)
            else:
                raise RuntimeError(
                    msg + " (set R2D2_PX_ERR_WARN=1 to continue anyway)")

    # -------------------------------------------------------------- weights
    def actor_state_for_sync(self) -> Dict[str, torch.Tensor]:
        """fp32 CPU state_dicts actors load (actor + critic for local priorities)."""
        return {
            "actor": {k: v.detach().float().cpu() for k, v in self.actor.state_dict().items()},
            "critic": {k: v.detach().float().cpu() for k, v in self.critic.state_dict().items()},
        }
