"""Flagship step-time A/B: budget machinery on/off + drain statistics."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from pytorch_r2d2_dpg_amd.config import Config
from pytorch_r2d2_dpg_amd.envs import make_env
from pytorch_r2d2_dpg_amd.learner import Learner
from pytorch_r2d2_dpg_amd.ops import hip_ext

cfg = Config()
cfg.env.name = "humanoid-run"
cfg.model.hidden_size = 512
cfg.model.trunk_sizes = (512, 512)
cfg.model.dtype = "bf16"
cfg.replay.seq_len = 80
cfg.replay.burn_in = 40
cfg.learner.batch_size = 256
spec = make_env(cfg.env.name).spec
lrn = Learner(cfg, spec, device="cuda")
lrn.collect_stats = False
B, L, H = 256, 85, 512
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
for _ in range(15):
    lrn.train_step(batch)
torch.cuda.synchronize()
s0 = hip_ext().px_budget_stats()
t0 = time.perf_counter()
N = 60
for _ in range(N):
    lrn.train_step(batch)
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / N * 1000
s1 = hip_ext().px_budget_stats()
print(f"ms_per_step={dt:.3f} launches={s1[0]-s0[0]} drains={s1[1]-s0[1]} "
      f"inflight_now={s1[2]} budget_env={os.environ.get('R2D2_PX_NO_BUDGET','')}", flush=True)
