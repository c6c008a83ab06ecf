"""Probe: gates_x projection layouts (VERDICT r1 item 9 / glue fusion).

The critic's input projection is [T*B, 512+A] @ [512+A, 4H] via
cat(feat, action) + addmm — K=533 for humanoid. Candidates:

  a) cat + addmm K=533 (current)
  b) addmm(b, feat, WxF) K=512 then out.addmm_(act, WxA) K=21 — no concat,
     library-friendly K
  c) same as (b) but the K=21 term via a single fused baddbmm? (skip —
     addmm_ is the simple form)

Interleaved rounds in one process (guide rule 24); CUDA-event timed.
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch


def bench(fn, iters=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters


def main():
    torch.manual_seed(0)
    T, B, F, A, H = 40, 256, 512, 21, 512
    M = T * B
    dt = torch.bfloat16
    feat = torch.randn(M, F, device="cuda", dtype=dt)
    act = torch.randn(M, A, device="cuda", dtype=dt)
    wx = torch.randn(4 * H, F + A, device="cuda", dtype=dt)
    b = torch.randn(4 * H, device="cuda", dtype=dt)
    wxf = wx[:, :F].contiguous()
    wxa = wx[:, F:].contiguous()

    def v_cat():
        x = torch.cat([feat, act], dim=-1)
        return torch.addmm(b, x, wx.t())

    def v_split():
        out = torch.addmm(b, feat, wxf.t())
        out.addmm_(act, wxa.t())
        return out

    # correctness
    ref = v_cat().float()
    got = v_split().float()
    err = (ref - got).abs().max().item()

    rounds = {"cat+addmm(K=533)": [], "split(K=512 + K=21)": []}
    for _ in range(5):  # interleaved rounds
        rounds["cat+addmm(K=533)"].append(bench(v_cat))
        rounds["split(K=512 + K=21)"].append(bench(v_split))
    for k, v in rounds.items():
        print(f"{k:24s} median={sorted(v)[len(v)//2]*1000:8.1f}us "
              f"min={min(v)*1000:8.1f}us")
    print(f"max abs diff: {err:.4f}")

    # pure K shapes for reference
    for K in (512, 533, 544):
        x = torch.randn(M, K, device="cuda", dtype=dt)
        w = torch.randn(4 * H, K, device="cuda", dtype=dt)
        t = bench(lambda: torch.addmm(b, x, w.t()))
        fl = 2 * M * K * 4 * H
        print(f"addmm K={K}: {t*1000:7.1f}us  {fl/(t/1000)/1e12:6.0f} TF")


if __name__ == "__main__":
    main()
