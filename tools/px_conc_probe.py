"""Probe: do two concurrent column-split LSTM kernels on separate streams
starve each other? Times sequential vs concurrent launches at B in {256,512}."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from pytorch_r2d2_dpg_amd.ops import hip_ext

ext = hip_ext()
dev = "cuda"

def mk(B, H, T):
    g = torch.Generator().manual_seed(0)
    gates_x = (torch.randn(T, B, 4*H, generator=g) * 0.1).to(dev, torch.bfloat16)
    h0 = torch.zeros(B, H, dtype=torch.bfloat16, device=dev)
    c0 = torch.zeros(B, H, dtype=torch.bfloat16, device=dev)
    wh = (torch.randn(4*H, H, generator=g) / H**0.5).to(dev, torch.bfloat16)
    return gates_x, h0, c0, wh

for B in (256, 512):
    H, T = 512, 40
    a1 = mk(B, H, T); a2 = mk(B, H, T)
    s1, s2 = torch.cuda.Stream(), torch.cuda.Stream()
    # warm
    ext.lstm_seq_fwd(*a1); torch.cuda.synchronize()
    # sequential
    t0 = time.perf_counter()
    for _ in range(5):
        ext.lstm_seq_fwd(*a1); ext.lstm_seq_fwd(*a2)
    torch.cuda.synchronize()
    seq_ms = (time.perf_counter() - t0) / 5 * 1000
    # concurrent
    t0 = time.perf_counter()
    for _ in range(5):
        with torch.cuda.stream(s1):
            ext.lstm_seq_fwd(*a1)
        with torch.cuda.stream(s2):
            ext.lstm_seq_fwd(*a2)
        torch.cuda.synchronize()
    conc_ms = (time.perf_counter() - t0) / 5 * 1000
    print(f"B={B}: sequential(2x) {seq_ms:.2f} ms   concurrent {conc_ms:.2f} ms   "
          f"px_err={ext.lstm_px_error()}", flush=True)
