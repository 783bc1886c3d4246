"""wgrad A/B: hipBLASLt TN (dy.t() @ x) vs transpose-then-NT kernel."""
import time

import torch

from skypilot_amd import ops

C = ops.native()
dev = torch.device("cuda:0")
M = 6 * 4096
shapes = [("qkv", 6144, 4096), ("o_proj", 4096, 4096),
          ("gateup", 28672, 4096), ("down", 4096, 14336),
          ("lmhead", 128256, 4096)]

def bench(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters

torch.manual_seed(0)
print(f"{'shape':>8} {'I':>7} {'J':>6} | blaslt_ms  ours_ms | blaslt_TF ours_TF")
for name, I, J in shapes:
    dy = (torch.randn(M, I, device=dev) * 0.1).bfloat16()
    x = (torch.randn(M, J, device=dev) * 0.1).bfloat16()
    fl = 2.0 * M * I * J
    t_ref = bench(lambda: dy.t() @ x)
    t_our = bench(lambda: C.wgrad_tn(dy, x))
    # numerics sanity at bench shape
    ours = C.wgrad_tn(dy, x)
    ref = (dy.t().float() @ x.float())
    err = (ours.float() - ref).abs()
    rel = float(err.max() / ref.abs().max())
    print(f"{name:>8} {I:7d} {J:6d} | {t_ref*1e3:8.2f} {t_our*1e3:8.2f} | "
          f"{fl/t_ref/1e12:8.0f} {fl/t_our/1e12:7.0f}  relmax={rel:.3e}")
    del dy, x, ours, ref, err
    torch.cuda.empty_cache()
