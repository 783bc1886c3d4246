"""Per-shape GEMM microbench: the exact Llama-3-8B training GEMMs
(fwd + dgrad + wgrad as autograd issues them) via hipBLASLt."""
import time

import torch

dev = torch.device("cuda:0")
T = 6 * 4096  # tokens at mb6
shapes = [
    ("qkv   ", T, 4096, 6144),
    ("o_proj", T, 4096, 4096),
    ("gateup", T, 4096, 28672),
    ("down  ", T, 14336, 4096),
    ("lmhead", T, 4096, 128256),
]

def bench(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters

total = {"fwd": 0.0, "dgrad": 0.0, "wgrad": 0.0}
print(f"{'name':>7} {'M':>6} {'K':>6} {'N':>7} | fwd TF  dgrad TF  wgrad TF")
for name, M, K, N in shapes:
    x = torch.randn(M, K, device=dev).bfloat16()
    w = torch.randn(N, K, device=dev).bfloat16()   # nn.Linear layout
    dy = torch.randn(M, N, device=dev).bfloat16()
    f = 2 * M * K * N
    t_fwd = bench(lambda: torch.nn.functional.linear(x, w))
    t_dgrad = bench(lambda: dy @ w)                 # [M,N]x[N,K]
    t_wgrad = bench(lambda: dy.t() @ x)             # [N,M]x[M,K]
    total["fwd"] += t_fwd; total["dgrad"] += t_dgrad; total["wgrad"] += t_wgrad
    print(f"{name} {M:6d} {K:6d} {N:7d} | {f/t_fwd/1e12:7.0f} {f/t_dgrad/1e12:8.0f} {f/t_wgrad/1e12:8.0f}")
tot = sum(total.values())
print(f"totals per step-ish: fwd {total['fwd']*1e3:.1f}ms dgrad {total['dgrad']*1e3:.1f}ms wgrad {total['wgrad']*1e3:.1f}ms  sum {tot*1e3:.1f}ms")
