#!/usr/bin/env python3
"""Per-kernel microbenchmarks (attention fwd/bwd, rmsnorm, CE, adamw) at
the flagship Llama-3-8B shapes.  Prints achieved TFLOP/s / GB/s."""
from __future__ import annotations

import argparse
import time

import torch

from skypilot_amd import ops


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def attn_flops(B, S, Hq, D, causal=True):
    f = 2 * 2 * B * Hq * S * S * D  # QK^T + PV, MACs x2
    return f / 2 if causal else f


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=4)
    ap.add_argument("--seq", type=int, default=4096)
    args = ap.parse_args()
    dev = torch.device("cuda:0")
    B, S, Hq, Hkv, D = args.batch, args.seq, 32, 8, 128
    scale = D ** -0.5

    q = torch.randn(B, S, Hq, D, device=dev).bfloat16() * 0.5
    k = torch.randn(B, S, Hkv, D, device=dev).bfloat16() * 0.5
    v = torch.randn(B, S, Hkv, D, device=dev).bfloat16() * 0.5
    dO = torch.randn(B, S, Hq, D, device=dev).bfloat16() * 0.5
    C = ops.native()

    O, lse = C.attn_fwd(q, k, v, scale, True)
    t = timeit(lambda: C.attn_fwd(q, k, v, scale, True))
    f = attn_flops(B, S, Hq, D)
    print(f"attn_fwd     {t*1e3:8.3f} ms  {f/t/1e12:7.1f} TF/s")

    t = timeit(lambda: C.attn_bwd(q, k, v, O, dO, lse, scale, True))
    print(f"attn_bwd     {t*1e3:8.3f} ms  {2.5*f/t/1e12:7.1f} TF/s")

    # RMSNorm at [B*S, 4096]
    H = 4096
    x = torch.randn(B * S, H, device=dev).bfloat16()
    w = torch.ones(H, device=dev).bfloat16()
    inv = torch.empty(B * S, device=dev, dtype=torch.float32)
    t = timeit(lambda: C.rmsnorm_fwd(x, w, inv, 1e-5))
    gb = 2 * x.numel() * 2 / 1e9
    print(f"rmsnorm_fwd  {t*1e3:8.3f} ms  {gb/t:7.0f} GB/s")
    y = C.rmsnorm_fwd(x, w, inv, 1e-5)
    t = timeit(lambda: C.rmsnorm_bwd(x, w, y, inv))
    gb = 3 * x.numel() * 2 / 1e9
    print(f"rmsnorm_bwd  {t*1e3:8.3f} ms  {gb/t:7.0f} GB/s")

    # Cross entropy at [8192, 128256]
    N, V = 8192, 128256
    logits = torch.randn(N, V, device=dev).bfloat16()
    tgt = torch.randint(0, V, (N,), device=dev, dtype=torch.int32)
    loss = torch.empty(N, device=dev, dtype=torch.float32)

    def ce():
        C.cross_entropy_fused(logits, tgt, 1.0 / N, -100)
    t = timeit(ce, iters=10)
    gb = 3 * logits.numel() * 2 / 1e9  # 2 reads + 1 write
    print(f"cross_entropy{t*1e3:8.3f} ms  {gb/t:7.0f} GB/s")

    # AdamW at 8B-ish params (one big tensor slice)
    n = 500_000_000
    p = torch.zeros(n, device=dev, dtype=torch.bfloat16)
    mp = torch.zeros(n, device=dev, dtype=torch.float32)
    g = torch.randn(n, device=dev, dtype=torch.bfloat16)
    m = torch.zeros_like(mp)
    vv = torch.zeros_like(mp)
    t = timeit(lambda: C.adamw_step([p], [mp], [g], [m], [vv], 1e-4, 0.9,
                                    0.95, 1e-8, 0.1, 1, 1.0, [True]),
               iters=5, warmup=2)
    gb = n * (4 * 3 * 2 + 2 * 2) / 1e9  # m,v,master r/w + p write + g read
    print(f"adamw        {t*1e3:8.3f} ms  {gb/t:7.0f} GB/s "
          f"({n/1e6:.0f}M params)")

    # RoPE
    x = torch.randn(B * S, Hq, D, device=dev).bfloat16()
    cos = torch.randn(S, D // 2, device=dev)
    sin = torch.randn(S, D // 2, device=dev)
    pos = torch.arange(S, device=dev, dtype=torch.int32).repeat(B)
    t = timeit(lambda: C.rope(x, cos, sin, pos, False))
    gb = 2 * x.numel() * 2 / 1e9
    print(f"rope         {t*1e3:8.3f} ms  {gb/t:7.0f} GB/s")

    # Decode GEMV (n=1, wq shape) — weight-BW roofline kernel
    xg = torch.randn(1, 4096, device=dev).bfloat16()
    wg = (torch.randn(4096, 4096, device=dev) * 0.02).bfloat16()
    t = timeit(lambda: C.skinny_gemm(xg, wg))
    print(f"skinny_gemv  {t*1e6:8.1f} us  {wg.numel()*2/t/1e9:7.0f} GB/s "
          f"(n=1 4096x4096)")

    # Fused residual+rmsnorm (decode shape)
    xr = torch.randn(1, 4096, device=dev).bfloat16()
    rr = torch.randn(1, 4096, device=dev).bfloat16()
    wr = torch.randn(4096, device=dev).bfloat16()
    t = timeit(lambda: C.rmsnorm_res(xr, rr, wr, 1e-5))
    print(f"rmsnorm_res  {t*1e6:8.1f} us  (decode n=1)")


if __name__ == "__main__":
    main()
