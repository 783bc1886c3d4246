"""A/B timing: attention forward v3 (8-wave 32x32) vs round-1 swapped.

Run twice with SKY_ATTN_FWD_V3=1/0 (the launcher caches the env at first
dispatch).  Prints ms and TF/s at the flagship shape (causal FLOPs)."""
import os
import sys
import time

import torch

from skypilot_amd import ops

dev = torch.device("cuda:0")
B = int(os.environ.get("AB_B", 4)); S = int(os.environ.get("AB_S", 4096)); Hq, Hkv, D = 32, 8, 128
torch.manual_seed(0)
q = (torch.randn(B, S, Hq, D, device=dev) * 0.5).bfloat16()
k = (torch.randn(B, S, Hkv, D, device=dev) * 0.5).bfloat16()
v = (torch.randn(B, S, Hkv, D, device=dev) * 0.5).bfloat16()
C = ops.native()

label = os.environ.get("SKY_ATTN_FWD_V3", "1")
CAUSAL = os.environ.get("SKY_BENCH_CAUSAL", "1") == "1"
O, lse = C.attn_fwd(q, k, v, D ** -0.5, CAUSAL)
torch.cuda.synchronize()
iters = 20
t0 = time.perf_counter()
for _ in range(iters):
    C.attn_fwd(q, k, v, D ** -0.5, CAUSAL)
torch.cuda.synchronize()
ms = (time.perf_counter() - t0) / iters * 1e3
# causal flops: 4 * B*Hq*D * S^2/2 (fwd = 2 matmuls)
fl = 4.0 * B * Hq * D * S * S / (2 if CAUSAL else 1)
print(f"[fwd v3={label} B={B} S={S} causal={CAUSAL}] {ms:.3f} ms  {fl / (ms * 1e-3) / 1e12:.0f} TF/s")
sys.stdout.flush()
