"""Minimal attention-only run for PMC profiling."""
import torch
from skypilot_amd import ops
dev = torch.device("cuda:0")
B, S, Hq, Hkv, D = 4, 4096, 32, 8, 128
q = (torch.randn(B, S, Hq, D, device=dev) * 0.5).bfloat16()
k = (torch.randn(B, S, Hkv, D, device=dev) * 0.5).bfloat16()
v = (torch.randn(B, S, Hkv, D, device=dev) * 0.5).bfloat16()
dO = (torch.randn(B, S, Hq, D, device=dev) * 0.5).bfloat16()
C = ops.native()
O, lse = C.attn_fwd(q, k, v, D ** -0.5, True)
for _ in range(3):
    C.attn_fwd(q, k, v, D ** -0.5, True)
    C.attn_bwd(q, k, v, O, dO, lse, D ** -0.5, True)
torch.cuda.synchronize()
print("done")
