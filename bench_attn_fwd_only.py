"""Forward-only attention loop for rocprofv3 PMC profiling."""
import torch

from skypilot_amd import ops

dev = torch.device("cuda:0")
B, S, Hq, Hkv, D = 4, 4096, 32, 8, 128
torch.manual_seed(0)
q = (torch.randn(B, S, Hq, D, device=dev) * 0.5).bfloat16()
k = (torch.randn(B, S, Hkv, D, device=dev) * 0.5).bfloat16()
v = (torch.randn(B, S, Hkv, D, device=dev) * 0.5).bfloat16()
C = ops.native()
for _ in range(5):
    C.attn_fwd(q, k, v, D ** -0.5, True)
torch.cuda.synchronize()
print("done")
