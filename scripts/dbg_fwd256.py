import os, sys
import torch
from skypilot_amd import ops
torch.manual_seed(7)
shapes = [(2, 256, 8, 2, True), (1, 256, 8, 8, True), (2, 512, 16, 4, True)]
C = ops.native()
for (B, S, Hq, Hkv, causal) in shapes:
    errs = []
    for rep in range(5):
        q = (torch.randn(B, S, Hq, 128, device="cuda") * 0.5).bfloat16()
        k = (torch.randn(B, S, Hkv, 128, device="cuda") * 0.5).bfloat16()
        v = (torch.randn(B, S, Hkv, 128, device="cuda") * 0.5).bfloat16()
        O, lse = C.attn_fwd(q, k, v, 128 ** -0.5, causal)
        qf, kf, vf = q.float(), k.float(), v.float()
        kk = kf.repeat_interleave(Hq // Hkv, dim=2)
        vv = vf.repeat_interleave(Hq // Hkv, dim=2)
        a = torch.einsum("bshd,bthd->bhst", qf, kk) * (128 ** -0.5)
        if causal:
            mask = torch.triu(torch.ones(S, S, device="cuda", dtype=torch.bool), 1)
            a = a.masked_fill(mask, float("-inf"))
        ref = torch.einsum("bhst,bthd->bshd", a.softmax(-1), vv)
        errs.append((O.float() - ref).abs().max().item())
    print(f"{(B,S,Hq,Hkv,causal)} errs={['%.4f' % e for e in errs]}", flush=True)
