"""Run the tr_b16 / permlane semantic probes and print mappings."""
import torch
from skypilot_amd import ops
C = ops.native()
ref = torch.zeros(1, device="cuda:0")
for mode, desc in [(0, "addr = lane*8 (contiguous b64)"),
                   (1, "uniform addr = 0"),
                   (2, "addr = ((l&15)+(l>>4)*64)*2")]:
    out = (C.trb16_probe(mode, ref).cpu().int() - 100)
    print(f"--- trb16 mode {mode}: {desc}")
    for l in range(0, 64, 2):
        rows = " | ".join(
            f"l{l+k:02d}:" + ",".join(f"{v:5d}" for v in out[l+k].tolist())
            for k in range(2))
        print(rows)
p = C.permlane_probe(ref).cpu().int()
print("--- permlane32_swap(a=1000+lane, b=2000+lane) -> (out0, out1)")
for l in range(64):
    print(f"  lane {l:2d}: out0={p[l,0].item()} out1={p[l,1].item()}")
