"""Single-stream decode A/B: bf16 weights vs fp8 decode weights."""
import os
import time

import torch

from skypilot_amd.serve.engine import Engine

eng = Engine("llama3-8b", device="cuda:0", max_seq=2048, max_batch=4)
if os.environ.get("FP8", "0") == "1":
    n = eng.enable_fp8_decode()
    print(f"fp8 weights registered: {n}")
eng.start()
prompt = list(range(1, 65))
eng.generate(prompt, max_tokens=16)  # warmup + graph capture
t0 = time.perf_counter()
out = eng.generate(prompt, max_tokens=256)
dt = time.perf_counter() - t0
print(f"decode: {len(out)} tokens in {dt:.3f}s = "
      f"{len(out)/dt:.1f} tok/s @1 stream")
