"""Offline hipBLASLt algorithm tuning for the trainer's GEMM shapes.

TunableOp sweeps hipBLASLt solutions per GEMM shape and records the
fastest in a CSV; bench.py / the trainer then load that CSV with tuning
OFF (pure lookup, no runtime cost).  Tuning is bounded hard via the
iteration/duration caps below — the shapes are exactly the Llama-3-8B
mb6 x 4096 training GEMMs (fwd + dgrad + wgrad come out of the same
fwd/bwd autograd pass).
"""
import os

os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "1")
os.environ.setdefault("PYTORCH_TUNABLEOP_VERBOSE", "1")
os.environ.setdefault("PYTORCH_TUNABLEOP_MAX_TUNING_ITERATIONS", "10")
os.environ.setdefault("PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS", "50")
os.environ.setdefault("PYTORCH_TUNABLEOP_MAX_WARMUP_ITERATIONS", "1")
os.environ.setdefault("PYTORCH_TUNABLEOP_MAX_WARMUP_DURATION_MS", "5")
os.environ.setdefault("PYTORCH_TUNABLEOP_ROCBLAS_ENABLED", "0")
os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME",
                      "gpurun_out/tunableop_gfx950.csv")

import time

import torch

def main():
    dev = "cuda:0"
    M = 6 * 4096
    h, m_, kv, vocab = 4096, 14336, 1024, 128256
    shapes = [  # (in_features, out_features) of every trainer Linear
        (h, h),          # wq / wo
        (h, kv),         # wk / wv
        (h, 2 * m_),     # w_gate_up (fused)
        (m_, h),         # w_down
        (h, vocab),      # lm_head
    ]
    weights = [torch.randn(o, i, device=dev, dtype=torch.bfloat16) * 0.02
               for i, o in shapes]
    for w in weights:
        w.requires_grad_(True)
    t0 = time.time()
    for it in range(3):  # first pass tunes; later passes hit the cache
        for (i, o), w in zip(shapes, weights):
            x = torch.randn(M, i, device=dev, dtype=torch.bfloat16,
                            requires_grad=True)
            y = torch.nn.functional.linear(x, w)
            y.backward(torch.randn_like(y))
            w.grad = None
        torch.cuda.synchronize()
        print(f"pass {it}: {time.time() - t0:.1f}s elapsed", flush=True)
    # force CSV flush
    torch.cuda.tunable.write_file()
    print("tuning done:", os.environ["PYTORCH_TUNABLEOP_FILENAME"])

if __name__ == "__main__":
    main()
