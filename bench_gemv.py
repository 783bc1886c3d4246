"""Per-shape decode-GEMV bench: skinny_gemm vs hipBLASLt (F.linear).
Reports effective weight-bandwidth (W bytes / time)."""
import torch, time
from skypilot_amd import ops
C = ops.native()
torch.manual_seed(0)
shapes = [(4096, 4096), (4096, 1024), (4096, 28672), (14336, 4096),
          (4096, 128256)]
for n in (1, 2, 4, 8):
    for i, o in shapes:
        x = (torch.randn(n, i, device="cuda") * 0.5).bfloat16()
        w = (torch.randn(o, i, device="cuda") * 0.02).bfloat16()
        for fn, name in ((lambda: C.skinny_gemm(x, w), "skinny"),
                         (lambda: torch.nn.functional.linear(x, w), "blaslt")):
            for _ in range(10):
                fn()
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(50):
                fn()
            torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / 50
            gbs = (o * i * 2) / dt / 1e9
            print(f"n={n} [{i:6d}->{o:6d}] {name}: {dt*1e6:8.1f} us "
                  f"{gbs:7.0f} GB/s", flush=True)
