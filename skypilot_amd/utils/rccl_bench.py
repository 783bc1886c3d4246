"""RCCL all-reduce bandwidth benchmark over xGMI.

Port of the reference's user-level NCCL test (reference:
examples/nccl_test.yaml — torch all_reduce, algbw/busbw; its published
output is 3.85 GB/s busbw over 2-node cloud TCP).  On one MI355X node
the ring is per-xGMI-link bound (~153 GB/s/link), so busbw should land
two orders of magnitude above the reference artifact.

Run under torchrun, one rank per GPU.
"""
from __future__ import annotations

import argparse
import json
import os
import time

import torch
import torch.distributed as dist


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--size-mb", type=int, default=1024)
    ap.add_argument("--iters", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--dtype", default="bf16")
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    use_gpu = torch.cuda.is_available()
    backend = "nccl" if use_gpu else "gloo"
    dist.init_process_group(backend, rank=rank, world_size=world)
    if use_gpu:
        torch.cuda.set_device(local_rank)
    dev = torch.device(f"cuda:{local_rank}" if use_gpu else "cpu")
    dtype = {"bf16": torch.bfloat16, "fp32": torch.float32}[args.dtype]

    n = args.size_mb * (1 << 20) // dtype.itemsize
    buf = torch.ones(n, dtype=dtype, device=dev)

    def sync():
        if use_gpu:
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        dist.all_reduce(buf)
    sync()
    dist.barrier()
    t0 = time.perf_counter()
    for _ in range(args.iters):
        dist.all_reduce(buf)
    sync()
    dt = (time.perf_counter() - t0) / args.iters

    size_bytes = n * dtype.itemsize
    algbw = size_bytes / dt / 1e9
    busbw = algbw * 2 * (world - 1) / world
    if rank == 0:
        print(json.dumps({
            "collective": "all_reduce", "backend": backend,
            "size_mb": args.size_mb, "dtype": args.dtype,
            "world_size": world, "iters": args.iters,
            "time_ms": dt * 1e3,
            "algbw_GBps": round(algbw, 2), "busbw_GBps": round(busbw, 2),
        }), flush=True)
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
