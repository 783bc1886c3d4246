#!/usr/bin/env python3
"""Headline benchmark: sky-launched Llama-3-8B bf16 training throughput.

Contract (driver-facing):
  python bench.py --gpus N --steps K --warmup W
N>1 is launched by the driver via torch.distributed.run (one rank per
GPU over RCCL).  W untimed warmup steps, then exactly K timed steps
bracketed by barrier + torch.cuda.synchronize on both sides; elapsed time
is MAX over ranks; rank 0 prints one JSON line.

Metric: tokens/sec aggregated over the whole job (weak scaling: per-GPU
work fixed at micro_batch x seq_len as N grows).  BASELINE.md publishes
no reference number for this metric (SkyPilot is an orchestrator), so
vs_baseline is null.
"""
from __future__ import annotations

import argparse
import json
import os
import sys

import torch
import torch.distributed as dist


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=8)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--model", default="llama3-8b")
    ap.add_argument("--micro-batch", type=int, default=6)
    ap.add_argument("--seq-len", type=int, default=4096)
    args = ap.parse_args()

    if not torch.cuda.is_available():
        print(json.dumps({"error": "no GPU available"}))
        return 1

    from skypilot_amd.train.trainer import TrainConfig, Trainer, \
        setup_distributed

    rank, world, local_rank = setup_distributed()
    if world == 1 and args.gpus > 1:
        print("error: --gpus > 1 requires torchrun (one rank per GPU)",
              file=sys.stderr)
        return 2

    cfg = TrainConfig(model=args.model, micro_batch=args.micro_batch,
                      seq_len=args.seq_len)
    tr = Trainer(cfg)

    import time

    def barrier_sync():
        if world > 1:
            dist.barrier()
        torch.cuda.synchronize()

    for _ in range(args.warmup):
        tr.train_step()
    barrier_sync()
    t0 = time.perf_counter()
    loss = 0.0
    for _ in range(args.steps):
        loss = tr.train_step()
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # Max over ranks.
    if world > 1:
        t = torch.tensor([elapsed], device=tr.device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    peak_gb = torch.cuda.max_memory_allocated() / 1e9
    print(f"[bench] rank {rank}: peak GPU memory {peak_gb:.1f} GB",
          file=sys.stderr, flush=True)

    tokens = args.steps * cfg.micro_batch * cfg.seq_len * world
    value = tokens / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if rank == 0:
        result = {
            "metric": "tokens/sec (node) sky-launched Llama-3-8B",
            "value": value,
            "unit": "tokens/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": cfg.micro_batch * world,
                "seq_len": cfg.seq_len,
                "parallelism": f"dp{world}",
                "final_loss": loss,
            },
        }
        print(json.dumps(result), flush=True)
    if world > 1:
        dist.destroy_process_group()
    return 0


if __name__ == "__main__":
    sys.exit(main())
