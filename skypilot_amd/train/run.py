"""Bundled training task entrypoint.

This is what the shipped task YAMLs run under the gang launcher:

    torchrun --standalone --nproc-per-node $SKYPILOT_NUM_GPUS_PER_NODE \
        --master-addr 127.0.0.1 -m skypilot_amd.train.run \
        --model llama3-8b --steps 200 --checkpoint-dir /ckpt

Checkpoint/resume follows the managed-jobs contract:
the checkpoint dir is a MOUNT-mode storage mount that survives
preemption; on restart the trainer resumes from the latest snapshot
(written by the pinned hipMemcpyAsync side-stream snapshotter).
"""
from __future__ import annotations

import argparse
import os
import sys
import time

import torch

from skypilot_amd.checkpoint.snapshotter import Snapshotter
from skypilot_amd.train.trainer import TrainConfig, Trainer, setup_distributed


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama3-8b")
    ap.add_argument("--steps", type=int, default=100)
    ap.add_argument("--micro-batch", type=int, default=4)
    ap.add_argument("--seq-len", type=int, default=4096)
    ap.add_argument("--lr", type=float, default=3e-4)
    ap.add_argument("--checkpoint-dir", default=os.environ.get(
        "SKY_AMD_CHECKPOINT_DIR"))
    ap.add_argument("--checkpoint-every", type=int, default=20)
    ap.add_argument("--tp", type=int, default=1)
    ap.add_argument("--grad-accum", type=int, default=1)
    ap.add_argument("--warmup-steps", type=int, default=0)
    ap.add_argument("--lr-decay-steps", type=int, default=0)
    ap.add_argument("--device", default=None)
    args = ap.parse_args()

    rank, world, _ = setup_distributed()
    device = args.device or ("cuda" if torch.cuda.is_available() else "cpu")
    cfg = TrainConfig(model=args.model, micro_batch=args.micro_batch,
                      seq_len=args.seq_len, lr=args.lr, device=device,
                      tp=args.tp, grad_accum=args.grad_accum,
                      warmup_steps=args.warmup_steps,
                      lr_decay_steps=args.lr_decay_steps)
    tr = Trainer(cfg)

    snap = None
    start_step = 0
    if args.checkpoint_dir:
        snap = Snapshotter(tr, args.checkpoint_dir)
        resumed = snap.try_resume()
        if resumed is not None:
            start_step = resumed
            if rank == 0:
                print(f"resumed from checkpoint at step {resumed}",
                      flush=True)

    t0 = time.perf_counter()
    for step in range(start_step, args.steps):
        loss = tr.train_step()
        if snap and (step + 1) % args.checkpoint_every == 0:
            snap.save(blocking=False)
        if rank == 0 and (step + 1) % 5 == 0:
            dt = time.perf_counter() - t0
            done = step + 1 - start_step
            tps = tr.tokens_per_step() * done / dt
            print(f"step {step+1}/{args.steps} loss {loss:.4f} "
                  f"tokens/s {tps:,.0f}", flush=True)
    if snap:
        snap.save(blocking=True)
        snap.wait()
    if rank == 0:
        print("training done", flush=True)
    if world > 1:
        torch.distributed.destroy_process_group()
    return 0


if __name__ == "__main__":
    sys.exit(main())
