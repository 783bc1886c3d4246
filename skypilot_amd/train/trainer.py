"""Bundled Llama trainer — the flagship training entrypoint.

This is what `bench.py` and the `sky launch` training task YAML run: one
process per GPU, bf16 Llama on the fused CDNA4 kernels, bucketed
all-reduce DDP over RCCL/xGMI, fused AdamW with fp32 master state.
Synthetic data (there is no network on the pool): random tokens of the
benchmark shape, random-init weights.
"""
from __future__ import annotations

import os
import time
from dataclasses import dataclass

import torch
import torch.distributed as dist

from skypilot_amd.models.llama import build_model
from skypilot_amd.parallel.ddp import BucketedDDP
from skypilot_amd.train.optim import FusedAdamW


@dataclass
class TrainConfig:
    model: str = "llama3-8b"
    micro_batch: int = 4
    seq_len: int = 4096
    lr: float = 3e-4
    weight_decay: float = 0.1
    seed: int = 1234
    bucket_mb: int = 64
    device: str = "cuda"
    tp: int = 1  # tensor-parallel degree (divides world size)
    grad_accum: int = 1          # micro-steps per optimizer step
    warmup_steps: int = 0        # linear LR warmup
    lr_decay_steps: int = 0      # cosine decay horizon (0 = constant)
    min_lr_ratio: float = 0.1


def dist_env():
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    return rank, world, local_rank


def setup_distributed(backend: str | None = None):
    rank, world, local_rank = dist_env()
    if world > 1 and not dist.is_initialized():
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29500")
        dist.init_process_group(backend, rank=rank, world_size=world)
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank)
    return rank, world, local_rank


class Trainer:
    def __init__(self, cfg: TrainConfig):
        self.cfg = cfg
        self.rank, self.world, self.local_rank = dist_env()
        dev = cfg.device
        if dev == "cuda":
            dev = f"cuda:{self.local_rank}"
        self.device = torch.device(dev)
        self.tp = max(1, cfg.tp)
        self.tp_group = self.dp_group = None
        if self.tp > 1:
            assert self.world % self.tp == 0, "tp must divide world size"
            # Consecutive ranks share a TP group (same xGMI neighborhood);
            # DP groups stride across TP groups.
            for i in range(self.world // self.tp):
                g = dist.new_group(list(range(i * self.tp,
                                              (i + 1) * self.tp)))
                if self.rank // self.tp == i:
                    self.tp_group = g
            for j in range(self.tp):
                g = dist.new_group(list(range(j, self.world, self.tp)))
                if self.rank % self.tp == j:
                    self.dp_group = g
            from skypilot_amd.parallel.tp import build_tp_model
            self.model = build_tp_model(
                cfg.model, self.tp, self.rank % self.tp,
                device=str(self.device), dtype=torch.bfloat16,
                group=self.tp_group, seed=cfg.seed)
        else:
            self.model = build_model(cfg.model, device=str(self.device),
                                     dtype=torch.bfloat16, seed=cfg.seed)
        self.ddp = BucketedDDP(self.model,
                               bucket_bytes=cfg.bucket_mb << 20,
                               process_group=self.dp_group)
        self.opt = FusedAdamW(self.model.parameters(), lr=cfg.lr,
                              weight_decay=cfg.weight_decay)
        self.step_count = 0
        # Data-parallel peers draw different data; TP peers the same.
        dp_rank = self.rank // self.tp if self.tp > 1 else self.rank
        g = torch.Generator(device="cpu").manual_seed(cfg.seed + dp_rank)
        self._gen = g

    def synthetic_batch(self):
        c = self.cfg
        vocab = self.model.cfg.vocab_size
        tok = torch.randint(0, vocab, (c.micro_batch, c.seq_len + 1),
                            generator=self._gen)
        tok = tok.to(self.device, non_blocking=True)
        return tok[:, :-1].contiguous(), tok[:, 1:].contiguous()

    def current_lr(self) -> float:
        """Linear warmup + cosine decay (step_count is 0-based here)."""
        import math as _math
        c = self.cfg
        step = self.step_count
        lr = c.lr
        if c.warmup_steps and step < c.warmup_steps:
            return lr * (step + 1) / c.warmup_steps
        if c.lr_decay_steps:
            t = min(1.0, (step - c.warmup_steps) /
                    max(1, c.lr_decay_steps - c.warmup_steps))
            floor = c.lr * c.min_lr_ratio
            return floor + 0.5 * (lr - floor) * (1 + _math.cos(_math.pi * t))
        return lr

    def train_step(self, batch=None) -> float:
        """One optimizer step = cfg.grad_accum micro-steps (gradients
        accumulate in the DDP buckets; the all-reduce fires on the final
        micro-step only)."""
        c = self.cfg
        self.ddp.zero_grad()
        loss = 0.0
        for micro in range(c.grad_accum):
            b = batch if batch is not None else self.synthetic_batch()
            tokens, targets = b
            self.ddp.mark_step_start(
                accumulating=(micro < c.grad_accum - 1))
            l = self.model.loss(tokens, targets)
            (l / c.grad_accum if c.grad_accum > 1 else l).backward()
            self.ddp.finish()
            loss = float(l.detach())
        self.opt.lr = self.current_lr()
        # An in-flight checkpoint snapshot reads optimizer state on a side
        # stream; order this step's in-place AdamW after those copies.
        ev = getattr(self, "_snapshot_event", None)
        if ev is not None and self.device.type == "cuda":
            torch.cuda.current_stream().wait_event(ev)
            self._snapshot_event = None
        self.opt.step(grad_scale=self.ddp.grad_scale)
        self.step_count += 1
        return loss

    def tokens_per_step(self) -> int:
        dp = self.world // self.tp
        return self.cfg.micro_batch * self.cfg.seq_len * dp * \
            self.cfg.grad_accum


def run_training(cfg: TrainConfig, steps: int, warmup: int = 2,
                 log_every: int = 1):
    """Used by the bundled train task; prints throughput per step."""
    rank, world, _ = setup_distributed()
    tr = Trainer(cfg)
    for i in range(warmup):
        tr.train_step()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(steps):
        loss = tr.train_step()
        if rank == 0 and (i + 1) % log_every == 0:
            dt = time.perf_counter() - t0
            tps = tr.tokens_per_step() * (i + 1) / dt
            print(f"step {i+1}/{steps} loss {loss:.4f} "
                  f"tokens/s {tps:,.0f}", flush=True)
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    return tr
