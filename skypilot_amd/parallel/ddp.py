"""Bucketed data parallelism over RCCL/xGMI — MI355X-native DDP.

Not a wrapper around ``torch.nn.parallel.DistributedDataParallel``: as
each parameter's gradient is accumulated, a post-accumulate hook moves it
into a flat per-bucket bf16 buffer and the bucket all-reduces
asynchronously the moment its last grad lands — communication overlaps
the rest of backward.  The fused AdamW consumes the bucket views
directly (``p._sky_grad`` attribute), so gradients are never re-gathered.

Bucket size defaults to 64 MiB: xGMI is point-to-point (7 links x
~153 GB/s per MI355X), so ring all-reduce is per-link bound and wants
fewer, larger transfers than an NVSwitch fabric.
The all-reduce SUMs; averaging is folded into the fused AdamW's
grad_scale so no extra pass over the gradients is needed.
"""
from __future__ import annotations

import torch
import torch.distributed as dist

DEFAULT_BUCKET_BYTES = 64 << 20


def _initialized() -> bool:
    return dist.is_available() and dist.is_initialized()


class GradBucket:
    def __init__(self, params, device, dtype):
        self.params = params
        self.numel = sum(p.numel() for p in params)
        self.flat = torch.zeros(self.numel, device=device, dtype=dtype)
        self.views = {}
        off = 0
        for p in params:
            self.views[p] = self.flat[off:off + p.numel()].view_as(p)
            off += p.numel()
        self.pending = 0
        self.work = None


class BucketedDDP:
    """Gradient-summing DDP with comm/compute overlap.

    Usage per step:
        ddp.mark_step_start(); loss.backward(); ddp.finish()
    After ``finish`` each param's gradient (summed over ranks) is in
    ``p._sky_grad``; divide-by-world is folded into the optimizer via
    ``ddp.grad_scale``.
    """

    def __init__(self, model: torch.nn.Module,
                 bucket_bytes: int = DEFAULT_BUCKET_BYTES,
                 process_group=None):
        self.model = model
        self.group = process_group
        self.world = dist.get_world_size(process_group) if _initialized() else 1
        params = [p for p in model.parameters() if p.requires_grad]
        # Reverse registration order approximates backward completion order,
        # so early buckets fill (and start reducing) first.
        params = list(reversed(params))
        self.buckets: list[GradBucket] = []
        cur, cur_bytes = [], 0
        for p in params:
            cur.append(p)
            cur_bytes += p.numel() * p.element_size()
            if cur_bytes >= bucket_bytes:
                self.buckets.append(GradBucket(cur, p.device, p.dtype))
                cur, cur_bytes = [], 0
        if cur:
            self.buckets.append(GradBucket(cur, cur[0].device, cur[0].dtype))
        self._p2b = {}
        for b in self.buckets:
            for p in b.params:
                self._p2b[p] = b
                p._sky_grad = b.views[p]
                p.register_post_accumulate_grad_hook(self._hook)
        self._accumulating = False
        if self.world > 1:
            for p in model.parameters():
                dist.broadcast(p.data, src=self._rank0(), group=self.group)

    def _rank0(self) -> int:
        return 0 if self.group is None else dist.get_global_rank(self.group, 0)

    def _hook(self, p):
        b = self._p2b[p]
        # Always accumulate: zero_grad() clears the flat buffers at the
        # start of each optimizer step, so add_ is correct for both the
        # single-micro-step case and every micro-step of grad accumulation.
        # (copy_ on the final micro-step would overwrite the accumulated
        # sum from earlier micro-steps.)
        b.views[p].add_(p.grad)
        p.grad = None  # free eagerly; flat buffer is the only grad storage
        b.pending -= 1
        if b.pending == 0 and self.world > 1 and not self._accumulating:
            b.work = dist.all_reduce(b.flat, op=dist.ReduceOp.SUM,
                                     group=self.group, async_op=True)

    def mark_step_start(self, accumulating: bool = False):
        """Call before each backward. ``accumulating=True`` defers the
        all-reduce (gradient accumulation micro-steps)."""
        self._accumulating = accumulating
        for b in self.buckets:
            b.pending = len(b.params)
            b.work = None

    def zero_grad(self):
        for b in self.buckets:
            b.flat.zero_()

    def finish(self):
        """Wait for outstanding bucket reductions (call after backward)."""
        for b in self.buckets:
            if b.work is not None:
                b.work.wait()
            elif self.world > 1 and not self._accumulating:
                dist.all_reduce(b.flat, op=dist.ReduceOp.SUM, group=self.group)

    @property
    def grad_scale(self) -> float:
        return 1.0 / self.world
