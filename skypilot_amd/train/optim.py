"""Fused AdamW with fp32 master weights for bf16 training on MI355X.

The update runs entirely in the hand-written CDNA4 kernel
(`ops/csrc/adamw.hip`) — ONE launch per step via the multi-tensor
chunk kernel (per-tensor launches measured 41 ms/step vs the ~16 ms
state-traffic bound for 8B params); device-side metadata (pointers,
sizes, chunk map) is built once and reused while pointers are
stable.  Master
weights and both moments are fp32; model params stay bf16 (refreshed
from the master copy each step inside the same kernel).  Gradient
averaging for DDP is folded in via ``grad_scale``.

On CPU (tests) a reference fp32 implementation with identical math runs
instead.
"""
from __future__ import annotations

import torch

from skypilot_amd import ops


class FusedAdamW:
    def __init__(self, params, lr=3e-4, betas=(0.9, 0.95), eps=1e-8,
                 weight_decay=0.1, decay_2d_only=True):
        self.params = [p for p in params if p.requires_grad]
        if not self.params:
            raise ValueError("no trainable params")
        self.lr, self.betas, self.eps, self.wd = lr, betas, eps, weight_decay
        self.step_count = 0
        self.master = [p.detach().float().clone() for p in self.params]
        self.exp_avg = [torch.zeros_like(m) for m in self.master]
        self.exp_avg_sq = [torch.zeros_like(m) for m in self.master]
        # Llama convention: decay weights of matrices, not norms/embedding
        # scales (1-D tensors).
        self.decay_mask = [
            (p.dim() >= 2) if decay_2d_only else True for p in self.params
        ]
        self._mt = None  # cached device metadata for the one-launch path

    def _grads(self):
        gs = []
        for p in self.params:
            g = getattr(p, "_sky_grad", None)
            if g is None:
                g = p.grad
            if g is None:
                raise RuntimeError("param has no gradient at optimizer step")
            gs.append(g)
        return gs

    _CHUNK = 65536

    def _mt_meta(self, grads):
        key = tuple(g.data_ptr() for g in grads)
        if self._mt is not None and self._mt["key"] == key:
            return self._mt
        dev = self.params[0].device
        T = len(self.params)
        ptrs = torch.empty(T, 5, dtype=torch.int64)
        sizes = torch.empty(T, dtype=torch.int64)
        wd = torch.empty(T, dtype=torch.float32)
        ct, cs = [], []
        aligned = True
        for i, (p, mp, g, m, v, dm) in enumerate(
                zip(self.params, self.master, grads, self.exp_avg,
                    self.exp_avg_sq, self.decay_mask)):
            ptrs[i] = torch.tensor([p.data_ptr(), mp.data_ptr(),
                                    g.data_ptr(), m.data_ptr(),
                                    v.data_ptr()], dtype=torch.int64)
            n = p.numel()
            sizes[i] = n
            wd[i] = self.wd if dm else 0.0
            aligned &= (g.data_ptr() % 8 == 0 and p.data_ptr() % 8 == 0)
            for off in range(0, n, self._CHUNK):
                ct.append(i)
                cs.append(off)
        if not aligned:
            return None  # odd bucket view: per-tensor fallback
        self._mt = {
            "key": key,
            "ptrs": ptrs.to(dev),
            "wd": wd.to(dev),
            "sizes": sizes.to(dev),
            "ct": torch.tensor(ct, dtype=torch.int32, device=dev),
            "cs": torch.tensor(cs, dtype=torch.int64, device=dev),
        }
        return self._mt

    @torch.no_grad()
    def step(self, grad_scale: float = 1.0):
        self.step_count += 1
        grads = self._grads()
        if self.params[0].is_cuda:
            C = ops.native()
            meta = self._mt_meta(grads)
            if meta is not None:
                C.adamw_step_mt(meta["ptrs"], meta["wd"], meta["sizes"],
                                meta["ct"], meta["cs"], self.lr,
                                self.betas[0], self.betas[1], self.eps,
                                self.step_count, grad_scale)
                return
            C.adamw_step([p.data for p in self.params], self.master, grads,
                         self.exp_avg, self.exp_avg_sq, self.lr,
                         self.betas[0], self.betas[1], self.eps, self.wd,
                         self.step_count, grad_scale, self.decay_mask)
            return
        # CPU reference path (same math, fp32).
        b1, b2 = self.betas
        bc1 = 1 - b1 ** self.step_count
        bc2 = 1 - b2 ** self.step_count
        for p, mp, g, m, v, dm in zip(self.params, self.master, grads,
                                      self.exp_avg, self.exp_avg_sq,
                                      self.decay_mask):
            gf = g.float() * grad_scale
            m.mul_(b1).add_(gf, alpha=1 - b1)
            v.mul_(b2).addcmul_(gf, gf, value=1 - b2)
            wd = self.wd if dm else 0.0
            update = (m / bc1) / ((v / bc2).sqrt() + self.eps) + wd * mp
            mp.add_(update, alpha=-self.lr)
            p.data.copy_(mp.to(p.dtype))

    def zero_grad(self):
        for p in self.params:
            p.grad = None

    def state_tensors(self):
        """Everything needed for checkpoint/resume (master + moments)."""
        return {
            "master": self.master,
            "exp_avg": self.exp_avg,
            "exp_avg_sq": self.exp_avg_sq,
            "step_count": self.step_count,
        }

    def load_state_tensors(self, state):
        for dst, src in zip(self.master, state["master"]):
            dst.copy_(src)
        for dst, src in zip(self.exp_avg, state["exp_avg"]):
            dst.copy_(src)
        for dst, src in zip(self.exp_avg_sq, state["exp_avg_sq"]):
            dst.copy_(src)
        self.step_count = state["step_count"]
        for p, mp in zip(self.params, self.master):
            p.data.copy_(mp.to(p.dtype))
