"""KV cache for the bundled inference engine — contiguous slot-based
cache sized against 288 GB HBM3E per MI355X (no reference counterpart;
SkyPilot delegates serving to user images)."""
from __future__ import annotations

from typing import List

import torch

from skypilot_amd.models.llama import LlamaConfig


class KVCache:
    """One [max_batch, max_seq, Hkv, D] K and V tensor per layer."""

    def __init__(self, cfg: LlamaConfig, max_batch: int, max_seq: int,
                 device, dtype=torch.bfloat16):
        self.cfg = cfg
        self.max_batch = max_batch
        self.max_seq = max_seq
        self.k: List[torch.Tensor] = []
        self.v: List[torch.Tensor] = []
        for _ in range(cfg.num_layers):
            shape = (max_batch, max_seq, cfg.num_kv_heads, cfg.head_dim)
            self.k.append(torch.zeros(shape, device=device, dtype=dtype))
            self.v.append(torch.zeros(shape, device=device, dtype=dtype))
        # host-side slot lengths (engine keeps the authoritative copy)
        self.lens = [0] * max_batch

    @staticmethod
    def bytes_needed(cfg: LlamaConfig, max_batch: int, max_seq: int) -> int:
        return (2 * cfg.num_layers * max_batch * max_seq *
                cfg.num_kv_heads * cfg.head_dim * 2)

    @classmethod
    def sized_for_memory(cls, cfg: LlamaConfig, max_seq: int,
                         budget_bytes: int, device,
                         cap: int = 256) -> "KVCache":
        """Pick max_batch to fill the HBM budget (288 GB minus weights)."""
        per_slot = cls.bytes_needed(cfg, 1, max_seq)
        max_batch = max(1, min(cap, budget_bytes // per_slot))
        return cls(cfg, max_batch, max_seq, device)

    # ---- writes -----------------------------------------------------------
    def write_prefill(self, layer: int, slot: int, k: torch.Tensor,
                      v: torch.Tensor, length: int) -> None:
        """k/v: [1, S_padded, Hkv, D]; store the first `length` rows."""
        self.k[layer][slot, :length] = k[0, :length]
        self.v[layer][slot, :length] = v[0, :length]

    def write_decode(self, layer: int, slots: torch.Tensor,
                     pos: torch.Tensor, k: torch.Tensor,
                     v: torch.Tensor) -> None:
        """k/v: [n, 1, Hkv, D] new tokens; slots/pos: [n] int64."""
        self.k[layer][slots, pos] = k[:, 0]
        self.v[layer][slots, pos] = v[:, 0]

    def free(self, slot: int) -> None:
        self.lens[slot] = 0
