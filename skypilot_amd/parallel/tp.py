"""Tensor parallelism over RCCL/xGMI for the 70B path.

Megatron-style sharding mapped to the MI355X node: column-parallel
QKV/gate-up GEMMs, row-parallel out/down GEMMs with one all-reduce each
(2 all-reduces per block over xGMI; 7 p2p links x ~153 GB/s, so TP=8
all-reduce of a [B,S,8192] bf16 activation is per-link bound —
SURVEY.md §2.12).  Llama-3-70B at TP=8 gives each rank 8 q-heads /
1 kv-head and a 3584-wide MLP shard.

No reference counterpart (SkyPilot implements no parallelism,
SURVEY.md §2.11); this is the bundled-entrypoint strategy layer.
"""
from __future__ import annotations

import math

import torch
import torch.distributed as dist
import torch.nn as nn
import torch.nn.functional as F

from skypilot_amd import ops
from skypilot_amd.models import llama as L


# Sentinel: a tp=1 build must not touch collectives even when the
# process happens to be in a >1-world process group (e.g. a reference
# model built alongside a TP engine).
_NO_COMM = "no_comm"


def _group_world(group) -> int:
    if group == _NO_COMM or not dist.is_initialized():
        return 1
    return dist.get_world_size(group)


class _AllReduceFwd(torch.autograd.Function):
    """All-reduce activations forward; identity backward (used after
    row-parallel GEMMs)."""

    @staticmethod
    def forward(ctx, x, group):
        if _group_world(group) > 1:
            x = x.contiguous()
            dist.all_reduce(x, group=group)
        return x

    @staticmethod
    def backward(ctx, g):
        return g, None


class _AllReduceBwd(torch.autograd.Function):
    """Identity forward; all-reduce gradient (used before column-parallel
    GEMMs, i.e. on the replicated input)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return x

    @staticmethod
    def backward(ctx, g):
        if _group_world(ctx.group) > 1:
            g = g.contiguous()
            dist.all_reduce(g, group=ctx.group)
        return g, None


class TPAttention(nn.Module):
    def __init__(self, cfg: L.LlamaConfig, tp: int, group):
        super().__init__()
        assert cfg.num_heads % tp == 0 and cfg.num_kv_heads % tp == 0, \
            f"heads {cfg.num_heads}/{cfg.num_kv_heads} not divisible by tp={tp}"
        self.cfg = cfg
        self.group = group
        h, d = cfg.hidden_size, cfg.head_dim
        self.n_q = cfg.num_heads // tp
        self.n_kv = cfg.num_kv_heads // tp
        self.wq = nn.Linear(h, self.n_q * d, bias=False)
        self.wk = nn.Linear(h, self.n_kv * d, bias=False)
        self.wv = nn.Linear(h, self.n_kv * d, bias=False)
        self.wo = nn.Linear(self.n_q * d, h, bias=False)
        self.scale = 1.0 / math.sqrt(d)

    def forward(self, x, cos, sin, positions, infer_ctx=None):
        B, S, _ = x.shape
        d = self.cfg.head_dim
        x = _AllReduceBwd.apply(x, self.group)
        q = self.wq(x).view(B, S, self.n_q, d)
        k = self.wk(x).view(B, S, self.n_kv, d)
        v = self.wv(x).view(B, S, self.n_kv, d)
        q = ops.rope(q.reshape(B * S, self.n_q, d), cos, sin,
                     positions).view(B, S, self.n_q, d)
        k = ops.rope(k.reshape(B * S, self.n_kv, d), cos, sin,
                     positions).view(B, S, self.n_kv, d)
        if infer_ctx is None:
            o = ops.attention(q, k, v, self.scale, causal=True)
        elif infer_ctx.mode == "prefill":
            # batched prefill (same contract as models/llama.py:117)
            slots = (infer_ctx.prefill_slots
                     if infer_ctx.prefill_slots is not None
                     else [infer_ctx.prefill_slot])
            lens = (infer_ctx.prefill_lens
                    if infer_ctx.prefill_lens is not None
                    else [infer_ctx.prefill_len])
            for i, (slot, ln) in enumerate(zip(slots, lens)):
                infer_ctx.cache.write_prefill(self.layer_idx, slot,
                                              k[i:i + 1], v[i:i + 1], ln)
            o = ops.attention(q, k, v, self.scale, causal=True)
        else:
            infer_ctx.cache.write_decode(self.layer_idx, infer_ctx.slots,
                                         infer_ctx.pos, k, v)
            o = ops.attn_decode(
                q.view(B, self.n_q, d), infer_ctx.cache.k[self.layer_idx],
                infer_ctx.cache.v[self.layer_idx], infer_ctx.kv_lens,
                infer_ctx.slot_ids_i32, self.scale).view(B, 1, self.n_q, d)
        out = self.wo(o.reshape(B, S, self.n_q * d))
        return _AllReduceFwd.apply(out, self.group)


class TPMLP(nn.Module):
    def __init__(self, cfg: L.LlamaConfig, tp: int, group):
        super().__init__()
        assert cfg.intermediate_size % tp == 0
        h = cfg.hidden_size
        self.m = cfg.intermediate_size // tp
        self.group = group
        self.w_gate_up = nn.Linear(h, 2 * self.m, bias=False)
        self.w_down = nn.Linear(self.m, h, bias=False)

    def forward(self, x, infer_ctx=None):
        x = _AllReduceBwd.apply(x, self.group)
        gu = self.w_gate_up(x)
        g, u = gu.split(self.m, dim=-1)
        out = self.w_down(F.silu(g) * u)
        return _AllReduceFwd.apply(out, self.group)


class TPBlock(nn.Module):
    def __init__(self, cfg, layer_idx, tp, group):
        super().__init__()
        self.attn = TPAttention(cfg, tp, group)
        self.attn.layer_idx = layer_idx
        self.mlp = TPMLP(cfg, tp, group)
        self.attn_norm = nn.Parameter(torch.ones(cfg.hidden_size))
        self.mlp_norm = nn.Parameter(torch.ones(cfg.hidden_size))
        self.eps = cfg.norm_eps

    def forward(self, x, cos, sin, positions, infer_ctx=None):
        x = x + self.attn(ops.rmsnorm(x, self.attn_norm, self.eps), cos,
                          sin, positions, infer_ctx)
        x = x + self.mlp(ops.rmsnorm(x, self.mlp_norm, self.eps))
        return x


class TPLlama(L.Llama):
    """Llama with tensor-parallel blocks; embed/norm/lm_head replicated
    (lm_head is 2.1 GB for 70B — cheap against 288 GB/GPU)."""

    def __init__(self, cfg: L.LlamaConfig, tp: int, group=None):
        nn.Module.__init__(self)
        if tp == 1:
            group = _NO_COMM
        self.cfg = cfg
        self.tp = tp
        self.group = group
        self.embed = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.blocks = nn.ModuleList(
            TPBlock(cfg, i, tp, group) for i in range(cfg.num_layers))
        self.final_norm = nn.Parameter(torch.ones(cfg.hidden_size))
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        self._rope = None


def shard_from_full(tp_model: TPLlama, full: L.Llama, rank: int, tp: int):
    """Populate a TP shard from a fully-materialized model (used for
    small-model tests and seed-consistent init)."""
    sd = tp_model.state_dict()
    fsd = full.state_dict()
    cfg = full.cfg
    d = cfg.head_dim
    for name, param in fsd.items():
        if name not in sd:
            continue
        if ".wq.weight" in name or ".wk.weight" in name or \
                ".wv.weight" in name:
            nh = (cfg.num_heads if ".wq." in name else cfg.num_kv_heads)
            per = nh // tp * d
            sd[name].copy_(param[rank * per:(rank + 1) * per])
        elif ".wo.weight" in name:
            per = cfg.num_heads // tp * d
            sd[name].copy_(param[:, rank * per:(rank + 1) * per])
        elif ".w_gate_up.weight" in name:
            m = cfg.intermediate_size
            per = m // tp
            gate = param[:m][rank * per:(rank + 1) * per]
            up = param[m:][rank * per:(rank + 1) * per]
            sd[name].copy_(torch.cat([gate, up], dim=0))
        elif ".w_down.weight" in name:
            per = cfg.intermediate_size // tp
            sd[name].copy_(param[:, rank * per:(rank + 1) * per])
        else:
            sd[name].copy_(param)
    tp_model.load_state_dict(sd)


def build_tp_model(name: str, tp: int, rank: int, device="cpu",
                   dtype=torch.bfloat16, group=None, seed: int = 0
                   ) -> TPLlama:
    """Build a TP shard with init consistent across ranks: every rank
    draws the same full weights (same seed) and keeps its shard."""
    cfg = L.CONFIGS[name]
    torch.manual_seed(seed)
    model = TPLlama(cfg, tp, group)
    if device != "cpu":
        model = model.to_empty(device=device)
    # Seed-consistent sharded init without materializing the full model:
    # draw each full weight on the target device, slice, free.
    gen_dev = device if device != "cpu" else "cpu"
    g = torch.Generator(device=gen_dev).manual_seed(seed)
    d = cfg.head_dim

    def draw(shape):
        return torch.empty(shape, device=gen_dev).normal_(
            0.0, 0.02, generator=g)

    with torch.no_grad():
        model.embed.weight.copy_(draw(model.embed.weight.shape))
        for blk in model.blocks:
            a, m = blk.attn, blk.mlp
            h = cfg.hidden_size
            full_q = draw((cfg.num_heads * d, h))
            per = cfg.num_heads // tp * d
            a.wq.weight.copy_(full_q[rank * per:(rank + 1) * per])
            full_k = draw((cfg.num_kv_heads * d, h))
            perk = cfg.num_kv_heads // tp * d
            a.wk.weight.copy_(full_k[rank * perk:(rank + 1) * perk])
            full_v = draw((cfg.num_kv_heads * d, h))
            a.wv.weight.copy_(full_v[rank * perk:(rank + 1) * perk])
            full_o = draw((h, cfg.num_heads * d))
            a.wo.weight.copy_(full_o[:, rank * per:(rank + 1) * per])
            mi = cfg.intermediate_size
            perm = mi // tp
            full_gu = draw((2 * mi, h))
            gate = full_gu[:mi][rank * perm:(rank + 1) * perm]
            up = full_gu[mi:][rank * perm:(rank + 1) * perm]
            m.w_gate_up.weight.copy_(torch.cat([gate, up], dim=0))
            full_dn = draw((h, mi))
            m.w_down.weight.copy_(full_dn[:, rank * perm:(rank + 1) * perm])
            blk.attn_norm.fill_(1.0)
            blk.mlp_norm.fill_(1.0)
        model.final_norm.fill_(1.0)
        model.lm_head.weight.copy_(draw(model.lm_head.weight.shape))
    model = model.to(dtype)
    # Per-shard view of the config for inference plumbing (the KV cache
    # holds this rank's kv heads only).
    import dataclasses
    model.cfg_shard = dataclasses.replace(
        cfg, num_heads=cfg.num_heads // tp,
        num_kv_heads=cfg.num_kv_heads // tp)
    return model
