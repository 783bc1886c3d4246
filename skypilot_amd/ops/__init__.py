"""skypilot_amd.ops — MI355X-native fused ops for the bundled trainer.

Each op routes to the hand-written CDNA4 HIP kernels (``_C.so``, built
in-tree by :mod:`skypilot_amd.ops.build`) whenever the tensors live on a
GPU.  On a GPU box a missing/unloadable extension is a **hard error** —
there is deliberately no silent eager fallback (the driver verifies the
native code actually loaded).  On CPU (the no-GPU CI container) the ops
fall back to plain fp32 PyTorch reference implementations, which are also
what the GPU numerics tests compare the kernels against.

Reference parity note: SkyPilot itself ships no kernels (SURVEY.md
§2.11); these ops are the MI355X-native additions the north star
requires for the bundled Llama train/serve entrypoints.
"""
from __future__ import annotations

import importlib.util
import math
import os
from pathlib import Path

import torch

_C = None
_LOAD_ERR: Exception | None = None


def _load():
    global _C, _LOAD_ERR
    if _C is not None:
        return _C
    so = Path(__file__).resolve().parent / "_C.so"
    try:
        if not so.exists():
            raise ImportError(
                f"native extension {so} not built; run "
                "`python -m skypilot_amd.ops.build`")
        spec = importlib.util.spec_from_file_location("skypilot_amd.ops._C",
                                                      so)
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)
        _C = mod
    except Exception as e:  # noqa: BLE001
        _LOAD_ERR = e
        raise
    return _C


def native():
    """The native module; raises loudly if unavailable."""
    return _load()


def native_available() -> bool:
    try:
        _load()
        return True
    except Exception:  # noqa: BLE001
        return False


def _require_native(opname: str):
    try:
        return _load()
    except Exception as e:  # noqa: BLE001
        raise RuntimeError(
            f"skypilot_amd.ops.{opname}: tensor is on GPU but the native "
            f"CDNA4 extension failed to load ({e}). Refusing to fall back "
            "to eager PyTorch on the GPU path.") from e


# ===========================================================================
# Reference (CPU / verification) implementations — plain fp32 PyTorch.
# ===========================================================================
def rmsnorm_ref(x: torch.Tensor, w: torch.Tensor, eps: float) -> torch.Tensor:
    xf = x.float()
    inv = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (xf * inv * w.float()).to(x.dtype)


def rope_ref(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
             positions: torch.Tensor, backward: bool = False) -> torch.Tensor:
    # x: [..., T, H, D]; cos/sin: [S, D/2]; positions: [T]
    D = x.shape[-1]
    xf = x.float()
    c = cos[positions]  # [T, D/2]
    s = sin[positions] * (-1.0 if backward else 1.0)
    c = c.unsqueeze(-2)  # [T, 1, D/2]
    s = s.unsqueeze(-2)
    x1, x2 = xf[..., : D // 2], xf[..., D // 2:]
    out = torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], dim=-1)
    return out.to(x.dtype)


def attention_ref(q, k, v, scale: float, causal: bool = True):
    # q: [B,S,Hq,D], k/v: [B,S,Hkv,D] -> [B,S,Hq,D]
    B, S, Hq, D = q.shape
    Hkv = k.shape[2]
    rep = Hq // Hkv
    qf = q.float().permute(0, 2, 1, 3)  # B,Hq,S,D
    kf = k.float().permute(0, 2, 1, 3).repeat_interleave(rep, dim=1)
    vf = v.float().permute(0, 2, 1, 3).repeat_interleave(rep, dim=1)
    att = torch.matmul(qf, kf.transpose(-1, -2)) * scale
    if causal:
        mask = torch.triu(torch.ones(S, S, dtype=torch.bool,
                                     device=q.device), 1)
        att = att.masked_fill(mask, float("-inf"))
    p = att.softmax(-1)
    out = torch.matmul(p, vf)
    return out.permute(0, 2, 1, 3).to(q.dtype)


def cross_entropy_ref(logits: torch.Tensor, targets: torch.Tensor,
                      ignore_index: int = -100) -> torch.Tensor:
    return torch.nn.functional.cross_entropy(
        logits.float(), targets.long(), ignore_index=ignore_index,
        reduction="mean")


# ===========================================================================
# Autograd wrappers.
# ===========================================================================
class _RMSNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, eps):
        if x.is_cuda:
            C = _require_native("rmsnorm")
            inv_rms = torch.empty(x.numel() // x.shape[-1],
                                  device=x.device, dtype=torch.float32)
            y = C.rmsnorm_fwd(x.contiguous(), w.contiguous(), inv_rms, eps)
            ctx.save_for_backward(x, w, inv_rms)
            ctx.native = True
            return y
        ctx.save_for_backward(x, w)
        ctx.eps = eps
        ctx.native = False
        return rmsnorm_ref(x, w, eps)

    @staticmethod
    def backward(ctx, dy):
        if ctx.native:
            x, w, inv_rms = ctx.saved_tensors
            C = native()
            dx, dw = C.rmsnorm_bwd(x.contiguous(), w.contiguous(),
                                   dy.contiguous(), inv_rms)
            return dx, dw.to(w.dtype), None
        x, w = ctx.saved_tensors
        xf, wf, dyf = x.float(), w.float(), dy.float()
        H = x.shape[-1]
        inv = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + ctx.eps)
        dot = (dyf * wf * xf).sum(-1, keepdim=True)
        dx = inv * (dyf * wf - xf * inv * inv * dot / H)
        dw = (dyf * xf * inv).reshape(-1, H).sum(0)
        return dx.to(x.dtype), dw.to(w.dtype), None


def rmsnorm(x: torch.Tensor, w: torch.Tensor, eps: float = 1e-5):
    return _RMSNormFn.apply(x, w, eps)


class _RoPEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, cos, sin, positions):
        ctx.save_for_backward(cos, sin, positions)
        if x.is_cuda:
            C = _require_native("rope")
            return C.rope(x.contiguous(), cos, sin, positions, False)
        return rope_ref(x, cos, sin, positions)

    @staticmethod
    def backward(ctx, dy):
        cos, sin, positions = ctx.saved_tensors
        if dy.is_cuda:
            C = native()
            return C.rope(dy.contiguous(), cos, sin, positions,
                          True), None, None, None
        return rope_ref(dy, cos, sin, positions, backward=True), \
            None, None, None


def rope(x, cos, sin, positions):
    return _RoPEFn.apply(x, cos, sin, positions)


class _AttentionFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, scale, causal):
        ctx.scale, ctx.causal = scale, causal
        if q.is_cuda:
            C = _require_native("attention")
            O, lse = C.attn_fwd(q.contiguous(), k.contiguous(),
                                v.contiguous(), scale, causal)
            ctx.save_for_backward(q, k, v, O, lse)
            ctx.native = True
            return O
        ctx.save_for_backward(q, k, v)
        ctx.native = False
        return attention_ref(q, k, v, scale, causal)

    @staticmethod
    def backward(ctx, dO):
        if ctx.native:
            q, k, v, O, lse = ctx.saved_tensors
            C = native()
            dq, dk, dv = C.attn_bwd(q, k, v, O, dO.contiguous(), lse,
                                    ctx.scale, ctx.causal)
            return dq, dk, dv, None, None
        q, k, v = ctx.saved_tensors
        with torch.enable_grad():
            qd = q.detach().requires_grad_()
            kd = k.detach().requires_grad_()
            vd = v.detach().requires_grad_()
            out = attention_ref(qd, kd, vd, ctx.scale, ctx.causal)
            gq, gk, gv = torch.autograd.grad(out, (qd, kd, vd), dO)
        return gq, gk, gv, None, None


def attention(q, k, v, scale=None, causal=True):
    """Flash attention, layout [B, S, H, D] with GQA (Hq multiple of Hkv)."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    return _AttentionFn.apply(q, k, v, scale, causal)


class _CrossEntropyFn(torch.autograd.Function):
    """Fused CE: forward computes mean loss AND writes d_logits into the
    logits buffer (scaled by 1/n_valid); backward scales by grad_output."""

    @staticmethod
    def forward(ctx, logits, targets, ignore_index):
        if logits.is_cuda:
            C = _require_native("cross_entropy")
            n_valid = int((targets != ignore_index).sum().item())
            n_valid = max(n_valid, 1)
            logits = logits.contiguous()
            loss = C.cross_entropy_fused(logits, targets.int(),
                                         1.0 / n_valid, ignore_index)
            ctx.save_for_backward(logits)  # now holds d_logits
            ctx.native = True
            return loss.sum() / n_valid
        ctx.native = False
        ctx.save_for_backward(logits, targets)
        ctx.ignore_index = ignore_index
        return cross_entropy_ref(logits, targets, ignore_index)

    @staticmethod
    def backward(ctx, g):
        if ctx.native:
            (dlogits,) = ctx.saved_tensors
            if not (isinstance(g, torch.Tensor) and g.numel() == 1
                    and float(g) == 1.0):
                dlogits = dlogits * g
            return dlogits, None, None
        logits, targets = ctx.saved_tensors
        with torch.enable_grad():
            ld = logits.detach().requires_grad_()
            loss = cross_entropy_ref(ld, targets, ctx.ignore_index)
            (gl,) = torch.autograd.grad(loss, (ld,), g)
        return gl, None, None


def fused_cross_entropy(logits, targets, ignore_index: int = -100):
    """NB: on GPU the logits buffer is overwritten with its gradient."""
    return _CrossEntropyFn.apply(logits, targets, ignore_index)


class _SwiGLUFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gu):
        ctx.save_for_backward(gu)
        if gu.is_cuda:
            C = _require_native("swiglu")
            return C.swiglu_fwd(gu.contiguous())
        M = gu.shape[-1] // 2
        g, u = gu.float().split(M, dim=-1)
        return (torch.nn.functional.silu(g) * u).to(gu.dtype)

    @staticmethod
    def backward(ctx, dy):
        (gu,) = ctx.saved_tensors
        if gu.is_cuda:
            C = native()
            return C.swiglu_bwd(gu, dy.contiguous())
        M = gu.shape[-1] // 2
        g, u = gu.float().split(M, dim=-1)
        dyf = dy.float()
        sg = torch.sigmoid(g)
        silu = g * sg
        dg = dyf * u * (sg + silu * (1 - sg))
        du = dyf * silu
        return torch.cat([dg, du], dim=-1).to(gu.dtype)


def swiglu(gu):
    """Fused silu(gate) * up over a fused [.., 2M] gate|up tensor."""
    return _SwiGLUFn.apply(gu)


def attn_decode_ref(q, kc, vc, kv_lens, slot_ids, scale):
    """q: [B, Hq, D]; kc/vc: [slots, S_max, Hkv, D]."""
    B, Hq, D = q.shape
    Hkv = kc.shape[2]
    rep = Hq // Hkv
    out = torch.empty_like(q)
    for i in range(B):
        slot = int(slot_ids[i])
        L = int(kv_lens[i])
        k = kc[slot, :L].float()  # [L, Hkv, D]
        v = vc[slot, :L].float()
        kf = k.repeat_interleave(rep, dim=1)  # [L, Hq, D]
        vf = v.repeat_interleave(rep, dim=1)
        att = torch.einsum("hd,lhd->hl", q[i].float(), kf) * scale
        p = att.softmax(-1)
        out[i] = torch.einsum("hl,lhd->hd", p, vf).to(q.dtype)
    return out


def attn_decode(q, kc, vc, kv_lens, slot_ids, scale):
    """Decode attention over the KV cache (no autograd)."""
    if q.is_cuda:
        C = _require_native("attn_decode")
        return C.attn_decode(q.contiguous(), kc, vc, kv_lens.int(),
                             slot_ids.int(), scale)
    return attn_decode_ref(q, kc, vc, kv_lens, slot_ids, scale)


# ---- fp8 decode weights (serving) -------------------------
# Per-output-row e4m3fn quantization: W_row ~= scale_r * fp8(W_row/scale_r).
# Registered weights reroute decode_linear through the fp8 GEMV (half the
# streamed bytes on the weight-bandwidth-bound decode step).
_FP8_WEIGHTS: dict = {}


def quantize_fp8_rowwise(w: torch.Tensor):
    """bf16 [O, I] -> (uint8 e4m3fn [O, I], fp32 scale [O])."""
    wf = w.float()
    scale = wf.abs().amax(dim=1).clamp(min=1e-8) / 448.0
    q = (wf / scale[:, None]).clamp(-448.0, 448.0)
    q8 = q.to(torch.float8_e4m3fn).view(torch.uint8)
    return q8.contiguous(), scale.contiguous()


def register_fp8_weight(w: torch.Tensor) -> None:
    """Quantize and register a weight for fp8 decode GEMV routing."""
    q8, scale = quantize_fp8_rowwise(w)
    _FP8_WEIGHTS[w.data_ptr()] = (q8, scale)


def clear_fp8_weights() -> None:
    _FP8_WEIGHTS.clear()


def decode_linear(x: torch.Tensor, weight: torch.Tensor) -> torch.Tensor:
    """F.linear for the decode step: y = x @ W^T with x [.., n, in],
    n <= 8.  On GPU this runs the weight-bandwidth-bound skinny GEMV
    (skinny_gemm.hip, ~3x hipBLASLt on these shapes); otherwise (CPU,
    n > 8, odd inner dim) it falls back to F.linear.  No autograd —
    inference only."""
    lead = x.shape[:-1]
    n = 1
    for d in lead:
        n *= d
    i, o = x.shape[-1], weight.shape[0]
    wbytes = 2 * i * o
    # Measured routing (bench_gemv.py on MI355X, docs/BENCHMARKS.md):
    # n=1 always wins (W streams at 6.5-7.3 TB/s; hipBLASLt floors at
    # ~19 us/GEMM); n>=2 wins on small-W shapes (launch-bound for
    # hipBLASLt) and, via the R=4 multi-row variant, on vocab-sized
    # projections (O >= 100k) up to n<=4.
    use_native = (n == 1 or wbytes <= 34_000_000
                  or (o >= 100_000 and n <= 4))
    # fp8 pays only while the step is weight-bandwidth-bound: at n >= 3
    # the bf16 multirow GEMV amortizes the weight stream and fp8's
    # unpack VALU cost dominates (measured: @8 streams 1521 -> 1069
    # tok/s with fp8 everywhere).
    if (x.is_cuda and x.dtype == torch.bfloat16 and 1 <= n <= 2
            and i % 1024 == 0 and _FP8_WEIGHTS):
        ent = _FP8_WEIGHTS.get(weight.data_ptr())
        # data_ptr can be recycled after a free; require matching shape
        if ent is not None and tuple(ent[0].shape) != tuple(weight.shape):
            ent = None
        if ent is not None:
            C = _require_native("skinny_gemm_fp8")
            y = C.skinny_gemm_fp8(x.reshape(n, i).contiguous(), ent[0],
                                  ent[1])
            return y.view(*lead, o)
    if (use_native and x.is_cuda and x.dtype == torch.bfloat16
            and 1 <= n <= 8 and i % 512 == 0):
        C = _require_native("skinny_gemm")
        y = C.skinny_gemm(x.reshape(n, i).contiguous(), weight)
        return y.view(*lead, o)
    return torch.nn.functional.linear(x, weight)


def decode_norm_linear(x, res, norm_w, eps, weight):
    """Fused rmsnorm(x [+ res]) * norm_w -> fp8 GEMV in ONE kernel
    (skinny_gemm.hip norm-fused variant) — decode is dispatch-gap
    bound, so deleting the rmsnorm_res launch saves its kernel time
    AND the ~5 us graph-replay gap.  Returns (x2, y) with x2 = x+res
    (x itself when res is None).  Falls back to rmsnorm_res +
    decode_linear when the fp8 table or shape doesn't apply."""
    lead = x.shape[:-1]
    n = 1
    for d in lead:
        n *= d
    i = x.shape[-1]
    o = weight.shape[0]
    # MEASURED NEGATIVE as the default path (371 -> 246 tok/s @1):
    # the dot loop's x reads moved from global (4-deep vm pipeline,
    # latency hidden under the nontemporal weight stream) to flat-LDS
    # loads, which serialize on lgkmcnt and stall the stream.  The
    # launch+gap saving (~10 us) never amortized the per-block slowdown.
    # Kept as a tested lib op behind SKY_FP8_NORM_FUSED=1.
    fused_on = os.environ.get("SKY_FP8_NORM_FUSED", "0") == "1"
    if (fused_on and x.is_cuda and x.dtype == torch.bfloat16
            and 1 <= n <= 2 and i % 1024 == 0 and i <= 4096
            and o % 2 == 0 and _FP8_WEIGHTS):
        ent = _FP8_WEIGHTS.get(weight.data_ptr())
        if ent is not None and tuple(ent[0].shape) != tuple(weight.shape):
            ent = None
        if ent is not None:
            C = _require_native("skinny_gemm_fp8_norm")
            res2 = (res.reshape(n, i).contiguous() if res is not None
                    else torch.empty(0, dtype=x.dtype, device=x.device))
            x2, y = C.skinny_gemm_fp8_norm(
                x.reshape(n, i).contiguous(), res2, norm_w, eps,
                ent[0], ent[1], res is not None)
            return x2.view(*lead, i), y.view(*lead, o)
    if res is None:
        return x, decode_linear(rmsnorm(x, norm_w, eps), weight)
    x2, h = rmsnorm_res(x, res, norm_w, eps)
    return x2, decode_linear(h, weight)


def rmsnorm_res(x, res, weight, eps: float = 1e-5):
    """Fused (x + res, rmsnorm(x + res) * w) for the decode step (one
    kernel instead of add + norm; decode_fused.hip).  No autograd."""
    if x.is_cuda and x.dtype == torch.bfloat16 and x.shape[-1] % 256 == 0:
        C = _require_native("rmsnorm_res")
        x2, h = C.rmsnorm_res(x.contiguous(), res.contiguous(), weight,
                              eps)
        return x2, h
    x2 = (x + res)
    return x2, rmsnorm(x2, weight, eps)


def decode_advance(logits, stage, ring, ctr):
    """Fused greedy decode advance: per-row argmax over `logits`
    [b, 1, V] plus the in-graph state bump — ring token store,
    stage[0] next-token feedback, stage[1]/[3]/[4] increments — in one
    launch (decode_fused.hip; replaces torch argmax+index_copy+copy_+
    3 adds; tie-break matches torch.argmax).  GPU-only: the decode
    graph never runs on CPU."""
    C = _require_native("decode_advance")
    C.decode_advance(logits.contiguous(), stage, ring, ctr)


def decode_swiglu_down(gu, weight):
    """silu(gate)*up followed by the down projection.  MEASURED
    NEGATIVE as a fused GEMV (skinny_gemm_swiglu kept for reference):
    fusing an elementwise producer into a GEMV recomputes it once PER
    OUTPUT ROW (O x M silu evaluations instead of M) — 289 -> 248
    tok/s single-stream.  The separate swiglu kernel + GEMV stays."""
    return decode_linear(swiglu(gu), weight)


def attn_decode_qkv(qkv, kc, vc, cos, sin, positions, kv_lens, slot_ids,
                    Hq, Hkv, scale):
    """Fused rope + KV-cache write + decode attention from the RAW
    packed qkv GEMV output (attention_decode.hip): the current token's
    roped k / raw v stay in registers for this step's attention (no
    write->read hazard) while one block per kv head writes the cache
    row for future steps.  Replaces rope_kvwrite + attn_decode (two
    launches -> one).  CPU/odd-D fallback composes the two ops."""
    D = kc.shape[3]
    if qkv.is_cuda and qkv.dtype == torch.bfloat16 and D == 128:
        C = _require_native("attn_decode_qkv")
        return C.attn_decode_qkv(qkv.contiguous(), kc, vc, cos, sin,
                                 positions.int(), kv_lens.int(),
                                 slot_ids.int(), Hq, Hkv, scale)
    q = rope_kvwrite(qkv, kc, vc, cos, sin, positions, slot_ids, Hq, Hkv)
    return attn_decode(q, kc, vc, kv_lens, slot_ids, scale)


def rope_kvwrite(qkv, kc, vc, cos, sin, positions, slot_ids, Hq, Hkv):
    """Packed-qkv rope + KV-cache scatter (decode_fused.hip): applies
    rope to the q and k segments of qkv [n, (Hq+2*Hkv)*D], writes the
    roped k and raw v rows into the caches at (slot_ids, positions),
    and returns contiguous roped q [n, Hq, D].  No autograd."""
    D = kc.shape[3]
    n = qkv.shape[0]
    if qkv.is_cuda and qkv.dtype == torch.bfloat16 and D == 128:
        C = _require_native("rope_kvwrite")
        return C.rope_kvwrite(qkv.contiguous(), kc, vc, cos, sin,
                              positions.int(), slot_ids.int(), Hq, Hkv)
    q, k, v = qkv.split([Hq * D, Hkv * D, Hkv * D], dim=-1)
    qr = rope(q.reshape(n, Hq, D), cos, sin, positions)
    kr = rope(k.reshape(n, Hkv, D), cos, sin, positions)
    pos = positions.long()
    kc[slot_ids.long(), pos] = kr
    vc[slot_ids.long(), pos] = v.reshape(n, Hkv, D)
    return qr


__all__ = [
    "rmsnorm", "rope", "attention", "fused_cross_entropy", "attn_decode",
    "swiglu", "decode_linear", "rmsnorm_res", "rope_kvwrite",
    "decode_swiglu_down",
    "native", "native_available", "rmsnorm_ref", "rope_ref",
    "attention_ref", "cross_entropy_ref", "attn_decode_ref",
]
