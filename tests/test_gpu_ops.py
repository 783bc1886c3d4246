"""GPU numerics: every HIP kernel vs its plain-PyTorch fp32 reference."""
import math

import pytest
import torch

from skypilot_amd import ops

pytestmark = pytest.mark.gpu


def dev():
    return torch.device("cuda:0")


def rel_err(a, b):
    a, b = a.float(), b.float()
    return ((a - b).norm() / (b.norm() + 1e-12)).item()


def test_mfma_layout_probe():
    """Asymmetric-input check of the documented fragment layouts."""
    torch.manual_seed(0)
    A = (torch.randn(16, 32) * 0.5).bfloat16().to(dev())
    B = (torch.randn(32, 16) * 0.5).bfloat16().to(dev())
    C = ops.native().mfma_probe(A, B)
    expect = A.float() @ B.float()
    assert rel_err(C, expect) < 2e-2, (C[:4, :4], expect[:4, :4])


def test_rmsnorm_fwd_bwd():
    torch.manual_seed(1)
    rows, H = 512, 4096
    x = (torch.randn(rows, H) * 2).bfloat16().to(dev()).requires_grad_()
    w = torch.randn(H).bfloat16().to(dev()).requires_grad_()
    y = ops.rmsnorm(x, w, 1e-5)
    ref = ops.rmsnorm_ref(x.detach().cpu().float(), w.detach().cpu().float(),
                          1e-5)
    assert rel_err(y.cpu(), ref) < 1e-2

    g = torch.randn_like(y)
    y.backward(g)
    x2 = x.detach().cpu().float().requires_grad_()
    w2 = w.detach().cpu().float().requires_grad_()
    inv = torch.rsqrt(x2.pow(2).mean(-1, keepdim=True) + 1e-5)
    (x2 * inv * w2).backward(g.cpu().float())
    assert rel_err(x.grad.cpu(), x2.grad) < 2e-2
    assert rel_err(w.grad.cpu(), w2.grad) < 2e-2


def test_rope_fwd_bwd():
    torch.manual_seed(2)
    T, H, D = 1024, 8, 128
    x = torch.randn(T, H, D).bfloat16().to(dev()).requires_grad_()
    half = D // 2
    inv_freq = 1.0 / (500000.0 ** (torch.arange(half).float() / half))
    freqs = torch.outer(torch.arange(T).float(), inv_freq)
    cos = freqs.cos().to(dev())
    sin = freqs.sin().to(dev())
    pos = torch.arange(T, dtype=torch.int32, device=dev())
    y = ops.rope(x, cos, sin, pos)
    ref = ops.rope_ref(x.detach().cpu(), cos.cpu(), sin.cpu(), pos.cpu())
    assert rel_err(y.cpu(), ref.float()) < 1e-2
    g = torch.randn_like(y)
    y.backward(g)
    ref_g = ops.rope_ref(g.cpu(), cos.cpu(), sin.cpu(), pos.cpu(),
                         backward=True)
    assert rel_err(x.grad.cpu(), ref_g.float()) < 1e-2


@pytest.mark.parametrize("B,S,Hq,Hkv", [(2, 256, 8, 2), (1, 1024, 4, 4),
                                        (1, 512, 4, 1)])
def test_attention_fwd(B, S, Hq, Hkv):
    torch.manual_seed(3)
    D = 128
    q = (torch.randn(B, S, Hq, D) * 0.5).bfloat16().to(dev())
    k = (torch.randn(B, S, Hkv, D) * 0.5).bfloat16().to(dev())
    v = (torch.randn(B, S, Hkv, D) * 0.5).bfloat16().to(dev())
    out = ops.attention(q, k, v)
    ref = ops.attention_ref(q.cpu().float(), k.cpu().float(),
                            v.cpu().float(), 1.0 / math.sqrt(D), causal=True)
    assert rel_err(out.cpu(), ref) < 2e-2


def test_attention_bwd():
    torch.manual_seed(4)
    B, S, Hq, Hkv, D = 1, 256, 4, 2, 128
    q0 = torch.randn(B, S, Hq, D) * 0.5
    k0 = torch.randn(B, S, Hkv, D) * 0.5
    v0 = torch.randn(B, S, Hkv, D) * 0.5
    g0 = torch.randn(B, S, Hq, D) * 0.5

    q = q0.bfloat16().to(dev()).requires_grad_()
    k = k0.bfloat16().to(dev()).requires_grad_()
    v = v0.bfloat16().to(dev()).requires_grad_()
    out = ops.attention(q, k, v)
    out.backward(g0.bfloat16().to(dev()))

    qr = q0.clone().requires_grad_()
    kr = k0.clone().requires_grad_()
    vr = v0.clone().requires_grad_()
    ref = ops.attention_ref(qr, kr, vr, 1.0 / math.sqrt(D), causal=True)
    ref.backward(g0)

    assert rel_err(q.grad.cpu(), qr.grad) < 3e-2
    assert rel_err(k.grad.cpu(), kr.grad) < 3e-2
    assert rel_err(v.grad.cpu(), vr.grad) < 3e-2


def test_cross_entropy_fused():
    torch.manual_seed(5)
    N, V = 512, 128256
    logits0 = torch.randn(N, V) * 2
    targets = torch.randint(0, V, (N,))
    targets[7] = -100

    logits = logits0.bfloat16().to(dev()).requires_grad_()
    loss = ops.fused_cross_entropy(logits, targets.int().to(dev()))
    expect = torch.nn.functional.cross_entropy(
        logits0.bfloat16().float(), targets.long(), ignore_index=-100)
    assert abs(loss.item() - expect.item()) < 2e-2 * abs(expect.item())

    loss.backward()
    l2 = logits0.bfloat16().float().requires_grad_()
    torch.nn.functional.cross_entropy(l2, targets.long(),
                                      ignore_index=-100).backward()
    assert rel_err(logits.grad.cpu(), l2.grad) < 2e-2


def test_adamw_matches_torch():
    torch.manual_seed(6)
    from skypilot_amd.train.optim import FusedAdamW
    shapes = [(256, 512), (1000,), (31,)]  # incl. non-multiple-of-4
    params = [torch.randn(*s).bfloat16().to(dev()).requires_grad_()
              for s in shapes]
    ref_params = [p.detach().float().cpu().clone().requires_grad_()
                  for p in params]
    opt = FusedAdamW(params, lr=1e-2, weight_decay=0.1, decay_2d_only=False)
    ref_opt = torch.optim.AdamW(ref_params, lr=1e-2, betas=(0.9, 0.95),
                                eps=1e-8, weight_decay=0.1)
    for step in range(5):
        grads = [torch.randn(*s) for s in shapes]
        for p, g in zip(params, grads):
            p.grad = g.bfloat16().to(dev())
        for p, g in zip(ref_params, grads):
            p.grad = g.bfloat16().float()
        opt.step()
        ref_opt.step()
    for p, rp, s in zip(params, ref_params, shapes):
        assert rel_err(p.cpu(), rp.detach().bfloat16().float()) < 2e-2, s


def test_model_loss_decreases_gpu():
    from skypilot_amd.train.trainer import TrainConfig, Trainer
    cfg = TrainConfig(model="llama-debug", micro_batch=2, seq_len=128)
    tr = Trainer(cfg)
    tok, tgt = tr.synthetic_batch()
    losses = [tr.train_step((tok, tgt)) for _ in range(8)]
    assert losses[-1] < losses[0], losses


def test_attn_decode_kernel():
    torch.manual_seed(7)
    B, Hq, Hkv, D, S_max = 5, 8, 2, 128, 512
    q = (torch.randn(B, Hq, D) * 0.5).bfloat16().to(dev())
    kc = (torch.randn(B + 2, S_max, Hkv, D) * 0.5).bfloat16().to(dev())
    vc = (torch.randn(B + 2, S_max, Hkv, D) * 0.5).bfloat16().to(dev())
    kv_lens = torch.tensor([3, 100, 512, 1, 77], dtype=torch.int32,
                           device=dev())
    slot_ids = torch.tensor([6, 0, 2, 4, 3], dtype=torch.int32,
                            device=dev())
    out = ops.attn_decode(q, kc, vc, kv_lens, slot_ids, 128 ** -0.5)
    ref = ops.attn_decode_ref(q.cpu().float(), kc.cpu().float(),
                              vc.cpu().float(), kv_lens.cpu(),
                              slot_ids.cpu(), 128 ** -0.5)
    assert rel_err(out.cpu(), ref) < 2e-2


def test_engine_cached_decode_matches_gpu():
    from skypilot_amd.serve.engine import Engine
    eng = Engine("llama-smoke", device="cuda:0", max_seq=256, max_batch=4)
    eng.start()
    try:
        prompt = [1, 5, 9, 200, 3, 77, 1000]
        out = eng.generate(prompt, max_tokens=8)
        ids = list(prompt)
        for _ in range(8):
            # full recompute, padded to the 64-row attention tile
            toks = torch.zeros(1, 64, dtype=torch.long, device=dev())
            toks[0, :len(ids)] = torch.tensor(ids, device=dev())
            logits = eng.model(toks)
            ids.append(int(logits[0, len(ids) - 1].argmax()))
        # bf16 non-associativity can flip an argmax occasionally; require
        # most tokens to match between cached and full recompute.
        matches = sum(a == b for a, b in zip(out, ids[len(prompt):]))
        assert matches >= 6, (out, ids[len(prompt):])
    finally:
        eng.stop()


def test_swiglu_fwd_bwd():
    torch.manual_seed(8)
    rows, M = 2048, 14336
    gu0 = torch.randn(rows, 2 * M)
    dy0 = torch.randn(rows, M)
    gu = gu0.bfloat16().to(dev()).requires_grad_()
    y = ops.swiglu(gu)
    y.backward(dy0.bfloat16().to(dev()))

    gu_ref = gu0.bfloat16().float().requires_grad_()
    g, u = gu_ref.split(M, dim=-1)
    y_ref = torch.nn.functional.silu(g) * u
    y_ref.backward(dy0)
    assert rel_err(y.cpu(), y_ref) < 2e-2
    assert rel_err(gu.grad.cpu(), gu_ref.grad) < 2e-2


def test_mfma32_layout_probe():
    torch.manual_seed(10)
    A = (torch.randn(32, 16) * 0.5).bfloat16().to(dev())
    B = (torch.randn(16, 32) * 0.5).bfloat16().to(dev())
    C = ops.native().mfma32_probe(A, B)
    expect = A.float() @ B.float()
    assert rel_err(C, expect) < 2e-2, (C[:3, :3], expect[:3, :3])


@pytest.mark.gpu
def test_skinny_gemm_matches_fp32_linear():
    """Decode GEMV (skinny_gemm.hip) vs plain fp32 F.linear."""
    C = ops.native()
    torch.manual_seed(3)
    for n, i, o in [(1, 4096, 4096), (3, 4096, 1024), (8, 14336, 4096),
                    (2, 4096, 128256), (5, 512, 512)]:
        x = (torch.randn(n, i, device="cuda") * 0.5).bfloat16()
        w = (torch.randn(o, i, device="cuda") * 0.02).bfloat16()
        y = C.skinny_gemm(x, w)
        ref = torch.nn.functional.linear(x.float(), w.float())
        assert y.shape == (n, o)
        torch.testing.assert_close(y.float(), ref, atol=0.02, rtol=0.02)
    # dispatcher: lead-dim flattening + CPU fallback parity
    x3 = (torch.randn(2, 1, 4096, device="cuda") * 0.5).bfloat16()
    w = (torch.randn(1024, 4096, device="cuda") * 0.02).bfloat16()
    out = ops.decode_linear(x3, w)
    assert out.shape == (2, 1, 1024)
    torch.testing.assert_close(
        out.float(), torch.nn.functional.linear(x3.float(), w.float()),
        atol=0.02, rtol=0.02)


@pytest.mark.gpu
def test_rmsnorm_res_matches_ref():
    """Fused residual-add + rmsnorm (decode_fused.hip) vs fp32 ref."""
    C = ops.native()
    torch.manual_seed(4)
    for n, h in [(1, 4096), (8, 4096), (3, 256), (2, 8192)]:
        x = (torch.randn(n, h, device="cuda")).bfloat16()
        r = (torch.randn(n, h, device="cuda")).bfloat16()
        w = (torch.randn(h, device="cuda") * 0.1 + 1).bfloat16()
        x2, out = C.rmsnorm_res(x, r, w, 1e-5)
        ref_x2 = (x + r)
        torch.testing.assert_close(x2, ref_x2)
        rf = ref_x2.float()
        ref = rf * torch.rsqrt(rf.pow(2).mean(-1, keepdim=True) + 1e-5) \
            * w.float()
        torch.testing.assert_close(out.float(), ref, atol=0.05, rtol=0.05)


@pytest.mark.gpu
def test_rope_kvwrite_matches_ref():
    """Packed rope + cache scatter (decode_fused.hip) vs composed ref."""
    C = ops.native()
    torch.manual_seed(5)
    n, Hq, Hkv, D, slots, S = 4, 8, 2, 128, 6, 64
    qkv = (torch.randn(n, (Hq + 2 * Hkv) * D, device="cuda")).bfloat16()
    kc = torch.zeros(slots, S, Hkv, D, device="cuda",
                     dtype=torch.bfloat16)
    vc = torch.zeros_like(kc)
    kc_ref, vc_ref = kc.clone(), vc.clone()
    half = D // 2
    t = torch.arange(S, device="cuda", dtype=torch.float32)
    inv = 1.0 / (10000 ** (torch.arange(half, device="cuda").float()
                           / half))
    freqs = torch.outer(t, inv)
    cos, sin = freqs.cos().contiguous(), freqs.sin().contiguous()
    positions = torch.tensor([0, 5, 63, 17], device="cuda",
                             dtype=torch.int32)
    slot_ids = torch.tensor([5, 0, 3, 2], device="cuda",
                            dtype=torch.int32)
    q = C.rope_kvwrite(qkv, kc, vc, cos, sin, positions, slot_ids,
                       Hq, Hkv)
    # composed reference (same ops the CPU fallback uses)
    qs, ks, vs = qkv.split([Hq * D, Hkv * D, Hkv * D], dim=-1)

    def rope_ref(xx, H):
        xx = xx.reshape(n, H, D).float()
        x1, x2 = xx[..., :half], xx[..., half:]
        c = cos[positions.long()].unsqueeze(1)
        s = sin[positions.long()].unsqueeze(1)
        return torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], -1)

    torch.testing.assert_close(q.float(), rope_ref(qs, Hq), atol=0.02,
                               rtol=0.02)
    kr = rope_ref(ks, Hkv).bfloat16()
    kc_ref[slot_ids.long(), positions.long()] = kr
    vc_ref[slot_ids.long(), positions.long()] = vs.reshape(n, Hkv, D)
    torch.testing.assert_close(kc.float(), kc_ref.float(), atol=0.02,
                               rtol=0.02)
    torch.testing.assert_close(vc, vc_ref)


@pytest.mark.gpu
def test_attention_shape_fuzz():
    """Swapped fwd/bwd kernels across shapes: batch, GQA ratio, causal,
    seq length — all against the fp32 reference."""
    torch.manual_seed(7)
    for (B, S, Hq, Hkv, causal) in [
            (1, 64, 8, 8, True), (3, 128, 16, 2, True),
            (2, 256, 32, 8, False), (1, 1024, 8, 2, True),
            (2, 192, 4, 4, False),
            # S % 256 == 0 routes to the v3 32x32 kernel (4-wave by
            # default; SKY_ATTN_FWD_V3_NW=8 selects the 8-wave twin)
            (2, 512, 16, 4, True), (1, 256, 8, 8, True),
            # v3 edge shapes around the r02 gl_lds barrier race:
            # minimal paired grid (nq=2), single-head, non-causal v3,
            # odd head ratios, and a deep-S single-block case
            (1, 256, 1, 1, True), (1, 256, 8, 8, False),
            (4, 768, 6, 2, True), (1, 2048, 1, 1, True),
            (3, 512, 5, 1, True)]:
        q = (torch.randn(B, S, Hq, 128, device="cuda") * 0.5).bfloat16()
        k = (torch.randn(B, S, Hkv, 128, device="cuda") * 0.5).bfloat16()
        v = (torch.randn(B, S, Hkv, 128, device="cuda") * 0.5).bfloat16()
        q.requires_grad_(True)
        k.requires_grad_(True)
        v.requires_grad_(True)
        out = ops.attention(q, k, v, 128 ** -0.5, causal=causal)
        g = torch.randn_like(out) * 0.5
        out.backward(g)
        qf = q.detach().float().requires_grad_(True)
        kf = k.detach().float().requires_grad_(True)
        vf = v.detach().float().requires_grad_(True)
        ref = ops.attention_ref(qf, kf, vf, 128 ** -0.5, causal=causal)
        ref.backward(g.float())
        cfg = (B, S, Hq, Hkv, causal)
        torch.testing.assert_close(out.float(), ref, atol=0.05,
                                   rtol=0.05, msg=str(cfg))
        torch.testing.assert_close(q.grad.float(), qf.grad, atol=0.08,
                                   rtol=0.08, msg=str(cfg))
        torch.testing.assert_close(k.grad.float(), kf.grad, atol=0.08,
                                   rtol=0.08, msg=str(cfg))
        torch.testing.assert_close(v.grad.float(), vf.grad, atol=0.08,
                                   rtol=0.08, msg=str(cfg))


@pytest.mark.gpu
def test_skinny_gemm_swiglu_matches_ref():
    """Fused silu(g)*u + down GEMV vs composed fp32 reference."""
    C = ops.native()
    torch.manual_seed(13)
    for n, m, o in [(1, 14336, 4096), (2, 512, 1024)]:
        gu = (torch.randn(n, 2 * m, device="cuda") * 0.5).bfloat16()
        w = (torch.randn(o, m, device="cuda") * 0.02).bfloat16()
        y = C.skinny_gemm_swiglu(gu, w)
        g, u = gu.float().split(m, dim=-1)
        x = torch.nn.functional.silu(g) * u
        ref = torch.nn.functional.linear(x, w.float())
        torch.testing.assert_close(y.float(), ref, atol=0.05, rtol=0.05)


@pytest.mark.gpu
def test_wgrad_tn_matches_ref():
    """Transpose-then-NT wgrad GEMM vs fp32 reference."""
    C = ops.native()
    torch.manual_seed(5)
    for (M, I, J) in [(256, 128, 256), (512, 256, 512), (1024, 384, 256)]:
        dy = (torch.randn(M, I, device="cuda") * 0.3).bfloat16()
        x = (torch.randn(M, J, device="cuda") * 0.3).bfloat16()
        out = C.wgrad_tn(dy, x)
        ref = dy.t().float() @ x.float()
        torch.testing.assert_close(out.float(), ref, atol=0.5, rtol=0.02,
                                   msg=str((M, I, J)))
        # transpose alone
        t = C.transpose_bf16(dy)
        assert torch.equal(t, dy.t().contiguous())


@pytest.mark.gpu
def test_skinny_gemm_fp8_matches_ref():
    """fp8 (e4m3fn rowwise) decode GEMV vs fp32 reference: the error
    must be bounded by the quantization error itself (VERDICT r01 #7
    fp8 numerics harness)."""
    C = ops.native()
    torch.manual_seed(11)
    for n, i, o in [(1, 4096, 4096), (2, 14336, 4096), (4, 1024, 2048),
                    (8, 2048, 1024)]:
        x = (torch.randn(n, i, device="cuda") * 0.5).bfloat16()
        w = (torch.randn(o, i, device="cuda") * 0.02).bfloat16()
        q8, scale = ops.quantize_fp8_rowwise(w)
        y = C.skinny_gemm_fp8(x, q8, scale)
        # reference through the SAME quantized weights (isolates kernel
        # error from quantization error)...
        wq = q8.view(torch.float8_e4m3fn).float() * scale[:, None]
        ref_q = x.float() @ wq.t()
        torch.testing.assert_close(y.float(), ref_q, atol=0.02, rtol=0.02)
        # ...and against the unquantized weights with fp8-scale bounds.
        ref = x.float() @ w.float().t()
        err = (y.float() - ref).abs().max()
        bound = 0.04 * ref.abs().max() + 0.5
        assert err < bound, (n, i, o, float(err), float(bound))


@pytest.mark.gpu
def test_fp8_decode_linear_routing():
    """Registered weights reroute ops.decode_linear through the fp8
    GEMV; unregistered weights keep the bf16 path."""
    torch.manual_seed(12)
    x = (torch.randn(1, 4096, device="cuda") * 0.5).bfloat16()
    w = (torch.randn(2048, 4096, device="cuda") * 0.02).bfloat16()
    y_bf16 = ops.decode_linear(x, w)
    ops.register_fp8_weight(w)
    try:
        y_fp8 = ops.decode_linear(x, w)
        # outputs differ slightly (quantization) but agree loosely
        assert not torch.equal(y_bf16, y_fp8)
        torch.testing.assert_close(y_fp8.float(), y_bf16.float(),
                                   atol=0.25, rtol=0.1)
    finally:
        ops.clear_fp8_weights()


@pytest.mark.gpu
def test_decode_advance_fused():
    """Fused argmax + decode state bump vs the torch reference chain
    (tie-break must match torch.argmax: lowest index)."""
    torch.manual_seed(11)
    for b, V, chunk in [(1, 128256, 8), (4, 4096, 8), (8, 512, 4)]:
        logits = (torch.randn(b, V, device="cuda") * 2).bfloat16()
        # force ties on row 0: two equal maxima, lowest index must win
        logits[0, 7] = 40.0
        logits[0, V - 3] = 40.0
        stage = torch.randint(0, 100, (6, b), dtype=torch.int32,
                              device="cuda")
        ring = torch.zeros(chunk, b, dtype=torch.long, device="cuda")
        ctr = torch.tensor([5], dtype=torch.long, device="cuda")
        ref_stage = stage.clone()
        ref_ring = ring.clone()
        am = logits.float().argmax(-1)
        ref_ring[5 % chunk] = am
        ref_stage[0] = am.int()
        for r in (1, 3, 4):
            ref_stage[r] += 1
        ops.decode_advance(logits, stage, ring, ctr)
        torch.cuda.synchronize()
        assert int(stage[0, 0]) == 7  # tie: lowest index
        assert torch.equal(stage, ref_stage), (b, V)
        assert torch.equal(ring, ref_ring), (b, V)


@pytest.mark.gpu
def test_decode_graph_chain_matches_eager():
    """The graph-captured greedy chain (with the fused advance) must
    produce the same tokens as step-by-step eager decode."""
    from skypilot_amd.serve.engine import Engine
    eng = Engine("llama-smoke", device="cuda:0", max_seq=256, max_batch=2)
    eng.start()
    try:
        prompt = list(range(1, 33))
        out1 = eng.generate(prompt, max_tokens=24)
        out2 = eng.generate(prompt, max_tokens=24)
        assert out1 == out2  # deterministic greedy
        assert len(out1) == 24
    finally:
        eng.stop()


@pytest.mark.gpu
def test_attn_decode_qkv_fused():
    """Fused rope+cache-write+attention vs the two-kernel chain
    (rope_kvwrite then attn_decode) on the same random cache state —
    outputs AND written cache rows must match."""
    torch.manual_seed(13)
    B, Hq, Hkv, D, S_max = 5, 8, 2, 128, 512
    half = D // 2
    inv_freq = 1.0 / (500000.0 ** (torch.arange(half).float() / half))
    freqs = torch.outer(torch.arange(S_max).float(), inv_freq)
    cos = freqs.cos().to(dev())
    sin = freqs.sin().to(dev())
    qkv = (torch.randn(B, (Hq + 2 * Hkv) * D) * 0.5).bfloat16().to(dev())
    kv_lens = torch.tensor([3, 100, 512, 1, 77], dtype=torch.int32,
                           device=dev())
    positions = kv_lens - 1
    slot_ids = torch.tensor([6, 0, 2, 4, 3], dtype=torch.int32,
                            device=dev())
    kc0 = (torch.randn(8, S_max, Hkv, D) * 0.5).bfloat16().to(dev())
    vc0 = (torch.randn(8, S_max, Hkv, D) * 0.5).bfloat16().to(dev())
    # reference chain
    kc_ref, vc_ref = kc0.clone(), vc0.clone()
    q = ops.rope_kvwrite(qkv, kc_ref, vc_ref, cos, sin, positions,
                         slot_ids, Hq, Hkv)
    o_ref = ops.attn_decode(q, kc_ref, vc_ref, kv_lens, slot_ids,
                            128 ** -0.5)
    # fused
    kc_f, vc_f = kc0.clone(), vc0.clone()
    o = ops.attn_decode_qkv(qkv, kc_f, vc_f, cos, sin, positions,
                            kv_lens, slot_ids, Hq, Hkv, 128 ** -0.5)
    torch.cuda.synchronize()
    assert rel_err(o, o_ref) < 2e-2
    # cache rows written identically (incl. the roped k)
    assert torch.equal(kc_f, kc_ref)
    assert torch.equal(vc_f, vc_ref)


@pytest.mark.gpu
def test_decode_norm_linear_fused(monkeypatch):
    """Norm-fused fp8 GEMV (lib op, off by default — measured negative
    as the routed path, docs/BENCHMARKS.md) vs the two-kernel chain on
    the SAME quantized weights."""
    monkeypatch.setenv("SKY_FP8_NORM_FUSED", "1")
    torch.manual_seed(17)
    from skypilot_amd.ops import (_FP8_WEIGHTS, decode_norm_linear,
                                  register_fp8_weight)
    I, O = 4096, 6144
    w = (torch.randn(O, I, device="cuda") * 0.02).bfloat16()
    register_fp8_weight(w)
    try:
        nw = torch.randn(I, device="cuda").bfloat16().abs() + 0.5
        for n, with_res in [(1, True), (2, True), (1, False)]:
            x = (torch.randn(n, I, device="cuda") * 0.5).bfloat16()
            res = ((torch.randn(n, I, device="cuda") * 0.5).bfloat16()
                   if with_res else None)
            x2, y = decode_norm_linear(x, res, nw, 1e-5, w)
            # reference chain on the same fp8 table
            if with_res:
                x2_ref, h = ops.rmsnorm_res(x, res, nw, 1e-5)
            else:
                x2_ref, h = x, ops.rmsnorm(x, nw, 1e-5)
            y_ref = ops.decode_linear(h, w)
            torch.cuda.synchronize()
            assert rel_err(y, y_ref) < 2e-2, (n, with_res)
            assert torch.allclose(x2.float(), x2_ref.float(),
                                  atol=1e-2, rtol=1e-2)
    finally:
        _FP8_WEIGHTS.pop(w.data_ptr(), None)
