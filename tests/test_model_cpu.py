"""CPU tests for the model / trainer / DDP stack (reference op paths)."""
import os

import pytest
import torch

from skypilot_amd.models.llama import CONFIGS, build_model
from skypilot_amd.train.optim import FusedAdamW


def test_model_forward_shapes():
    m = build_model("llama-debug", dtype=torch.float32)
    tok = torch.randint(0, 512, (2, 64))
    logits = m(tok)
    assert logits.shape == (2, 64, 512)


def test_model_loss_decreases_cpu():
    torch.manual_seed(0)
    m = build_model("llama-debug", dtype=torch.float32)
    opt = FusedAdamW(m.parameters(), lr=1e-3)
    tok = torch.randint(0, 512, (2, 65))
    inputs, targets = tok[:, :-1], tok[:, 1:].contiguous()
    losses = []
    for _ in range(6):
        opt.zero_grad()
        loss = m.loss(inputs, targets)
        loss.backward()
        opt.step()
        losses.append(float(loss))
    assert losses[-1] < losses[0], losses


def test_flops_per_token_sane():
    m = build_model("llama-debug", dtype=torch.float32)
    f = m.flops_per_token(128)
    assert f > 6 * m.num_params()


def test_config_inventory():
    assert "llama3-8b" in CONFIGS and "llama3-70b" in CONFIGS
    c = CONFIGS["llama3-8b"]
    assert c.hidden_size == 4096 and c.num_layers == 32
    assert c.num_kv_heads == 8 and c.vocab_size == 128256


def _ddp_worker(rank, world, port):
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": str(world), "LOCAL_RANK": str(rank),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        torch.manual_seed(100 + rank)  # different init per rank
        from skypilot_amd.parallel.ddp import BucketedDDP
        m = build_model("llama-debug", dtype=torch.float32, seed=rank)
        ddp = BucketedDDP(m, bucket_bytes=1 << 20)
        torch.manual_seed(7 + rank)
        tok = torch.randint(0, 512, (1, 33))
        ddp.zero_grad()
        ddp.mark_step_start()
        loss = m.loss(tok[:, :-1], tok[:, 1:].contiguous())
        loss.backward()
        ddp.finish()
        # After broadcast at init, params identical; grads summed: verify a
        # grad is identical across ranks by all-gathering a slice.
        g = m.lm_head.weight._sky_grad.flatten()[:128].clone()
        gather = [torch.zeros_like(g) for _ in range(world)]
        dist.all_gather(gather, g)
        assert torch.allclose(gather[0], gather[1], atol=1e-6)
        opt = FusedAdamW(m.parameters(), lr=1e-3)
        opt.step(grad_scale=ddp.grad_scale)
        p = m.lm_head.weight.flatten()[:128].clone()
        gather2 = [torch.zeros_like(p) for _ in range(world)]
        dist.all_gather(gather2, p)
        assert torch.allclose(gather2[0], gather2[1], atol=1e-6)
    finally:
        dist.destroy_process_group()


def test_ddp_two_process_gloo():
    import multiprocessing as mp
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_ddp_worker, args=(r, 2, port))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=300)
    for p in procs:
        assert p.exitcode == 0


def test_grad_accum_matches_big_batch():
    """grad_accum=2 with batch B must match one step at batch 2B."""
    from skypilot_amd.train.trainer import TrainConfig, Trainer
    torch.manual_seed(0)
    # Tiny lr: bf16 grad noise is amplified to ~lr by Adam's normalizer,
    # so keep the one-step divergence below bf16 parameter resolution.
    cfg1 = TrainConfig(model="llama-debug", micro_batch=2, seq_len=64,
                       device="cpu", seed=11, grad_accum=1, lr=1e-5)
    cfg2 = TrainConfig(model="llama-debug", micro_batch=1, seq_len=64,
                       device="cpu", seed=11, grad_accum=2, lr=1e-5)
    tr1, tr2 = Trainer(cfg1), Trainer(cfg2)
    tok = torch.randint(0, 512, (2, 65))
    batch_big = (tok[:, :-1], tok[:, 1:].contiguous())
    tr1.train_step(batch_big)
    # same data split into two micro-batches
    b1 = (tok[:1, :-1], tok[:1, 1:].contiguous())
    b2 = (tok[1:, :-1], tok[1:, 1:].contiguous())
    c = tr2.cfg
    tr2.ddp.zero_grad()
    tr2.ddp.mark_step_start(accumulating=True)
    (tr2.model.loss(*b1) / 2).backward()
    tr2.ddp.finish()
    tr2.ddp.mark_step_start(accumulating=False)
    (tr2.model.loss(*b2) / 2).backward()
    tr2.ddp.finish()
    # Compare the RAW accumulated gradients (not post-Adam weights, which
    # tiny-lr can mask): the accumulated sum over both micro-batches must
    # match the big-batch gradient, not just the last micro-batch's.
    for pa, pb in zip(tr1.model.parameters(), tr2.model.parameters()):
        ga, gb = pa._sky_grad.float(), pb._sky_grad.float()
        # bf16 buckets: allow a few ulp (2^-8 relative) of rounding noise
        assert torch.allclose(ga, gb, rtol=0.05, atol=1e-4), \
            "accumulated grad != big-batch grad"
    tr2.opt.lr = tr2.current_lr()
    tr2.opt.step(grad_scale=tr2.ddp.grad_scale)
    p1 = tr1.model.lm_head.weight.detach()
    p2 = tr2.model.lm_head.weight.detach()
    assert torch.allclose(p1.float(), p2.float(), atol=5e-4, rtol=0)


def test_grad_accum_sums_not_overwrites():
    """The final micro-step must ADD into the bucket, not overwrite it.

    Regression test for the round-1 bug where mark_step_start(False) made
    the hook copy_ the last micro-batch's grad over the accumulated sum.
    Uses two different micro-batches so sum != last.
    """
    from skypilot_amd.train.trainer import TrainConfig, Trainer
    torch.manual_seed(3)
    cfg = TrainConfig(model="llama-debug", micro_batch=1, seq_len=32,
                      device="cpu", seed=7, grad_accum=2)
    tr = Trainer(cfg)
    t1 = torch.randint(0, 512, (1, 33))
    t2 = torch.randint(0, 512, (1, 33))
    b1 = (t1[:, :-1], t1[:, 1:].contiguous())
    b2 = (t2[:, :-1], t2[:, 1:].contiguous())
    # grad of b2 alone
    tr.ddp.zero_grad()
    tr.ddp.mark_step_start(accumulating=False)
    tr.model.loss(*b2).backward()
    tr.ddp.finish()
    g_last = tr.model.lm_head.weight._sky_grad.float().clone()
    # accumulated grad of b1 then b2
    tr.ddp.zero_grad()
    tr.ddp.mark_step_start(accumulating=True)
    tr.model.loss(*b1).backward()
    tr.ddp.finish()
    tr.ddp.mark_step_start(accumulating=False)
    tr.model.loss(*b2).backward()
    tr.ddp.finish()
    g_sum = tr.model.lm_head.weight._sky_grad.float()
    rel = float((g_sum - g_last).abs().max() / g_last.abs().max())
    assert rel > 0.05, "accumulated grad == last micro-batch grad (overwrite bug)"


def test_lr_schedule():
    from skypilot_amd.train.trainer import TrainConfig, Trainer
    cfg = TrainConfig(model="llama-debug", micro_batch=1, seq_len=64,
                      device="cpu", lr=1e-3, warmup_steps=10,
                      lr_decay_steps=100)
    tr = Trainer(cfg)
    tr.step_count = 0
    assert abs(tr.current_lr() - 1e-4) < 1e-9       # warmup start
    tr.step_count = 9
    assert abs(tr.current_lr() - 1e-3) < 1e-9       # warmup end
    tr.step_count = 100
    assert abs(tr.current_lr() - 1e-4) < 1e-6       # cosine floor


def test_qwen2_family_train_and_bias():
    """Qwen2-style family (GQA + q/k/v biases, 1e6 rope base): training
    reduces loss, and the biases genuinely participate (zeroing them
    changes the forward)."""
    import torch
    from skypilot_amd.train.trainer import TrainConfig, Trainer
    torch.manual_seed(0)
    cfg = TrainConfig(model="qwen2-debug", micro_batch=2, seq_len=64,
                      device="cpu")
    tr = Trainer(cfg)
    tok, tgt = tr.synthetic_batch()
    losses = [tr.train_step((tok, tgt)) for _ in range(6)]
    assert losses[-1] < losses[0], losses
    # bias participation
    m = tr.model
    toks = torch.randint(0, 512, (1, 64))
    with torch.no_grad():
        y1 = m(toks).clone()
        for blk in m.blocks:
            blk.attn.wq.bias.zero_()
            blk.attn.wk.bias.zero_()
            blk.attn.wv.bias.zero_()
        y2 = m(toks)
    assert not torch.allclose(y1, y2)


def test_qwen2_engine_decode_matches_recompute():
    """Cached decode (packed-qkv + bias path) matches full recompute
    for the bias family."""
    import torch
    from skypilot_amd.serve.engine import Engine
    torch.manual_seed(1)
    eng = Engine("qwen2-debug", device="cpu", max_seq=256, max_batch=2)
    eng.start()
    try:
        prompt = [3, 7, 11, 200, 5]
        out = eng.generate(prompt, max_tokens=8)
        ids = list(prompt)
        for _ in range(8):
            logits = eng.model(torch.tensor([ids]))
            ids.append(int(logits[0, -1].argmax()))
        assert out == ids[len(prompt):], (out, ids[len(prompt):])
    finally:
        eng.stop()
