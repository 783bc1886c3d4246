"""Tensor-parallel correctness: TP=2 sharded forward/backward must match
the single-process model (gloo, CPU)."""
import os

import pytest
import torch


def _tp_worker(rank, world, port, q):
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": str(world), "LOCAL_RANK": str(rank),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from skypilot_amd.models.llama import build_model
        from skypilot_amd.parallel.tp import TPLlama, shard_from_full
        from skypilot_amd.models.llama import CONFIGS
        torch.manual_seed(0)
        full = build_model("llama-smoke", dtype=torch.float32)
        tp_model = TPLlama(CONFIGS["llama-smoke"], world).to(torch.float32)
        shard_from_full(tp_model, full, rank, world)

        torch.manual_seed(42)
        tokens = torch.randint(0, 4096, (1, 64))
        logits_tp = tp_model(tokens)
        logits_full = full(tokens)
        err = (logits_tp - logits_full).abs().max().item()
        assert err < 1e-3, f"rank {rank}: fwd err {err}"

        # Backward: grads of a shared (replicated) param must match the
        # full model's after the TP all-reduces.
        loss_tp = logits_tp.float().pow(2).mean()
        loss_tp.backward()
        loss_full = logits_full.float().pow(2).mean()
        loss_full.backward()
        g_tp = tp_model.blocks[0].attn_norm.grad
        g_full = full.blocks[0].attn_norm.grad
        err = (g_tp - g_full).abs().max().item()
        assert err < 1e-4, f"rank {rank}: norm-grad err {err}"
        # Sharded wq grad matches the corresponding rows of the full grad.
        d = full.cfg.head_dim
        per = full.cfg.num_heads // world * d
        gq_full = full.blocks[0].attn.wq.weight.grad[
            rank * per:(rank + 1) * per]
        gq_tp = tp_model.blocks[0].attn.wq.weight.grad
        err = (gq_tp - gq_full).abs().max().item()
        assert err < 1e-4, f"rank {rank}: wq-grad err {err}"
        q.put((rank, "ok"))
    except Exception as e:  # noqa: BLE001
        q.put((rank, f"{type(e).__name__}: {e}"))
        raise
    finally:
        dist.destroy_process_group()


def test_tp2_matches_single_process():
    import multiprocessing as mp
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_tp_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=300) for _ in range(2)]
    for p in procs:
        p.join(timeout=60)
    for rank, res in results:
        assert res == "ok", f"rank {rank}: {res}"
