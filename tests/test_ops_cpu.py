"""CPU tests of the op wrappers + reference implementations.

The GPU numerics tests (test_gpu_ops.py) compare the HIP kernels against
these same references, so these tests pin the references themselves to
plain-PyTorch ground truth.
"""
import math

import pytest
import torch

from skypilot_amd import ops


def test_rmsnorm_matches_manual():
    torch.manual_seed(0)
    x = torch.randn(8, 256, dtype=torch.bfloat16)
    w = torch.randn(256, dtype=torch.bfloat16)
    y = ops.rmsnorm(x, w, 1e-5)
    xf = x.float()
    expect = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + 1e-5)
    expect = (expect * w.float()).to(torch.bfloat16)
    assert torch.allclose(y.float(), expect.float(), atol=1e-2, rtol=1e-2)


def test_rmsnorm_autograd_matches_torch():
    torch.manual_seed(1)
    x = torch.randn(8, 128, dtype=torch.float32, requires_grad=True)
    w = torch.randn(128, dtype=torch.float32, requires_grad=True)
    y = ops.rmsnorm(x, w, 1e-5)
    g = torch.randn_like(y)
    y.backward(g)

    x2 = x.detach().clone().requires_grad_()
    w2 = w.detach().clone().requires_grad_()
    inv = torch.rsqrt(x2.pow(2).mean(-1, keepdim=True) + 1e-5)
    (x2 * inv * w2).backward(g)
    assert torch.allclose(x.grad, x2.grad, atol=1e-4, rtol=1e-4)
    assert torch.allclose(w.grad, w2.grad, atol=1e-4, rtol=1e-4)


def test_rope_round_trip():
    torch.manual_seed(2)
    T, H, D = 6, 2, 128
    x = torch.randn(T, H, D, dtype=torch.bfloat16)
    pos = torch.arange(T, dtype=torch.int32)
    half = D // 2
    inv_freq = 1.0 / (10000.0 ** (torch.arange(half).float() / half))
    freqs = torch.outer(torch.arange(T).float(), inv_freq)
    cos, sin = freqs.cos(), freqs.sin()
    y = ops.rope(x, cos, sin, pos)
    back = ops.rope_ref(y, cos, sin, pos, backward=True)
    assert torch.allclose(back.float(), x.float(), atol=3e-2, rtol=3e-2)


def test_rope_preserves_norm():
    torch.manual_seed(3)
    x = torch.randn(4, 1, 128)
    pos = torch.arange(4, dtype=torch.int32)
    half = 64
    inv_freq = 1.0 / (500000.0 ** (torch.arange(half).float() / half))
    freqs = torch.outer(torch.arange(4).float(), inv_freq)
    y = ops.rope(x, freqs.cos(), freqs.sin(), pos)
    assert torch.allclose(x.norm(dim=-1), y.norm(dim=-1), atol=1e-4,
                          rtol=1e-4)


def test_attention_ref_matches_sdpa():
    torch.manual_seed(4)
    B, S, Hq, Hkv, D = 2, 64, 4, 2, 128
    q = torch.randn(B, S, Hq, D)
    k = torch.randn(B, S, Hkv, D)
    v = torch.randn(B, S, Hkv, D)
    out = ops.attention_ref(q, k, v, 1.0 / math.sqrt(D), causal=True)
    qs = q.permute(0, 2, 1, 3)
    ks = k.permute(0, 2, 1, 3).repeat_interleave(2, dim=1)
    vs = v.permute(0, 2, 1, 3).repeat_interleave(2, dim=1)
    expect = torch.nn.functional.scaled_dot_product_attention(
        qs, ks, vs, is_causal=True).permute(0, 2, 1, 3)
    assert torch.allclose(out, expect, atol=1e-4, rtol=1e-4)


def test_attention_autograd_runs():
    B, S, Hq, Hkv, D = 1, 64, 2, 1, 128
    q = torch.randn(B, S, Hq, D, requires_grad=True)
    k = torch.randn(B, S, Hkv, D, requires_grad=True)
    v = torch.randn(B, S, Hkv, D, requires_grad=True)
    out = ops.attention(q, k, v)
    out.sum().backward()
    assert q.grad is not None and k.grad is not None and v.grad is not None


def test_cross_entropy_matches_torch():
    torch.manual_seed(5)
    logits = torch.randn(16, 512, requires_grad=True)
    targets = torch.randint(0, 512, (16,))
    loss = ops.fused_cross_entropy(logits, targets.int())
    expect = torch.nn.functional.cross_entropy(logits.detach(),
                                               targets.long())
    assert torch.allclose(loss, expect, atol=1e-5)
    loss.backward()
    assert logits.grad is not None


def test_cross_entropy_ignore_index():
    logits = torch.randn(8, 64)
    targets = torch.randint(0, 64, (8,))
    targets[3] = -100
    loss = ops.fused_cross_entropy(logits, targets.int())
    expect = torch.nn.functional.cross_entropy(logits, targets.long(),
                                               ignore_index=-100)
    assert torch.allclose(loss, expect, atol=1e-5)


def test_native_extension_loads():
    # The .so is cross-compiled for gfx950 in this container; it must at
    # least load (GPU calls are tested under -m gpu).
    assert ops.native_available(), "in-tree _C.so missing or unloadable"


def test_swiglu_cpu_matches_manual():
    torch.manual_seed(9)
    gu = torch.randn(4, 64, requires_grad=True)
    y = ops.swiglu(gu)
    g, u = gu.detach().split(32, dim=-1)
    expect = torch.nn.functional.silu(g) * u
    assert torch.allclose(y, expect, atol=1e-5)
    y.sum().backward()
    assert gu.grad is not None


def test_all_example_yamls_parse():
    """Every shipped example must parse into a valid Task (or pipeline)."""
    from pathlib import Path

    import yaml as _yaml

    from skypilot_amd.task import Task
    ex = Path(__file__).parent.parent / "examples"
    files = sorted(ex.glob("*.yaml"))
    assert len(files) >= 7
    for f in files:
        cfg = _yaml.safe_load(f.read_text())
        if "tasks" in cfg:  # pipeline / job-group form
            for sub in cfg["tasks"]:
                t = Task.from_yaml_config(sub)
                assert t.run, f
            continue
        t = Task.from_yaml_config(cfg)
        assert t.run or t.service, f


def test_task_yaml_round_trip():
    from skypilot_amd.task import Task
    cfg = {
        "name": "rt", "num_nodes": 2,
        "resources": {"accelerators": "MI355X:4", "cpus": "8+",
                      "job_recovery": {"strategy": "FAILOVER",
                                       "max_restarts_on_errors": 2},
                      "labels": {"team": "ml"}},
        "envs": {"A": "1"},
        "run": "echo hi",
    }
    t = Task.from_yaml_config(cfg)
    out = t.to_yaml_config()
    t2 = Task.from_yaml_config(out)
    assert t2.num_nodes == 2
    assert t2.resources.accelerator_count == 4
    assert t2.resources.cpus == 8 and t2.resources.cpus_is_min
    assert t2.resources.job_recovery.max_restarts_on_errors == 2
    assert t2.envs == {"A": "1"}
