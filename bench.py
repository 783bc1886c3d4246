#!/usr/bin/env python3
"""Headline benchmark: sky-launched Llama-3-8B bf16 training throughput.

Contract (driver-facing):
  python bench.py --gpus N --steps K --warmup W
N>1 is launched by the driver via torch.distributed.run (one rank per
GPU over RCCL); each rank then runs the training loop directly.  When
invoked WITHOUT torchrun (the driver's N=1 case), the benchmark goes
through the FULL orchestrator stack — ``execution.launch`` provisions
the local pool, the node agent queues the job, the gang driver spawns
the training process — and the measured JSON is parsed back from the
job log, so the claimed metric ("sky-launched") is what is actually
measured (this round's fix; round 1 constructed the Trainer directly).

W untimed warmup steps, then exactly K timed steps bracketed by
barrier + torch.cuda.synchronize on both sides; elapsed time is MAX over
ranks; rank 0 prints one JSON line.

Metric: tokens/sec aggregated over the whole job (weak scaling: per-GPU
work fixed at micro_batch x seq_len as N grows).  BASELINE.md publishes
no reference number for this metric (SkyPilot is an orchestrator), so
vs_baseline is null.  Job-start latency (launch call -> job RUNNING) is
reported alongside in the config (reference instrumentation model:
sky/utils/timeline.py + cluster_events, global_user_state.py:1001).
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time
from pathlib import Path

REPO_ROOT = Path(__file__).resolve().parent


def parse_args():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=8)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--model", default="llama3-8b")
    ap.add_argument("--micro-batch", type=int, default=6)
    ap.add_argument("--seq-len", type=int, default=4096)
    ap.add_argument("--direct", action="store_true",
                    help="run the training loop in this process (the "
                         "per-rank inner entrypoint; skips the "
                         "orchestrator)")
    return ap.parse_args()


def run_direct(args, extra_config=None) -> int:
    """Per-rank training loop: the timed region of the benchmark."""
    import torch
    import torch.distributed as dist

    on_gpu = torch.cuda.is_available()
    if not on_gpu and os.environ.get("SKY_BENCH_ALLOW_CPU") != "1":
        # Test hook only: SKY_BENCH_ALLOW_CPU=1 runs the same code path
        # on CPU (tiny model) so the sky-launched round trip is testable
        # without a GPU.  Real benchmark numbers always come from GPUs.
        print(json.dumps({"error": "no GPU available"}))
        return 1

    from skypilot_amd.train.trainer import TrainConfig, Trainer, \
        setup_distributed

    rank, world, local_rank = setup_distributed()
    if world == 1 and args.gpus > 1:
        print("error: --gpus > 1 requires torchrun (one rank per GPU)",
              file=sys.stderr)
        return 2

    cfg = TrainConfig(model=args.model, micro_batch=args.micro_batch,
                      seq_len=args.seq_len,
                      device="cuda" if on_gpu else "cpu")
    tr = Trainer(cfg)

    def barrier_sync():
        if world > 1:
            dist.barrier()
        if on_gpu:
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        tr.train_step()
    barrier_sync()
    t0 = time.perf_counter()
    loss = 0.0
    for _ in range(args.steps):
        loss = tr.train_step()
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # Max over ranks.
    if world > 1:
        t = torch.tensor([elapsed], device=tr.device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    if on_gpu:
        peak_gb = torch.cuda.max_memory_allocated() / 1e9
        print(f"[bench] rank {rank}: peak GPU memory {peak_gb:.1f} GB",
              file=sys.stderr, flush=True)

    tokens = args.steps * cfg.micro_batch * cfg.seq_len * world
    value = tokens / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if rank == 0:
        config = {
            "model": args.model,
            "global_batch": cfg.micro_batch * world,
            "seq_len": cfg.seq_len,
            "parallelism": f"dp{world}",
            "final_loss": loss,
        }
        if extra_config:
            config.update(extra_config)
        result = {
            "metric": "tokens/sec (node) sky-launched Llama-3-8B",
            "value": value,
            "unit": "tokens/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "config": config,
        }
        print(json.dumps(result), flush=True)
    if world > 1:
        dist.destroy_process_group()
    return 0


def run_sky_launched(args) -> int:
    """The full stack: provision -> agent -> gang driver -> training.

    Launches this script's --direct inner path as a sky task on the
    local MI355X pool, waits for the job, parses the measured JSON from
    the job log and re-prints it augmented with job-start latency.
    """
    from skypilot_amd import execution, global_state
    from skypilot_amd.agent.client import AgentClient
    from skypilot_amd.backends.pool_backend import PoolBackend
    from skypilot_amd.task import Task

    inner = (f"python {REPO_ROOT}/bench.py --direct --gpus {args.gpus} "
             f"--steps {args.steps} --warmup {args.warmup} "
             f"--model {args.model} --micro-batch {args.micro_batch} "
             f"--seq-len {args.seq_len}")
    if args.gpus > 1:
        inner = (f"torchrun --standalone --master-addr 127.0.0.1 "
                 f"--nproc-per-node {args.gpus} {REPO_ROOT}/bench.py "
                 f"--direct --gpus {args.gpus} --steps {args.steps} "
                 f"--warmup {args.warmup} --model {args.model} "
                 f"--micro-batch {args.micro_batch} "
                 f"--seq-len {args.seq_len}")
    envs = {"PYTHONPATH": str(REPO_ROOT)}
    if os.environ.get("SKY_BENCH_ALLOW_CPU") == "1":
        envs["SKY_BENCH_ALLOW_CPU"] = "1"
    task = Task.from_yaml_config({
        "name": "bench-train",
        "run": inner,
        "envs": envs,
        "resources": {"accelerators": f"MI355X:{args.gpus}"},
    })
    cluster = "bench-cluster"
    backend = PoolBackend()
    t_launch = time.time()
    job_id, handle = execution.launch(task, cluster, detach_run=True)
    agent = AgentClient(handle["agent_port"],
                        token=handle.get("agent_token"))

    # Job-start latency: launch() call -> agent marks the job RUNNING.
    started_at = None
    deadline = time.time() + 600
    while time.time() < deadline:
        j = agent.get_job(job_id)
        if j and j.get("started_at"):
            started_at = j["started_at"]
            break
        time.sleep(0.2)
    job_start_latency = (started_at - t_launch) if started_at else None

    final = backend.wait_job(handle, job_id, timeout=3600)
    log_dir = Path(agent.get_job(job_id)["log_dir"])
    text = ""
    for name in ("run.log", "0-node.log"):
        p = log_dir / name
        if p.exists():
            text = p.read_text(errors="replace")
            break

    result = None
    for line in reversed(text.splitlines()):
        if line.startswith('{"metric"'):
            try:
                result = json.loads(line)
                break
            except ValueError:
                continue

    # Teardown so back-to-back runs don't leak agents.
    try:
        backend.teardown(handle, terminate=True)
    except Exception as e:  # noqa: BLE001 — teardown is best-effort here
        print(f"[bench] teardown failed: {e}", file=sys.stderr)

    status = final.get("status") if isinstance(final, dict) else final
    if result is None:
        print(f"[bench] sky-launched job ended {status}; no metric JSON "
              f"in {log_dir}; log tail:\n{text[-2000:]}", file=sys.stderr)
        return 1
    result["config"]["sky_launched"] = True
    result["config"]["job_start_latency_s"] = (
        round(job_start_latency, 3) if job_start_latency is not None
        else None)
    print(json.dumps(result), flush=True)
    return 0


def main() -> int:
    args = parse_args()
    under_torchrun = "TORCHELASTIC_RUN_ID" in os.environ or (
        "RANK" in os.environ and "WORLD_SIZE" in os.environ)
    if args.direct or under_torchrun:
        return run_direct(args)
    try:
        return run_sky_launched(args)
    except Exception as e:  # noqa: BLE001
        # Orchestration failed — fall back to the direct trainer so a
        # measurement still exists, and say so honestly in the config.
        import traceback
        traceback.print_exc()
        print(f"[bench] sky-launched path failed ({e}); falling back to "
              "direct trainer (config.sky_launched=false)",
              file=sys.stderr)
        try:
            # Release the GPU before the direct run: the failed sky
            # attempt may have left the cluster (and its job) alive.
            from skypilot_amd import global_state
            from skypilot_amd.backends.pool_backend import PoolBackend
            rec = global_state.get_cluster("bench-cluster")
            if rec:
                PoolBackend().teardown(rec["handle"], terminate=True)
                time.sleep(3.0)
        except Exception:  # noqa: BLE001
            pass
        return run_direct(args, extra_config={"sky_launched": False})


if __name__ == "__main__":
    sys.exit(main())
