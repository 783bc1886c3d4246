#!/usr/bin/env python3
"""Serving throughput bench: continuous-batching decode tokens/s for the
bundled Llama engine on one MI355X (the per-replica number behind
BASELINE config 4)."""
from __future__ import annotations

import argparse
import threading
import time

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama3-8b")
    ap.add_argument("--prompt-len", type=int, default=128)
    ap.add_argument("--gen-len", type=int, default=128)
    ap.add_argument("--batches", default="1,8,32,64")
    ap.add_argument("--max-batch", type=int, default=64)
    args = ap.parse_args()

    from skypilot_amd.serve.engine import Engine
    eng = Engine(args.model,
                 device="cuda:0" if torch.cuda.is_available() else "cpu",
                 max_seq=4096, max_batch=args.max_batch)
    eng.start()
    prompt = list(range(2, 2 + args.prompt_len))
    # warmup
    eng.generate(prompt, max_tokens=8)

    for nb in [int(x) for x in args.batches.split(",")]:
        results = []
        t0 = time.perf_counter()

        def worker():
            out = eng.generate(prompt, max_tokens=args.gen_len,
                               timeout=1200)
            results.append(len(out))

        threads = [threading.Thread(target=worker) for _ in range(nb)]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        dt = time.perf_counter() - t0
        toks = sum(results)
        print(f"concurrency {nb:3d}: {toks} tokens in {dt:6.2f}s = "
              f"{toks/dt:8.1f} tok/s decode "
              f"(ttft incl. prefill; prompt {args.prompt_len})",
              flush=True)
    print(f"graph buckets captured: {eng.stats['graph_buckets']} "
          f"(use_graphs={eng.use_graphs})")
    eng.stop()


if __name__ == "__main__":
    main()
