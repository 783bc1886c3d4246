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
    ap.add_argument("--fp8", action="store_true",
                    help="rowwise e4m3fn decode weights")
    args = ap.parse_args()

    from skypilot_amd.serve.engine import Engine
    eng = Engine(args.model,
                 device="cuda:0" if torch.cuda.is_available() else "cpu",
                 max_seq=4096, max_batch=args.max_batch)
    if args.fp8:
        print(f"fp8 decode weights: {eng.enable_fp8_decode()} tensors")
    eng.start()
    prompt = list(range(2, 2 + args.prompt_len))
    # warmup
    eng.generate(prompt, max_tokens=8)

    from skypilot_amd.serve.engine import Request
    for nb in [int(x) for x in args.batches.split(",")]:
        results = []
        ttfts = []
        t0 = time.perf_counter()

        def worker():
            req = eng.submit(Request(prompt_ids=list(prompt),
                                     max_tokens=args.gen_len))
            req.done.wait(1200)
            results.append(len(req.out_ids))
            if req.first_token_at:
                ttfts.append(req.first_token_at - req.created)

        threads = [threading.Thread(target=worker) for _ in range(nb)]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        dt = time.perf_counter() - t0
        toks = sum(results)
        ttfts.sort()
        p50 = ttfts[len(ttfts) // 2] * 1e3 if ttfts else 0
        p95 = ttfts[int(len(ttfts) * 0.95)] * 1e3 if ttfts else 0
        print(f"concurrency {nb:3d}: {toks} tokens in {dt:6.2f}s = "
              f"{toks/dt:8.1f} tok/s decode | ttft p50 {p50:6.1f} ms "
              f"p95 {p95:6.1f} ms (prompt {args.prompt_len})",
              flush=True)
    print(f"graph buckets captured: {eng.stats['graph_buckets']} "
          f"(use_graphs={eng.use_graphs})")
    eng.stop()


if __name__ == "__main__":
    main()
