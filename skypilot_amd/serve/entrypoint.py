"""OpenAI-compatible inference entrypoint for the bundled Llama models.

This is the `run:` command of the shipped serving task YAML:

    python -m skypilot_amd.serve.entrypoint --model llama3-8b --port $PORT

Endpoints: /health, /v1/models, /v1/completions, /v1/chat/completions,
/stats.  With no network on the pool there are no pretrained weights or
tokenizer files; the server runs random-init weights with a byte-level
tokenizer (data: synthetic) — the serving bench measures engine/kernel
throughput, not sample quality.  A checkpoint dir produced by the
bundled trainer can be loaded with --checkpoint-dir.
"""
from __future__ import annotations

import argparse
import os
import time
from typing import List, Optional

import uvicorn
from fastapi import FastAPI
from pydantic import BaseModel

from skypilot_amd.serve.engine import Engine, Request


class ByteTokenizer:
    """Byte-level fallback tokenizer (ids 0-255 + BOS=256, EOS=257)."""
    BOS = 256
    EOS = 257

    def encode(self, text: str) -> List[int]:
        return [self.BOS] + list(text.encode("utf-8", errors="replace"))

    def decode(self, ids: List[int]) -> str:
        return bytes(i for i in ids if 0 <= i < 256).decode(
            "utf-8", errors="replace")


class CompletionRequest(BaseModel):
    model: str = ""
    prompt: str = ""
    max_tokens: int = 64
    temperature: float = 0.0
    top_p: float = 1.0
    stop: Optional[List[str]] = None
    stream: bool = False


class ChatMessage(BaseModel):
    role: str
    content: str


class ChatRequest(BaseModel):
    model: str = ""
    messages: List[ChatMessage] = []
    max_tokens: int = 64
    temperature: float = 0.0


def create_app(engine: Engine, model_name: str) -> FastAPI:
    app = FastAPI(title="skypilot-amd inference")
    tok = ByteTokenizer()

    @app.get("/health")
    def health():
        return {"ok": True, "model": model_name,
                "max_batch": engine.max_batch}

    @app.get("/stats")
    def stats():
        return {**engine.stats, "active": len(engine.active),
                "free_slots": len(engine.free_slots)}

    @app.get("/v1/models")
    def models():
        return {"object": "list",
                "data": [{"id": model_name, "object": "model"}]}

    def _complete(prompt: str, max_tokens: int, temperature: float):
        ids = tok.encode(prompt)
        t0 = time.time()
        out = engine.generate(ids, max_tokens=max_tokens,
                              temperature=temperature)
        return out, time.time() - t0, len(ids)

    @app.post("/v1/completions")
    def completions(req: CompletionRequest):
        if req.stream:
            import json as _json
            import queue as _queue

            from fastapi.responses import StreamingResponse
            from skypilot_amd.serve.engine import Request as _Req
            q = _queue.Queue()
            r = _Req(prompt_ids=tok.encode(req.prompt),
                     max_tokens=req.max_tokens,
                     temperature=req.temperature, top_p=req.top_p,
                     stream_queue=q)
            engine.submit(r)

            def sse():
                while True:
                    item = q.get(timeout=600)
                    if item is None:
                        yield "data: [DONE]\n\n"
                        return
                    chunk = {"id": "cmpl-local",
                             "object": "text_completion",
                             "model": model_name,
                             "choices": [{"index": 0,
                                          "text": tok.decode([item]),
                                          "finish_reason": None}]}
                    yield f"data: {_json.dumps(chunk)}\n\n"

            return StreamingResponse(sse(), media_type="text/event-stream")
        out, dt, n_prompt = _complete(req.prompt, req.max_tokens,
                                      req.temperature)
        text = tok.decode(out)
        finish = "length"
        for stop in req.stop or []:
            i = text.find(stop)
            if i >= 0:
                text = text[:i]
                finish = "stop"
                break
        return {
            "id": "cmpl-local", "object": "text_completion",
            "model": model_name,
            "choices": [{"index": 0, "text": text,
                         "finish_reason": finish}],
            "usage": {"prompt_tokens": n_prompt,
                      "completion_tokens": len(out),
                      "total_tokens": n_prompt + len(out),
                      "latency_s": dt},
        }

    @app.post("/v1/chat/completions")
    def chat(req: ChatRequest):
        prompt = "\n".join(f"{m.role}: {m.content}" for m in req.messages)
        out, dt, n_prompt = _complete(prompt, req.max_tokens,
                                      req.temperature)
        return {
            "id": "chatcmpl-local", "object": "chat.completion",
            "model": model_name,
            "choices": [{"index": 0,
                         "message": {"role": "assistant",
                                     "content": tok.decode(out)},
                         "finish_reason": "length"}],
            "usage": {"prompt_tokens": n_prompt,
                      "completion_tokens": len(out),
                      "total_tokens": n_prompt + len(out),
                      "latency_s": dt},
        }

    return app


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama3-8b")
    ap.add_argument("--port", type=int,
                    default=int(os.environ.get("PORT", 8000)))
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--max-seq", type=int, default=4096)
    ap.add_argument("--max-batch", type=int, default=None)
    ap.add_argument("--device", default=None)
    ap.add_argument("--fp8-decode", action="store_true",
                    help="quantize weights to rowwise e4m3fn for the "
                         "decode GEMVs (~2x single-stream decode)")
    ap.add_argument("--tp", type=int, default=1,
                    help="tensor-parallel degree; run under torchrun "
                         "with one rank per GPU (70B serving: --tp 8)")
    args = ap.parse_args()
    if args.tp > 1:
        # TP serving (BASELINE config 5's serve twin): every rank builds
        # its shard; rank 0 leads (scheduler + HTTP), others follow the
        # broadcast step plan.  Launched by the gang driver as
        # torchrun --nproc-per-node TP ... --tp TP.
        import torch
        import torch.distributed as dist
        from skypilot_amd.parallel.tp import build_tp_model
        rank = int(os.environ.get("RANK", "0"))
        backend = "nccl" if torch.cuda.is_available() else "gloo"
        if not dist.is_initialized():
            dist.init_process_group(backend)
        if torch.cuda.is_available():
            torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)))
        device = args.device or (
            f"cuda:{int(os.environ.get('LOCAL_RANK', rank))}"
            if torch.cuda.is_available() else "cpu")
        shard = build_tp_model(args.model, tp=args.tp, rank=rank,
                               device=device)
        engine = Engine(args.model, device=device, max_seq=args.max_seq,
                        max_batch=args.max_batch, model=shard,
                        tp_rank=rank, tp_world=args.tp)
        if args.fp8_decode:
            engine.enable_fp8_decode()
        if rank > 0:
            engine.follower_loop()
            return
        engine.start()
        app = create_app(engine, args.model)
        print(f"serving {args.model} tp={args.tp} on "
              f"{args.host}:{args.port} (max_batch={engine.max_batch})",
              flush=True)
        uvicorn.run(app, host=args.host, port=args.port,
                    log_level="warning")
        return
    engine = Engine(args.model, device=args.device, max_seq=args.max_seq,
                    max_batch=args.max_batch)
    if args.fp8_decode:
        n8 = engine.enable_fp8_decode()
        print(f"fp8 decode weights: {n8} tensors quantized", flush=True)
    engine.start()
    app = create_app(engine, args.model)
    print(f"serving {args.model} on {args.host}:{args.port} "
          f"(max_batch={engine.max_batch})", flush=True)
    uvicorn.run(app, host=args.host, port=args.port, log_level="warning")


if __name__ == "__main__":
    main()
