"""Inference engine — continuous batching over a slot-based KV cache.

MI355X-native serving core for the bundled OpenAI-compatible entrypoint
(no reference counterpart; SkyPilot points users at vLLM images,
SURVEY.md §2.11).  Prefill runs the causal flash kernel (prompt padded
to the 64-row tile); decode steps batch every active sequence into one
forward through the decode-attention kernel.  max_batch is sized to the
288 GB HBM budget (KVCache.sized_for_memory).
"""
from __future__ import annotations

import os
import queue
import threading
import time
import uuid
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from skypilot_amd import ops
from skypilot_amd.models.llama import build_model
from skypilot_amd.serve.kv_cache import KVCache


@dataclass
class InferenceContext:
    cache: KVCache
    mode: str  # "prefill" | "decode"
    prefill_slot: int = -1
    prefill_len: int = 0
    # batched prefill: one row per new sequence (all rows share
    # positions 0..pad; rows beyond a sequence's true length are
    # ignored at sampling and never written to the cache)
    prefill_slots: Optional[list] = None
    prefill_lens: Optional[list] = None
    slots: Optional[torch.Tensor] = None        # int64 [n] (decode)
    pos: Optional[torch.Tensor] = None          # int64 [n] write positions
    kv_lens: Optional[torch.Tensor] = None      # int32 [n] incl. new token
    slot_ids_i32: Optional[torch.Tensor] = None  # int32 [n]


@dataclass
class Request:
    prompt_ids: List[int]
    max_tokens: int = 64
    temperature: float = 0.0
    top_p: float = 1.0
    stop_id: Optional[int] = None
    stream_queue: Optional["queue.Queue"] = None  # per-token streaming
    id: str = field(default_factory=lambda: uuid.uuid4().hex[:12])
    out_ids: List[int] = field(default_factory=list)
    done: threading.Event = field(default_factory=threading.Event)
    created: float = field(default_factory=time.time)
    first_token_at: Optional[float] = None
    finished_at: Optional[float] = None


class Engine:
    def __init__(self, model_name: str, device: Optional[str] = None,
                 max_seq: int = 4096, max_batch: Optional[int] = None,
                 hbm_budget_gb: Optional[float] = None,
                 use_graphs: bool = True, model=None, tp_group=None,
                 tp_rank: int = 0, tp_world: int = 1):
        self.device = torch.device(device or (
            "cuda" if torch.cuda.is_available() else "cpu"))
        dtype = torch.bfloat16
        # Tensor-parallel serving (BASELINE config 5's serve twin): a
        # pre-built TP shard is passed in; rank 0 leads (scheduler +
        # HTTP), ranks > 0 mirror its forwards via broadcast step plans
        # (see follower_loop).  All ranks hold identical activations
        # after each block's all-reduce, so token selection needs no
        # extra communication.
        self.tp_group = tp_group
        self.tp_rank = tp_rank
        # TP participation is EXPLICIT (tp_world > 1): a process may be
        # part of a distributed group for other reasons (e.g. a non-TP
        # engine built alongside) and must not broadcast step plans.
        self.tp_world = tp_world
        if model is not None:
            self.model = model
        else:
            self.model = build_model(model_name, device=str(self.device),
                                     dtype=dtype)
        self.model.eval()
        self.cfg = getattr(self.model, "cfg_shard", self.model.cfg)
        self.max_seq = min(max_seq, self.cfg.max_seq_len)
        if max_batch is None:
            if hbm_budget_gb is None:
                if self.device.type == "cuda":
                    free, total = torch.cuda.mem_get_info(self.device)
                    hbm_budget_gb = (free / 1e9) * 0.8
                else:
                    hbm_budget_gb = 0.5
            self.cache = KVCache.sized_for_memory(
                self.cfg, self.max_seq, int(hbm_budget_gb * 1e9),
                self.device)
        else:
            # +1 slot when graphs are on: the graph-padding scratch slot
            # must not cost a unit of serving concurrency.
            extra = 1 if (use_graphs and self.device.type == "cuda") else 0
            self.cache = KVCache(self.cfg, max_batch + extra, self.max_seq,
                                 self.device)
        self.max_batch = self.cache.max_batch
        self.free_slots = list(range(self.max_batch))
        self.active: Dict[int, Request] = {}  # slot -> request
        self.pending: "queue.Queue[Request]" = queue.Queue()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.stats = {"requests": 0, "tokens_generated": 0,
                      "prefill_tokens": 0, "graph_buckets": []}
        # hipGraph decode: the whole decode forward (~7 kernels/layer +
        # head = ~230 launches for 8B) replays as ONE graph launch per
        # power-of-two batch bucket.  Low-concurrency decode is
        # launch-bound, so this is the serving latency lever.  Padded
        # rows point at a reserved scratch slot so their K/V writes
        # never touch a live sequence.
        self.use_graphs = (use_graphs and self.device.type == "cuda"
                           and self.max_batch >= 2)
        self._graphs: Dict[int, tuple] = {}
        self._scratch_slot = -1
        if self.use_graphs:
            self._scratch_slot = self.free_slots.pop()
            self.max_batch -= 1

    def enable_fp8_decode(self, min_bytes: int = 1 << 20) -> int:
        """Quantize every large linear weight to row-wise e4m3fn and
        route the decode GEMVs through the fp8 kernel (half the streamed
        bytes on the weight-bandwidth-bound decode path; prefill stays
        bf16).  Returns the number of weights registered."""
        from skypilot_amd import ops as _ops
        n = 0
        for name, p in self.model.named_parameters():
            if (p.ndim == 2 and p.dtype == torch.bfloat16
                    and p.numel() * 2 >= min_bytes
                    and p.shape[1] % 1024 == 0
                    and "embed" not in name):
                _ops.register_fp8_weight(p.data)
                n += 1
        # The decode path lazily packs [q|k|v] into a single GEMV weight
        # (models/llama.py _wqkv) that is NOT a named parameter —
        # materialize and register it here too (it is ~10% of the
        # streamed decode bytes).
        for blk in getattr(self.model, "blocks", []):
            a = getattr(blk, "attn", None)
            if a is None or not hasattr(a, "wq"):
                continue
            if getattr(a, "_wqkv", None) is None and a.wq.weight.is_cuda:
                a._wqkv = torch.cat(
                    [a.wq.weight, a.wk.weight, a.wv.weight], dim=0)
            if (getattr(a, "_wqkv", None) is not None
                    and a._wqkv.shape[1] % 1024 == 0):
                _ops.register_fp8_weight(a._wqkv)
                n += 1
        return n

    # ------------------------------------------------------------------
    def submit(self, req: Request) -> Request:
        if len(req.prompt_ids) >= self.max_seq:
            req.prompt_ids = req.prompt_ids[-(self.max_seq - 1 -
                                              req.max_tokens):]
        self.stats["requests"] += 1
        self.pending.put(req)
        return req

    def generate(self, prompt_ids: List[int], max_tokens: int = 64,
                 temperature: float = 0.0,
                 timeout: float = 300.0) -> List[int]:
        req = self.submit(Request(prompt_ids=prompt_ids,
                                  max_tokens=max_tokens,
                                  temperature=temperature))
        if not req.done.wait(timeout):
            raise TimeoutError("generation timed out")
        return req.out_ids

    def start(self):
        if self._thread is None:
            self._thread = threading.Thread(target=self._loop, daemon=True)
            self._thread.start()

    def stop(self):
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=10)
            self._thread = None
        if self.tp_world > 1 and self.tp_rank == 0:
            self._tp_bcast({"op": "stop"})

    # ------------------------------------------------------------------
    @torch.no_grad()
    def _prefill(self, reqs: List[Request]) -> None:
        """Batched prefill: all admitted prompts run as ONE padded
        causal forward (serial per-prompt prefills dominated the ramp
        at high concurrency: 64 x ~9 ms before the first decode)."""
        n = len(reqs)
        slots = [self.free_slots.pop() for _ in range(n)]
        lens = [len(r.prompt_ids) for r in reqs]
        self._tp_bcast({"op": "prefill", "slots": slots, "lens": lens,
                        "prompts": [r.prompt_ids for r in reqs]})
        pad = (max(lens) + 63) // 64 * 64
        toks = torch.zeros(n, pad, dtype=torch.long, device=self.device)
        for i, r in enumerate(reqs):
            toks[i, :lens[i]] = torch.tensor(r.prompt_ids,
                                             device=self.device)
        positions = torch.arange(pad, dtype=torch.int32,
                                 device=self.device).repeat(n)  # [n*pad]
        ctx = InferenceContext(cache=self.cache, mode="prefill",
                               prefill_slots=slots, prefill_lens=lens)
        logits = self.model(toks, positions, ctx)
        last = logits[torch.arange(n), torch.tensor(lens) - 1]  # [n, V]
        for i, (req, slot) in enumerate(zip(reqs, slots)):
            next_id = self._sample(last[i], req.temperature, req.top_p)
            self.cache.lens[slot] = lens[i]
            req.out_ids.append(next_id)
            if req.stream_queue is not None:
                req.stream_queue.put(next_id)
            req.first_token_at = time.time()
            self.active[slot] = req
            self.stats["prefill_tokens"] += lens[i]
            self._maybe_finish(slot, req, next_id)

    # -------------------------- hipGraph decode ------------------------
    # Max chained graph replays between host syncs (SKY_DECODE_CHUNK to
    # tune; measured flat 8..32 at 1 stream, so 8 keeps token-streaming
    # latency low).
    CHUNK = int(os.environ.get("SKY_DECODE_CHUNK", "8"))

    def _bucket(self, n: int) -> int:
        b = 1
        while b < n:
            b <<= 1
        return b

    def _get_graph(self, b: int):
        """Lazily capture the decode forward for batch bucket `b`."""
        if b in self._graphs:
            return self._graphs[b]
        dev = self.device
        # All six per-step inputs travel as ONE pinned-host -> device
        # copy of an int32 [6, b] block; the unpack (casts/views) is
        # captured inside the graph.  Rows: 0 toks, 1 positions,
        # 2 slots, 3 pos, 4 kv_lens, 5 slot_i32.
        st = {
            "host": torch.empty(6, b, dtype=torch.int32,
                                pin_memory=True),
            "stage": torch.zeros(6, b, dtype=torch.int32, device=dev),
        }
        st["stage"][2] = self._scratch_slot
        st["stage"][5] = self._scratch_slot
        st["stage"][4] = 1

        def unpack():
            stg = st["stage"]
            return (stg[0].long().view(b, 1), stg[1],
                    InferenceContext(
                        cache=self.cache, mode="decode",
                        slots=stg[2].long(), pos=stg[3].long(),
                        kv_lens=stg[4], slot_ids_i32=stg[5]))

        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(2):  # warmup on a side stream (autotune etc.)
                toks, positions, ctx = unpack()
                self.model(toks, positions, ctx)
        torch.cuda.current_stream().wait_stream(side)
        st["ring"] = torch.zeros(self.CHUNK, b, dtype=torch.long,
                                 device=dev)
        st["ctr"] = torch.zeros(1, dtype=torch.long, device=dev)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            toks, positions, ctx = unpack()
            st["out"] = self.model(toks, positions, ctx)
            # Self-advancing state: the fused advance kernel does the
            # argmax AND feeds it back into the token stage + bumps
            # positions/pos/kv_lens + stores the token in the ring, so
            # greedy decode replays CHUNK times with zero host work in
            # between (torch's argmax alone measured 43 us/step —
            # profiles/r02_fp8_decode_kernel_stats.txt).  The ctr
            # increment stays a separate captured op so the kernel's
            # ring-row read is ordered before it.
            ops.decode_advance(st["out"][:, 0], st["stage"], st["ring"],
                               st["ctr"])
            st["ctr"] += 1
        self._graphs[b] = (g, st)
        self.stats["graph_buckets"] = sorted(self._graphs)
        return self._graphs[b]

    @torch.no_grad()
    def _decode_step(self) -> None:
        slots = sorted(self.active.keys())
        if not slots:
            return
        reqs = [self.active[s] for s in slots]
        n = len(slots)
        lens = [self.cache.lens[s] for s in slots]
        tok_l = [r.out_ids[-1] for r in reqs]
        if self.use_graphs:
            b = self._bucket(n)
            all_greedy = all(r.temperature == 0 for r in reqs)
            chunk = 1
            if all_greedy and self.pending.empty():
                rem = min(r.max_tokens - len(r.out_ids) for r in reqs)
                cap = min(self.max_seq - 2 - l for l in lens)
                chunk = max(1, min(self.CHUNK, rem, cap))
            # TP: the plan must go out BEFORE capture — _get_graph's
            # warmup forwards contain all-reduces that need every rank
            # participating (the follower captures on receipt).
            self._tp_bcast({"op": "decode", "graph": True, "bucket": b,
                            "slots": slots, "lens": lens, "toks": tok_l,
                            "chunk": chunk})
            try:
                g, st = self._get_graph(b)
            except Exception as e:  # capture unsupported -> eager forever
                if self.tp_world > 1:
                    raise  # followers already committed to the graph plan
                import sys
                print(f"[engine] hipGraph capture failed, falling back "
                      f"to eager decode: {e!r}", file=sys.stderr)
                self.use_graphs = False
                return self._decode_step()
            scr = self._scratch_slot
            h = st["host"].numpy()
            h[0, :n] = tok_l
            h[0, n:] = 0
            h[1, :n] = lens
            h[1, n:] = 0
            h[2, :n] = slots
            h[2, n:] = scr
            h[3] = h[1]
            h[4] = h[1] + 1
            h[5] = h[2]
            st["ctr"].zero_()
            st["stage"].copy_(st["host"], non_blocking=True)
            for _ in range(chunk):
                g.replay()
            if all_greedy:
                ring = st["ring"][:chunk, :n].tolist()  # one sync/chunk
                for t in range(chunk):
                    for i, slot in enumerate(slots):
                        if slot not in self.active:  # finished earlier t
                            continue
                        req = self.active[slot]
                        next_id = ring[t][i]
                        self.cache.lens[slot] += 1
                        req.out_ids.append(next_id)
                        if req.stream_queue is not None:
                            req.stream_queue.put(next_id)
                        self.stats["tokens_generated"] += 1
                        self._maybe_finish(slot, req, next_id)
                return
            logits = st["out"]  # [b, 1, V]
        else:
            self._tp_bcast({"op": "decode", "graph": False,
                            "slots": slots, "lens": lens, "toks": tok_l,
                            "chunk": 1})
            toks = torch.tensor([[t] for t in tok_l], dtype=torch.long,
                                device=self.device)
            slots_t = torch.tensor(slots, dtype=torch.long,
                                   device=self.device)
            pos_t = torch.tensor(lens, dtype=torch.long,
                                 device=self.device)
            positions = torch.tensor(lens, dtype=torch.int32,
                                     device=self.device)
            kv_lens = torch.tensor([l + 1 for l in lens],
                                   dtype=torch.int32, device=self.device)
            ctx = InferenceContext(
                cache=self.cache, mode="decode", slots=slots_t, pos=pos_t,
                kv_lens=kv_lens, slot_ids_i32=slots_t.int())
            logits = self.model(toks, positions, ctx)  # [n, 1, V]
        # Greedy rows batch into one argmax + one sync; sampled rows
        # (temperature > 0) go through _sample individually.  (The
        # all-greedy graph path already returned above.)
        greedy_ids = None
        if all(r.temperature == 0 for r in reqs):
            greedy_ids = logits[:n, 0].argmax(-1).tolist()
        for i, slot in enumerate(slots):
            self.cache.lens[slot] += 1
            req = self.active[slot]
            if greedy_ids is not None:
                next_id = greedy_ids[i]
            else:
                next_id = self._sample(logits[i, 0], req.temperature,
                                       req.top_p)
            req.out_ids.append(next_id)
            if req.stream_queue is not None:
                req.stream_queue.put(next_id)
            self.stats["tokens_generated"] += 1
            self._maybe_finish(slot, req, next_id)

    def _sample(self, logits: torch.Tensor, temperature: float,
                top_p: float = 1.0) -> int:
        if temperature and temperature > 0:
            probs = (logits.float() / temperature).softmax(-1)
            if top_p < 1.0:
                sp, idx = probs.sort(descending=True)
                keep = (sp.cumsum(0) - sp) <= top_p
                keep[0] = True
                probs = torch.zeros_like(probs).scatter(
                    0, idx[keep], sp[keep])
                probs = probs / probs.sum()
            return int(torch.multinomial(probs, 1).item())
        return int(logits.argmax().item())

    def _maybe_finish(self, slot: int, req: Request, last_id: int) -> None:
        if (len(req.out_ids) >= req.max_tokens or
                (req.stop_id is not None and last_id == req.stop_id) or
                self.cache.lens[slot] + 1 >= self.max_seq):
            del self.active[slot]
            self.cache.free(slot)
            self.free_slots.append(slot)
            req.finished_at = time.time()
            if req.stream_queue is not None:
                req.stream_queue.put(None)  # end-of-stream marker
            req.done.set()

    # ---------------------- tensor-parallel plumbing -------------------
    def _tp_bcast(self, obj) -> None:
        if self.tp_world > 1 and self.tp_rank == 0:
            dist.broadcast_object_list([obj], src=0, group=self.tp_group)

    def follower_loop(self) -> None:
        """TP rank > 0: execute the leader's broadcast step plan.  The
        follower keeps no scheduler state — every forward's slots/
        lengths/tokens arrive in the plan, and its KV cache stays
        consistent because it runs the identical deterministic
        forwards."""
        assert self.tp_world > 1 and self.tp_rank > 0
        while True:
            lst = [None]
            dist.broadcast_object_list(lst, src=0, group=self.tp_group)
            msg = lst[0]
            if msg["op"] == "stop":
                return
            if msg["op"] == "prefill":
                self._follow_prefill(msg)
            else:
                self._follow_decode(msg)

    @torch.no_grad()
    def _follow_prefill(self, msg) -> None:
        slots, lens = msg["slots"], msg["lens"]
        n = len(slots)
        pad = (max(lens) + 63) // 64 * 64
        toks = torch.zeros(n, pad, dtype=torch.long, device=self.device)
        for i, ids in enumerate(msg["prompts"]):
            toks[i, :lens[i]] = torch.tensor(ids, device=self.device)
        positions = torch.arange(pad, dtype=torch.int32,
                                 device=self.device).repeat(n)
        ctx = InferenceContext(cache=self.cache, mode="prefill",
                               prefill_slots=slots, prefill_lens=lens)
        self.model(toks, positions, ctx)

    @torch.no_grad()
    def _follow_decode(self, msg) -> None:
        slots, lens, tok_l = msg["slots"], msg["lens"], msg["toks"]
        n = len(slots)
        if msg["graph"]:
            g, st = self._get_graph(msg["bucket"])
            scr = self._scratch_slot
            h = st["host"].numpy()
            h[0, :n] = tok_l
            h[0, n:] = 0
            h[1, :n] = lens
            h[1, n:] = 0
            h[2, :n] = slots
            h[2, n:] = scr
            h[3] = h[1]
            h[4] = h[1] + 1
            h[5] = h[2]
            st["ctr"].zero_()
            st["stage"].copy_(st["host"], non_blocking=True)
            for _ in range(msg["chunk"]):
                g.replay()
            torch.cuda.synchronize()
            return
        toks = torch.tensor([[t] for t in tok_l], dtype=torch.long,
                            device=self.device)
        slots_t = torch.tensor(slots, dtype=torch.long, device=self.device)
        pos_t = torch.tensor(lens, dtype=torch.long, device=self.device)
        positions = torch.tensor(lens, dtype=torch.int32,
                                 device=self.device)
        kv_lens = torch.tensor([l + 1 for l in lens], dtype=torch.int32,
                               device=self.device)
        ctx = InferenceContext(cache=self.cache, mode="decode",
                               slots=slots_t, pos=pos_t, kv_lens=kv_lens,
                               slot_ids_i32=slots_t.int())
        self.model(toks, positions, ctx)

    def _loop(self):
        try:
            self._loop_inner()
        except Exception:
            # A fatal engine error must not leave TP followers blocked
            # in their broadcast receive forever.
            if self.tp_world > 1 and self.tp_rank == 0:
                try:
                    self._tp_bcast({"op": "stop"})
                except Exception:  # noqa: BLE001
                    pass
            raise

    def _loop_inner(self):
        while not self._stop.is_set():
            did = False
            # Admit pending requests while slots are free — batched
            # (up to 16 per prefill forward).
            batch: List[Request] = []
            while (self.free_slots and len(batch) < len(self.free_slots)
                   and len(batch) < 16 and not self.pending.empty()):
                try:
                    batch.append(self.pending.get_nowait())
                except queue.Empty:
                    break
            if batch:
                self._prefill(batch)
                did = True
            if self.active:
                self._decode_step()
                did = True
            if not did:
                time.sleep(0.005)
