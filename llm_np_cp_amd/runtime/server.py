"""HTTP serving layer with CONTINUOUS request batching.

The reference has no server (no CLI, no server); this is the
deployment-facing wrapper around the engine: OpenAI-style
``/v1/completions`` and ``/v1/chat/completions`` endpoints (both with
``"stream": true`` SSE token streaming) plus ``/health`` and ``/stats``.
Chat messages go through the tokenizer's chat template when the
checkpoint ships one, else a neutral ``<|role|>`` fallback template.

Concurrent requests are NOT serialized: a scheduler thread owns the
engine and runs a ragged lockstep group (per-sequence KV pools, per-row
device positions).  Between 16-token chunks, finished rows RETIRE (EOS
or per-request max_tokens; the last row compacts into the freed slot)
and queued compatible requests JOIN mid-flight (``prefill_row`` into a
free slot) — continuous batching, not fixed groups.  Requests with
CPU-only sampling strategies (top_k/top_p) or a non-batch engine fall
back to single-sequence generate().

    python -m llm_np_cp_amd.runtime.server --model llama-3.2-1b \
        --port 8080 --max-batch 8
    curl -d '{"prompt": "Once upon a time", "max_tokens": 32}' \
         -H 'Content-Type: application/json' localhost:8080/v1/completions
"""

from __future__ import annotations

import json
import queue
import threading
import time
from dataclasses import dataclass, field
from typing import Optional

try:  # pydantic model must live at module scope (ForwardRef resolution)
    from typing import Literal

    from pydantic import BaseModel, Field

    _Strategy = Literal["min_p", "greedy", "top_k", "top_p", "temperature"]

    class CompletionRequest(BaseModel):
        prompt: str
        max_tokens: int = Field(128, ge=1)
        temperature: float = Field(1.0, gt=0)
        min_p: float = Field(0.1, ge=0, le=1)
        strategy: _Strategy = "min_p"
        seed: Optional[int] = None
        stop_on_eos: bool = True
        stream: bool = False      # SSE token stream (OpenAI-style)
        stop: Optional[list] = None   # stop strings (OpenAI semantics)
        logprobs: Optional[int] = None  # top-N per-token logprobs
        n: int = Field(1, ge=1, le=16)  # independent samples
        logit_bias: Optional[dict] = None  # {token_id: bias}
        echo: bool = False        # prepend the prompt to the text

    class ChatMessage(BaseModel):
        role: str
        content: str

    class ChatCompletionRequest(BaseModel):
        messages: list = Field(..., min_length=1)
        max_tokens: int = Field(128, ge=1)
        temperature: float = Field(1.0, gt=0)
        min_p: float = Field(0.1, ge=0, le=1)
        strategy: _Strategy = "min_p"
        seed: Optional[int] = None
        stop_on_eos: bool = True
        stream: bool = False
        stop: Optional[list] = None
        logprobs: Optional[int] = None
        n: int = Field(1, ge=1, le=16)
        logit_bias: Optional[dict] = None
except ImportError:  # pragma: no cover - serving is optional
    CompletionRequest = None
    ChatMessage = None
    ChatCompletionRequest = None


class ClientDisconnected(Exception):
    """Raised inside the token callback to abort generation for a
    streaming client that went away (SSE cancellation)."""


@dataclass
class _Pending:
    req: "CompletionRequest"
    done: threading.Event = field(default_factory=threading.Event)
    result: Optional[dict] = None
    error: Optional[Exception] = None
    t0: float = field(default_factory=time.time)
    on_token: Optional[object] = None   # str -> None; SSE streaming hook


class BatchScheduler:
    """Owns the engine thread; continuous ragged batching for
    compatible concurrent requests."""

    def __init__(self, generate_one, run_group, max_batch: int,
                 window_s: float = 0.004):
        self.generate_one = generate_one
        self.run_group = run_group  # (pendings, poll_fn) -> resolves them
        self.max_batch = max_batch
        self.window_s = window_s
        self.q: "queue.Queue[_Pending]" = queue.Queue()
        self.stats = {"requests": 0, "batches": 0, "max_group": 0,
                      "joined_mid_flight": 0}
        t = threading.Thread(target=self._loop, daemon=True)
        t.start()

    def submit(self, req) -> dict:
        p = _Pending(req)
        self.q.put(p)
        p.done.wait()
        if p.error is not None:
            raise p.error
        return p.result

    def submit_async(self, req, on_token=None) -> _Pending:
        """Enqueue without blocking; the caller streams from on_token
        and watches pending.done (SSE path)."""
        p = _Pending(req, on_token=on_token)
        self.q.put(p)
        return p

    @staticmethod
    def _key(req):
        return (req.strategy, round(req.min_p, 6),
                round(req.temperature, 6), req.stop_on_eos)

    def _batchable(self, req) -> bool:
        # streaming and stop-string requests take the generate_one
        # path (per-token callback / text-level truncation); the
        # lockstep group only surfaces ids per chunk
        return (self.max_batch > 1 and req.strategy in ("greedy", "min_p")
                and not getattr(req, "stream", False)
                and not getattr(req, "stop", None)
                and getattr(req, "logprobs", None) is None
                and not getattr(req, "logit_bias", None)
                and not getattr(req, "echo", False)
                and getattr(req, "n", 1) == 1)

    def _poll_compatible(self, key, deferred):
        """Non-blocking: next queued pending with this sampling key;
        others are deferred (re-queued after the group).  Once anything
        is deferred, stop admitting joiners so the running group drains
        and the deferred requests cannot starve behind it."""
        if deferred:
            return None
        while True:
            try:
                p = self.q.get_nowait()
            except queue.Empty:
                return None
            if self._batchable(p.req) and self._key(p.req) == key:
                return p
            deferred.append(p)

    def _loop(self):
        while True:
            first = self.q.get()
            self.stats["batches"] += 1
            if not self._batchable(first.req):
                self.stats["requests"] += 1
                self.stats["max_group"] = max(self.stats["max_group"], 1)
                try:
                    if first.on_token is not None:
                        first.result = self.generate_one(
                            first.req, on_token=first.on_token)
                    else:
                        first.result = self.generate_one(first.req)
                except Exception as e:
                    first.error = e
                first.done.set()
                continue
            key = self._key(first.req)
            group = [first]
            deadline = time.monotonic() + self.window_s
            deferred = []
            while len(group) < self.max_batch:
                tmo = deadline - time.monotonic()
                if tmo <= 0:
                    break
                try:
                    nxt = self.q.get(timeout=tmo)
                except queue.Empty:
                    break
                if self._batchable(nxt.req) and self._key(nxt.req) == key:
                    group.append(nxt)
                else:
                    deferred.append(nxt)
            try:
                self.run_group(
                    group, lambda: self._poll_compatible(key, deferred),
                    self.stats)
            except Exception as e:
                for p in group:
                    if not p.done.is_set():
                        p.error = e
                        p.done.set()
            for p in deferred:  # re-enter scheduling
                self.q.put(p)


def _msg_text(c) -> str:
    """OpenAI allows content to be a string OR a list of typed parts
    ({"type": "text", "text": ...}); flatten the text parts."""
    if isinstance(c, list):
        return "".join(p.get("text", "") if isinstance(p, dict)
                       else str(p) for p in c)
    return str(c)


def chat_prompt(tok, messages) -> str:
    """Assemble chat messages into a prompt: the tokenizer's own
    chat_template when the checkpoint ships one (AutoTokenizer), else a
    neutral <|role|> fallback usable with any tokenizer."""
    msgs = [{"role": m["role"], "content": _msg_text(m["content"])}
            if isinstance(m, dict) else
            {"role": m.role, "content": _msg_text(m.content)}
            for m in messages]
    template = getattr(tok, "chat_template", None)
    if template and hasattr(tok, "apply_chat_template"):
        return tok.apply_chat_template(msgs, tokenize=False,
                                       add_generation_prompt=True)
    lines = [f"<|{m['role']}|>\n{m['content']}" for m in msgs]
    return "\n".join(lines) + "\n<|assistant|>\n"


def build_app(model_name: str = "llama-3.2-1b", backend: str = "auto",
              dtype: str = "bf16", max_seq: int = 4096,
              max_batch: int = 8, kv_dtype: str = "bf16",
              batch_window_ms: float = 4.0, lora: str = None,
              _engine=None):
    """``_engine``: test hook — a pre-built batch-capable engine
    (GPUModel interface: prefill_row/decode_rows/compact_row/bt_*)
    injected in place of the loaded model, so the continuous-batching
    scheduler logic is CPU-testable against a deterministic fake."""
    from fastapi import Body, FastAPI

    import llm_np_cp_amd as L

    tok, model, cfg = L.load_model(model_name, backend=backend,
                                   dtype=dtype, max_seq=max_seq,
                                   kv_dtype=kv_dtype, max_batch=max_batch,
                                   lora=lora)
    if _engine is not None:
        model = _engine
        can_batch = max_batch > 1
    else:
        # GPU engines built with batch pools serve the lockstep group
        can_batch = (max_batch > 1
                     and type(model).__name__ == "GPUModel")

    eos = getattr(cfg, "eos_token_id", None)
    eos_set = (set() if eos is None else {int(eos)}
               if isinstance(eos, (int, float)) else {int(e) for e in eos})

    # Prometheus metrics (SURVEY §5 observability); per-app registry so
    # repeated build_app calls (tests) never collide on timeseries names
    try:
        from prometheus_client import (CollectorRegistry, Counter,
                                       Histogram)
        metrics_reg = CollectorRegistry()
        m_requests = Counter("llm_requests_total",
                             "completed requests", ["endpoint"],
                             registry=metrics_reg)
        m_tokens = Counter("llm_completion_tokens_total",
                           "tokens generated", registry=metrics_reg)
        m_latency = Histogram("llm_request_seconds",
                              "end-to-end request latency",
                              registry=metrics_reg)
    except ImportError:  # pragma: no cover - metrics are optional
        metrics_reg = None

    def _payload(req, ids, text, total_s, prefill_s, tps,
                 finish_reason=None) -> dict:
        if metrics_reg is not None:
            m_tokens.inc(len(ids))
            m_latency.observe(total_s)
        if finish_reason is None:
            hit_eos = bool(ids) and int(ids[-1]) in eos_set
            finish_reason = "stop" if hit_eos else "length"
        return {
            "object": "text_completion",
            "model": model_name,
            "choices": [{"text": text, "index": 0,
                         "finish_reason": finish_reason}],
            "usage": {
                "prompt_tokens": len(tok.encode(req.prompt)),
                "completion_tokens": len(ids),
            },
            "timings": {
                "total_s": total_s,
                "prefill_s": prefill_s,
                "decode_tokens_per_s": tps,
            },
        }

    def generate_one(req, on_token=None) -> dict:
        n = getattr(req, "n", 1) or 1
        t0 = time.time()
        pay = None
        for idx in range(n):
            seed = (req.seed + idx) if (req.seed is not None and n > 1) \
                else req.seed
            bias = getattr(req, "logit_bias", None)
            params = L.SamplingParams(
                strategy=req.strategy, min_p=req.min_p,
                temperature=req.temperature, seed=seed,
                logit_bias={int(k): float(vv) for k, vv in bias.items()}
                if bias else None)
            out = L.generate(req.prompt, tok, model,
                             max_tokens=req.max_tokens, params=params,
                             stream=False, stop_on_eos=req.stop_on_eos,
                             stop=getattr(req, "stop", None),
                             logprobs=getattr(req, "logprobs", None),
                             on_token=on_token if idx == 0 else None)
            text = (req.prompt + out.text
                    if getattr(req, "echo", False) else out.text)
            if pay is None:
                pay = _payload(req, out.token_ids, text,
                               time.time() - t0, out.prefill_time_s,
                               out.decode_tokens_per_s,
                               finish_reason=out.finish_reason)
            else:
                pay["choices"].append({"text": text, "index": idx,
                                       "finish_reason": out.finish_reason})
                pay["usage"]["completion_tokens"] += len(out.token_ids)
                if metrics_reg is not None:  # extra choices count too
                    m_tokens.inc(len(out.token_ids))
            if out.logprobs is not None:
                pay["choices"][idx]["logprobs"] = {
                    "tokens": [e["token"] for e in out.logprobs],
                    "token_logprobs": [e["logprob"]
                                       for e in out.logprobs],
                    "top_logprobs": [{t["token"]: t["logprob"]
                                      for t in e["top"]}
                                     for e in out.logprobs],
                }
        pay["timings"]["total_s"] = time.time() - t0
        return pay

    class _Row:
        __slots__ = ("pending", "ids", "first_consumed")

        def __init__(self, pending):
            self.pending = pending
            self.ids = []
            self.first_consumed = False

    def run_group(pendings, poll, stats) -> None:
        """Continuous ragged batching: chunks of 16 steps; retired rows
        compact; compatible queued requests join between chunks."""
        if not can_batch:
            for p in pendings:
                stats["requests"] += 1
                try:
                    p.result = generate_one(p.req)
                except Exception as e:
                    p.error = e
                p.done.set()
            return
        r0 = pendings[0].req
        greedy = r0.strategy == "greedy"
        min_p, temp, stop_eos = r0.min_p, r0.temperature, r0.stop_on_eos
        rows: list = []  # slot -> _Row

        def admit(p):
            slot = len(rows)
            t_p = time.time()
            model.prefill_row(slot, tok.encode(p.req.prompt),
                              greedy=greedy, min_p=min_p, temperature=temp)
            p.prefill_s = time.time() - t_p
            rows.append(_Row(p))
            stats["requests"] += 1

        def finish(row):
            p = row.pending
            dt = time.time() - p.t0
            ids = row.ids[:p.req.max_tokens]
            p.result = _payload(p.req, ids, tok.decode(ids), dt,
                                getattr(p, "prefill_s", 0.0),
                                len(ids) / max(dt, 1e-9))
            p.done.set()

        for p in pendings:
            admit(p)
        stats["max_group"] = max(stats["max_group"], len(rows))
        room = model.max_seq - 2
        while rows:
            # mid-flight admission into free slots
            while len(rows) < max_batch:
                p2 = poll()
                if p2 is None:
                    break
                admit(p2)
                stats["joined_mid_flight"] += 1
                stats["max_group"] = max(stats["max_group"], len(rows))
            B = len(rows)
            chunk = min(16, room - max(model._host_lens[:B]))
            if chunk <= 0:  # a row ran out of KV room: retire longest
                longest = max(range(B), key=lambda b: model._host_lens[b])
                finish(rows[longest])
                model.compact_row(longest, B - 1)
                rows[longest] = rows[B - 1]
                rows.pop()
                continue
            out = model.decode_rows(B, chunk, greedy=greedy, min_p=min_p,
                                    temperature=temp)
            # the first sampled token (from prefill_row) precedes chunk 1
            for b in range(B):
                row = rows[b]
                if not row.first_consumed:
                    # ring holds first+chunk ids; decode_rows returned
                    # the last `chunk` — prepend the first token once
                    n0 = int(model.bt_nout[b].item())
                    first_id = int(model.bt_ring[b, 0].item()) \
                        if n0 >= 1 else None
                    if first_id is not None:
                        row.ids.append(first_id)
                    row.first_consumed = True
                row.ids.extend(int(t) for t in out[b])
            # retire rows (EOS / max_tokens), compacting from the end
            b = 0
            while b < len(rows):
                row = rows[b]
                req = row.pending.req
                hit_eos = (stop_eos and
                           any(int(t) in eos_set for t in row.ids))
                if hit_eos:
                    cut = next(i for i, t in enumerate(row.ids)
                               if int(t) in eos_set)
                    row.ids = row.ids[:cut + 1]
                if hit_eos or len(row.ids) >= req.max_tokens:
                    finish(row)
                    last = len(rows) - 1
                    model.compact_row(b, last)
                    rows[b] = rows[last]
                    rows.pop()
                    # re-check slot b (now holds the moved row)
                    continue
                b += 1

    sched = BatchScheduler(generate_one, run_group,
                           max_batch if can_batch else 1,
                           window_s=batch_window_ms / 1e3)
    app = FastAPI(title="llm_np_cp_amd", version=L.__version__)

    @app.get("/v1/models")
    def models():
        return {"object": "list",
                "data": [{"id": model_name, "object": "model",
                          "owned_by": "llm_np_cp_amd",
                          "meta": {"model_type": cfg.model_type,
                                   "dtype": dtype,
                                   "max_seq": model.max_seq
                                   if hasattr(model, "max_seq")
                                   else max_seq}}]}

    @app.get("/health")
    def health():
        return {"status": "ok", "model": model_name,
                "model_type": cfg.model_type,
                "backend": type(model).__name__,
                "max_batch": max_batch if can_batch else 1}

    @app.get("/stats")
    def stats():
        return dict(sched.stats)

    @app.get("/metrics")
    def metrics():
        from fastapi import HTTPException, Response
        if metrics_reg is None:
            raise HTTPException(status_code=404,
                                detail="prometheus_client not installed")
        from prometheus_client import CONTENT_TYPE_LATEST, generate_latest
        return Response(generate_latest(metrics_reg),
                        media_type=CONTENT_TYPE_LATEST)

    def _sse(req, chunk_of):
        """Run req through the scheduler, yielding one SSE event per
        decoded token piece, then a final [DONE].  Token callbacks fire
        on the scheduler thread; all of them happen-before done.set(),
        so `done and empty` is a safe termination check.

        CANCELLATION: when the client disconnects, Starlette closes the
        generator; the finally block sets the cancel flag and the next
        token callback raises inside generate() on the scheduler thread
        — generation stops instead of decoding to max_tokens for a
        reader that is gone."""
        from fastapi.responses import StreamingResponse

        q: "queue.Queue[str]" = queue.Queue()
        cancelled = threading.Event()

        def push(piece: str):
            if cancelled.is_set():
                raise ClientDisconnected()
            q.put(piece)

        p = sched.submit_async(req, on_token=push)

        def gen():
            try:
                while not (p.done.is_set() and q.empty()):
                    try:
                        piece = q.get(timeout=0.05)
                    except queue.Empty:
                        continue
                    yield "data: " + json.dumps(chunk_of(piece)) + "\n\n"
                if p.error is not None:
                    yield ("data: " + json.dumps({"error": str(p.error)})
                           + "\n\n")
                yield "data: [DONE]\n\n"
            finally:
                cancelled.set()

        return StreamingResponse(gen(), media_type="text/event-stream")

    @app.post("/v1/completions")
    def completions(req: CompletionRequest = Body(...)):
        if metrics_reg is not None:
            m_requests.labels(endpoint="completions").inc()
        if req.stream:
            return _sse(req, lambda piece: {
                "object": "text_completion.chunk", "model": model_name,
                "choices": [{"text": piece, "index": 0}]})
        try:
            return sched.submit(req)
        except ValueError as e:  # engine-level rejection (e.g. KV room)
            from fastapi import HTTPException
            raise HTTPException(status_code=400, detail=str(e))

    @app.post("/v1/chat/completions")
    def chat_completions(req: ChatCompletionRequest = Body(...)):
        if metrics_reg is not None:
            m_requests.labels(endpoint="chat").inc()
        creq = CompletionRequest(
            prompt=chat_prompt(tok, req.messages),
            max_tokens=req.max_tokens,
            temperature=req.temperature, min_p=req.min_p,
            strategy=req.strategy, seed=req.seed,
            stop_on_eos=req.stop_on_eos, stream=req.stream,
            stop=req.stop, logprobs=req.logprobs, n=req.n,
            logit_bias=req.logit_bias)
        if req.stream:
            return _sse(creq, lambda piece: {
                "object": "chat.completion.chunk", "model": model_name,
                "choices": [{"delta": {"content": piece}, "index": 0}]})
        try:
            out = sched.submit(creq)
        except ValueError as e:
            from fastapi import HTTPException
            raise HTTPException(status_code=400, detail=str(e))
        choices = []
        for i, ch in enumerate(out["choices"]):
            c = {"index": i,
                 "message": {"role": "assistant",
                             "content": ch["text"]},
                 "finish_reason": ch["finish_reason"]}
            if "logprobs" in ch:
                c["logprobs"] = ch["logprobs"]
            choices.append(c)
        return {
            "object": "chat.completion", "model": model_name,
            "choices": choices,
            "usage": out["usage"], "timings": out["timings"],
        }

    return app


def main():
    import argparse

    import uvicorn

    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama-3.2-1b")
    ap.add_argument("--backend", default="auto")
    ap.add_argument("--dtype", default="bf16",
                    choices=["bf16", "fp8", "fp4"])
    ap.add_argument("--kv-dtype", default="bf16", choices=["bf16", "fp8"])
    ap.add_argument("--max-seq", type=int, default=4096)
    ap.add_argument("--max-batch", type=int, default=8)
    ap.add_argument("--batch-window-ms", type=float, default=4.0)
    ap.add_argument("--lora", default=None,
                    help="PEFT adapter dir merged at load")
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=8080)
    args = ap.parse_args()
    app = build_app(args.model, args.backend, args.dtype, args.max_seq,
                    max_batch=args.max_batch, kv_dtype=args.kv_dtype,
                    batch_window_ms=args.batch_window_ms, lora=args.lora)
    uvicorn.run(app, host=args.host, port=args.port)


if __name__ == "__main__":
    main()
