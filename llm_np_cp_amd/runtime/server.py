"""HTTP serving layer with dynamic request batching.

The reference has no server (no CLI, no server); this is the
deployment-facing wrapper around the engine: an OpenAI-style
``/v1/completions`` endpoint plus ``/health`` and ``/stats``.

Concurrent requests are NOT serialized: a scheduler thread owns the
engine and groups compatible in-flight requests (same sampling mode)
into one RAGGED lockstep batch (``GPUModel.prefill_batch`` /
``decode_batch`` — per-sequence KV pools, per-row device positions),
decoding in chunks and retiring rows on EOS / per-request max_tokens.
Requests with CPU-only sampling strategies (top_k/top_p) or a non-batch
engine fall back to single-sequence generate().

    python -m llm_np_cp_amd.runtime.server --model llama-3.2-1b \
        --port 8080 --max-batch 8
    curl -d '{"prompt": "Once upon a time", "max_tokens": 32}' \
         -H 'Content-Type: application/json' localhost:8080/v1/completions
"""

from __future__ import annotations

import queue
import threading
import time
from dataclasses import dataclass, field
from typing import Optional

try:  # pydantic model must live at module scope (ForwardRef resolution)
    from pydantic import BaseModel

    class CompletionRequest(BaseModel):
        prompt: str
        max_tokens: int = 128
        temperature: float = 1.0
        min_p: float = 0.1
        strategy: str = "min_p"   # min_p | greedy | top_k | top_p
        seed: Optional[int] = None
        stop_on_eos: bool = True
except ImportError:  # pragma: no cover - serving is optional
    CompletionRequest = None


@dataclass
class _Pending:
    req: "CompletionRequest"
    done: threading.Event = field(default_factory=threading.Event)
    result: Optional[dict] = None
    error: Optional[Exception] = None


class BatchScheduler:
    """Groups compatible concurrent requests into ragged lockstep
    batches on a single engine-owning thread."""

    def __init__(self, generate_one, run_group, max_batch: int,
                 window_s: float = 0.004):
        self.generate_one = generate_one
        self.run_group = run_group
        self.max_batch = max_batch
        self.window_s = window_s
        self.q: "queue.Queue[_Pending]" = queue.Queue()
        self.stats = {"requests": 0, "batches": 0, "max_group": 0}
        t = threading.Thread(target=self._loop, daemon=True)
        t.start()

    def submit(self, req) -> dict:
        p = _Pending(req)
        self.q.put(p)
        p.done.wait()
        if p.error is not None:
            raise p.error
        return p.result

    @staticmethod
    def _key(req):
        return (req.strategy, round(req.min_p, 6),
                round(req.temperature, 6), req.stop_on_eos)

    def _batchable(self, req) -> bool:
        return self.max_batch > 1 and req.strategy in ("greedy", "min_p")

    def _loop(self):
        while True:
            first = self.q.get()
            group = [first]
            if self._batchable(first.req):
                key = self._key(first.req)
                deadline = time.monotonic() + self.window_s
                while len(group) < self.max_batch:
                    tmo = deadline - time.monotonic()
                    if tmo <= 0:
                        break
                    try:
                        nxt = self.q.get(timeout=tmo)
                    except queue.Empty:
                        break
                    if self._batchable(nxt.req) and \
                            self._key(nxt.req) == key:
                        group.append(nxt)
                    else:
                        self.q.put(nxt)
                        break
            self.stats["requests"] += len(group)
            self.stats["batches"] += 1
            self.stats["max_group"] = max(self.stats["max_group"],
                                          len(group))
            try:
                if len(group) == 1 and not self._batchable(first.req):
                    group[0].result = self.generate_one(group[0].req)
                else:
                    results = self.run_group([p.req for p in group])
                    for p, r in zip(group, results):
                        p.result = r
            except Exception as e:  # surface to every waiter
                for p in group:
                    p.error = e
            finally:
                for p in group:
                    p.done.set()


def build_app(model_name: str = "llama-3.2-1b", backend: str = "auto",
              dtype: str = "bf16", max_seq: int = 4096,
              max_batch: int = 8, kv_dtype: str = "bf16",
              batch_window_ms: float = 4.0):
    from fastapi import Body, FastAPI

    import llm_np_cp_amd as L

    tok, model, cfg = L.load_model(model_name, backend=backend,
                                   dtype=dtype, max_seq=max_seq,
                                   kv_dtype=kv_dtype)
    can_batch = False
    if max_batch > 1 and type(model).__name__ == "GPUModel":
        # rebuild with batch pools (load_model has no max_batch knob to
        # keep its reference-parity signature lean)
        from ..core.config import PRESETS
        from ..io.loader import random_weights, load_weights_numpy, \
            load_config
        from ..models.engine import GPUModel
        import os
        if os.path.isdir(model_name):
            weights = load_weights_numpy(model_name)
        else:
            weights = random_weights(cfg, seed=0)
        model = GPUModel(cfg, weights, dtype=dtype, max_seq=max_seq,
                         kv_dtype=kv_dtype, max_batch=max_batch)
        can_batch = True

    eos = getattr(cfg, "eos_token_id", None)
    eos_set = (set() if eos is None else {int(eos)}
               if isinstance(eos, (int, float)) else {int(e) for e in eos})

    def generate_one(req) -> dict:
        params = L.SamplingParams(strategy=req.strategy, min_p=req.min_p,
                                  temperature=req.temperature,
                                  seed=req.seed)
        t0 = time.time()
        out = L.generate(req.prompt, tok, model,
                         max_tokens=req.max_tokens, params=params,
                         stream=False, stop_on_eos=req.stop_on_eos)
        return _payload(req, out.token_ids, out.text, time.time() - t0,
                        out.prefill_time_s, out.decode_tokens_per_s)

    def run_group(reqs) -> list:
        if not can_batch or len(reqs) == 1:
            return [generate_one(r) for r in reqs]
        t0 = time.time()
        prompts = [tok.encode(r.prompt) for r in reqs]
        B = len(reqs)
        greedy = reqs[0].strategy == "greedy"
        min_p = reqs[0].min_p
        temp = reqs[0].temperature
        stop_eos = reqs[0].stop_on_eos
        maxn = max(r.max_tokens for r in reqs)
        room = model.max_seq - max(len(p) for p in prompts) - 1
        maxn = min(maxn, room)
        model.prefill_batch(prompts)
        tp = time.time() - t0
        rows = [[] for _ in range(B)]
        done = [False] * B
        produced = 0
        first = True
        while produced < maxn and not all(done):
            n = min(32, maxn - produced)
            ids = model.decode_batch(n, greedy=greedy, min_p=min_p,
                                     temperature=temp,
                                     first_from_logits=first)
            first = False
            produced += n
            for b in range(B):
                if done[b]:
                    continue
                for t in ids[b].tolist():
                    rows[b].append(int(t))
                    if len(rows[b]) >= reqs[b].max_tokens:
                        done[b] = True
                        break
                    if stop_eos and int(t) in eos_set:
                        done[b] = True
                        break
        dt = time.time() - t0
        n_out = sum(len(r) for r in rows)
        return [_payload(reqs[b], rows[b], tok.decode(rows[b]), dt, tp,
                         n_out / max(dt - tp, 1e-9))
                for b in range(B)]

    def _payload(req, ids, text, total_s, prefill_s, tps) -> dict:
        return {
            "object": "text_completion",
            "model": model_name,
            "choices": [{"text": text, "index": 0,
                         "finish_reason": "stop"}],
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

    sched = BatchScheduler(generate_one, run_group,
                           max_batch if can_batch else 1,
                           window_s=batch_window_ms / 1e3)
    app = FastAPI(title="llm_np_cp_amd", version=L.__version__)

    @app.get("/health")
    def health():
        return {"status": "ok", "model": model_name,
                "model_type": cfg.model_type,
                "backend": type(model).__name__,
                "max_batch": max_batch if can_batch else 1}

    @app.get("/stats")
    def stats():
        return dict(sched.stats)

    @app.post("/v1/completions")
    def completions(req: CompletionRequest = Body(...)):
        return sched.submit(req)

    return app


def main():
    import argparse

    import uvicorn

    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama-3.2-1b")
    ap.add_argument("--backend", default="auto")
    ap.add_argument("--dtype", default="bf16", choices=["bf16", "fp8"])
    ap.add_argument("--kv-dtype", default="bf16", choices=["bf16", "fp8"])
    ap.add_argument("--max-seq", type=int, default=4096)
    ap.add_argument("--max-batch", type=int, default=8)
    ap.add_argument("--batch-window-ms", type=float, default=4.0)
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=8080)
    args = ap.parse_args()
    app = build_app(args.model, args.backend, args.dtype, args.max_seq,
                    max_batch=args.max_batch, kv_dtype=args.kv_dtype,
                    batch_window_ms=args.batch_window_ms)
    uvicorn.run(app, host=args.host, port=args.port)


if __name__ == "__main__":
    main()
