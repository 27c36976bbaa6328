"""Minimal HTTP serving layer (single-sequence, batch-1 engine).

The reference has no server (no CLI, no server); this is the
deployment-facing wrapper around the same ``generate()`` capability:
an OpenAI-style ``/v1/completions`` endpoint plus ``/health``.  One
model instance, requests served sequentially (the engine is a batch-1
decode engine by design — matching the reference's scope).

    python -m llm_np_cp_amd.runtime.server --model llama-3.2-1b --port 8080
    curl -d '{"prompt": "Once upon a time", "max_tokens": 32}' \
         -H 'Content-Type: application/json' localhost:8080/v1/completions
"""

from __future__ import annotations

import threading
import time
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


def build_app(model_name: str = "llama-3.2-1b", backend: str = "auto",
              dtype: str = "bf16", max_seq: int = 4096):
    from fastapi import Body, FastAPI

    import llm_np_cp_amd as L

    tok, model, cfg = L.load_model(model_name, backend=backend,
                                   dtype=dtype, max_seq=max_seq)
    lock = threading.Lock()  # batch-1 engine: serialize requests
    app = FastAPI(title="llm_np_cp_amd", version=L.__version__)

    @app.get("/health")
    def health():
        return {"status": "ok", "model": model_name,
                "model_type": cfg.model_type, "backend": type(model).__name__}

    @app.post("/v1/completions")
    def completions(req: CompletionRequest = Body(...)):
        params = L.SamplingParams(strategy=req.strategy, min_p=req.min_p,
                                  temperature=req.temperature, seed=req.seed)
        t0 = time.time()
        with lock:
            out = L.generate(req.prompt, tok, model,
                             max_tokens=req.max_tokens, params=params,
                             stream=False, stop_on_eos=req.stop_on_eos)
        dt = time.time() - t0
        return {
            "object": "text_completion",
            "model": model_name,
            "choices": [{"text": out.text, "index": 0,
                         "finish_reason": "stop"}],
            "usage": {
                "prompt_tokens": len(tok.encode(req.prompt)),
                "completion_tokens": len(out.token_ids),
            },
            "timings": {
                "total_s": dt,
                "prefill_s": out.prefill_time_s,
                "decode_tokens_per_s": out.decode_tokens_per_s,
            },
        }

    return app


def main():
    import argparse

    import uvicorn

    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama-3.2-1b")
    ap.add_argument("--backend", default="auto")
    ap.add_argument("--dtype", default="bf16", choices=["bf16", "fp8"])
    ap.add_argument("--max-seq", type=int, default=4096)
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=8080)
    args = ap.parse_args()
    app = build_app(args.model, args.backend, args.dtype, args.max_seq)
    uvicorn.run(app, host=args.host, port=args.port)


if __name__ == "__main__":
    main()
