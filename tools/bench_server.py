#!/usr/bin/env python3
"""Serving throughput: N concurrent clients against the continuous-
batching server (in-process TestClient).  Reports aggregate completion
tokens/sec and the /stats grouping evidence."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from concurrent.futures import ThreadPoolExecutor

from csrc.build import ensure_built
ensure_built()
from fastapi.testclient import TestClient
from llm_np_cp_amd.runtime.server import build_app

model = sys.argv[1] if len(sys.argv) > 1 else "llama-3.2-1b"
dtype = sys.argv[2] if len(sys.argv) > 2 else "fp8"
clients = int(sys.argv[3]) if len(sys.argv) > 3 else 16
max_tokens = int(sys.argv[4]) if len(sys.argv) > 4 else 128

app = build_app(model, backend="gpu", dtype=dtype, max_seq=2048,
                max_batch=16, batch_window_ms=4.0)
client = TestClient(app)
prompts = [f"request {i}: once upon a time in a datacenter"
           for i in range(clients)]

def one(i):
    r = client.post("/v1/completions", json={
        "prompt": prompts[i], "max_tokens": max_tokens,
        "strategy": "greedy", "stop_on_eos": False})
    assert r.status_code == 200, r.text
    return r.json()["usage"]["completion_tokens"]

one(0)  # warm (captures graphs)
t0 = time.perf_counter()
with ThreadPoolExecutor(max_workers=clients) as ex:
    toks = sum(ex.map(one, range(clients)))
dt = time.perf_counter() - t0
stats = client.get("/stats").json()
print(f"{model} {dtype}: {clients} concurrent x {max_tokens} tokens -> "
      f"{toks} tokens in {dt:.2f}s = {toks/dt:.0f} tok/s aggregate")
print("scheduler stats:", stats)
