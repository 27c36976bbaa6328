#!/usr/bin/env python3
"""Client for the HTTP server (start it first):

    python -m llm_np_cp_amd.runtime.server --model llama-3.2-1b \
        --dtype fp8 --max-batch 8 --port 8080
    python examples/server_client.py [--stream]

Uses only the standard library; shows a plain completion, a chat
completion, and SSE streaming."""
import http.client
import json
import sys

HOST, PORT = "127.0.0.1", 8080


def post(path, body):
    c = http.client.HTTPConnection(HOST, PORT, timeout=300)
    c.request("POST", path, body=json.dumps(body),
              headers={"Content-Type": "application/json"})
    return c, c.getresponse()


if "--stream" in sys.argv:
    c, r = post("/v1/completions", {
        "prompt": "Once upon a time", "max_tokens": 64,
        "strategy": "min_p", "stream": True, "stop_on_eos": False})
    for raw in r:
        for line in raw.split(b"\n"):
            if not line.startswith(b"data: "):
                continue
            payload = line[len(b"data: "):]
            if payload == b"[DONE]":
                print()
                sys.exit(0)
            evt = json.loads(payload)
            sys.stdout.write(evt["choices"][0].get("text", ""))
            sys.stdout.flush()
else:
    c, r = post("/v1/completions", {
        "prompt": "Once upon a time", "max_tokens": 48,
        "strategy": "min_p", "stop": ["\n\n"], "logprobs": 3})
    body = json.loads(r.read())
    print(body["choices"][0]["text"])
    print("--", body["usage"], body["timings"])

    c, r = post("/v1/chat/completions", {
        "messages": [{"role": "system", "content": "Be brief."},
                     {"role": "user", "content": "Hello!"}],
        "max_tokens": 32})
    body = json.loads(r.read())
    print(body["choices"][0]["message"]["content"])
