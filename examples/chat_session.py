#!/usr/bin/env python3
"""Interactive multi-turn chat with KV reuse: each turn prefills only
the new tokens (ChatSession keeps the KV cache across turns).

    python examples/chat_session.py [model] [dtype]
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import llm_np_cp_amd as L

model_name = sys.argv[1] if len(sys.argv) > 1 else "tiny-llama"
dtype = sys.argv[2] if len(sys.argv) > 2 else "bf16"
tok, model, cfg = L.load_model(model_name, dtype=dtype)
session = L.ChatSession(tok, model,
                        params=L.SamplingParams(strategy="min_p", seed=0))

print(f"{model_name} ready ({session.max_seq}-token session pool). "
      f"Ctrl-D to exit.")
while True:
    try:
        text = input("> ")
    except EOFError:
        break
    if not text.strip():
        continue
    r = session.send(f"<|user|>\n{text}\n<|assistant|>\n", max_tokens=128)
    print(r.text)
    print(f"  [{session.seq_len} tokens resident; turn prefill "
          f"{r.prefill_time_s * 1e3:.1f} ms, "
          f"{r.decode_tokens_per_s:.0f} tok/s]")
