#!/usr/bin/env python3
"""Speculative decoding: a cheap DRAFT proposes k tokens per TARGET
verify pass.  Greedy mode is token-identical to the target's own
decode; pass SamplingParams for stochastic speculative sampling
(distribution provably unchanged).

    python examples/speculative_decode.py              # tiny CPU demo
    # real deployment shape: quantized self-draft on GPU
    # python examples/speculative_decode.py llama-3.1-8b bf16 fp4
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import llm_np_cp_amd as L

model = sys.argv[1] if len(sys.argv) > 1 else "tiny-llama"
t_dtype = sys.argv[2] if len(sys.argv) > 2 else "bf16"
d_dtype = sys.argv[3] if len(sys.argv) > 3 else t_dtype

tok, target, _ = L.load_model(model, dtype=t_dtype)
_, draft, _ = L.load_model(model, dtype=d_dtype)

res = L.generate_speculative("Once upon a time", tok, draft, target,
                             max_tokens=48, k=4, stop_on_eos=False)
s = res.spec_stats
print(res.text)
print(f"--\naccepted {s['accepted']}/{s['proposed']} drafts over "
      f"{s['verify_passes']} verify passes "
      f"({s['accepted'] / max(s['proposed'], 1):.0%})")
