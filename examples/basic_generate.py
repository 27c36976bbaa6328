#!/usr/bin/env python3
"""Minimal end-to-end generation (mirrors the reference's __main__:
load -> generate 'Once upon a time' with streaming output).

    python examples/basic_generate.py                  # preset, CPU/GPU auto
    python examples/basic_generate.py /path/to/ckpt    # any HF Llama/Gemma/
                                                       # Qwen/Mistral/Mixtral
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import llm_np_cp_amd as L

model_name = sys.argv[1] if len(sys.argv) > 1 else "tiny-llama"
tok, model, cfg = L.load_model(model_name, backend="auto")
out = L.generate("Once upon a time", tok, model, max_tokens=64,
                 params=L.SamplingParams(strategy="min_p", seed=0),
                 stop_on_eos=False)
print(f"\n--\n{len(out.token_ids)} tokens, "
      f"{out.decode_tokens_per_s:.1f} tok/s decode, "
      f"prefill {out.prefill_time_s * 1e3:.1f} ms")
