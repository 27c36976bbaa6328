#!/usr/bin/env python3
"""Long decode soak: verify device-side state stays sane over thousands
of graph replays (tickets re-armed, RNG advancing, no NaNs, ids in
range, KV length bookkeeping exact)."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from csrc.build import ensure_built
ensure_built()
import llm_np_cp_amd as L
from llm_np_cp_amd.io.loader import LazyRandomWeights
from llm_np_cp_amd.models.engine import GPUModel

def soak(preset, dtype, steps, greedy):
    cfg = L.preset_config(preset)
    m = GPUModel(cfg, LazyRandomWeights(cfg, 0), max_seq=steps + 128,
                 dtype=dtype)
    prompt = np.random.default_rng(0).integers(0, cfg.vocab_size, size=64)
    m.prefill(prompt)
    ids = m.decode(steps, greedy=greedy, min_p=0.1, use_graph=True)
    assert len(ids) == steps, (len(ids), steps)
    assert ids.min() >= 0 and ids.max() < cfg.vocab_size
    # the final sampled token is not yet processed into the cache:
    # len = prompt + (steps - 1) processed decode tokens
    assert int(m.len_buf.item()) == 64 + steps - 1, int(m.len_buf.item())
    assert int(m.attn_cnt.sum().item()) == 0  # tickets re-armed
    logits = m.b_logits.cpu().numpy()
    assert np.isfinite(logits).all(), "non-finite logits after soak"
    uniq = len(set(ids.tolist()))
    print(f"{preset} {dtype} greedy={greedy}: {steps} steps OK, "
          f"{uniq} distinct tokens, len={int(m.len_buf.item())}")

soak("llama-3.2-1b", "fp8", 4000, True)
soak("llama-3.2-1b", "fp8", 1000, False)   # min-p stochastic path
soak("gemma-2-2b", "bf16", 1500, True)
print("SOAK PASSED")
