#!/usr/bin/env python3
"""Prefill (TTFT) timing: Llama-3.2-1B, 2048-token synthetic prompt."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch
from csrc.build import ensure_built
ensure_built()
import llm_np_cp_amd as L
from llm_np_cp_amd.io.loader import LazyRandomWeights
from llm_np_cp_amd.models.engine import GPUModel

cfg = L.preset_config(sys.argv[1] if len(sys.argv) > 1 else "llama-3.2-1b")
m = GPUModel(cfg, LazyRandomWeights(cfg, 0), max_seq=4096)
prompt = np.random.default_rng(0).integers(0, cfg.vocab_size, size=2048)
m.prefill(prompt[:64])  # warm
for n in (512, 2048):
    m.make_cache(4096)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    m.prefill(prompt[:n])
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"prefill {n:5d} tokens: {dt*1e3:8.2f} ms "
          f"({n/dt:8.0f} tok/s prefill)")
