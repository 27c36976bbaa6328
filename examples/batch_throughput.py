#!/usr/bin/env python3
"""Throughput mode (GPU): B prompts decode in lockstep over
per-sequence KV pools; the skinny fp8 MFMA GEMM streams each weight
matrix ONCE per step for all rows.

    python examples/batch_throughput.py [preset] [batch] [tokens]
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

import llm_np_cp_amd as L
from llm_np_cp_amd.io.loader import random_weights
from llm_np_cp_amd.models.engine import GPUModel

preset = sys.argv[1] if len(sys.argv) > 1 else "llama-3.2-1b"
B = int(sys.argv[2]) if len(sys.argv) > 2 else 8
N = int(sys.argv[3]) if len(sys.argv) > 3 else 256

cfg = L.preset_config(preset)
model = GPUModel(cfg, random_weights(cfg), dtype="fp8", max_seq=1024,
                 max_batch=B)
rng = np.random.default_rng(0)
prompts = [rng.integers(0, cfg.vocab_size, size=rng.integers(8, 64))
           for _ in range(B)]
model.prefill_batch(prompts)
model.decode_batch(8)                      # warm + capture the graph
t0 = time.perf_counter()
ids = model.decode_batch(N, first_from_logits=False)
dt = time.perf_counter() - t0
print(f"{preset} fp8 batch={B}: {B * N / dt:.0f} aggregate tok/s "
      f"({dt / N * 1e3:.2f} ms/step), out shape {ids.shape}")
