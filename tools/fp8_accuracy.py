#!/usr/bin/env python3
"""Quantify quantized-weight decode accuracy vs the bf16 engine
(llama-1b).  Usage: python tools/fp8_accuracy.py [fp8|fp4]"""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from csrc.build import ensure_built
ensure_built()
import llm_np_cp_amd as L
from llm_np_cp_amd.io.loader import LazyRandomWeights
from llm_np_cp_amd.models.engine import GPUModel

qdtype = sys.argv[1] if len(sys.argv) > 1 else "fp8"
cfg = L.preset_config("llama-3.2-1b")
w = LazyRandomWeights(cfg, 0)
bf = GPUModel(cfg, w, max_seq=512)
f8 = GPUModel(cfg, w, max_seq=512, dtype=qdtype)
rng = np.random.default_rng(0)
prompt = rng.integers(0, cfg.vocab_size, size=64)

cb = bf.make_cache(512); cf = f8.make_cache(512)
lb = bf.forward(prompt, cb, 0)[0]
lf = f8.forward(prompt, cf, 0)[0]
# TEACHER-FORCED: both engines consume the bf16 engine's token each
# step, so the per-step KL compares the SAME context (separate rollouts
# diverge after the first disagreement and the KL stops meaning much)
agree = 0; kl_sum = 0.0; n_steps = 64
for i in range(n_steps):
    tb = int(np.argmax(lb)); tf = int(np.argmax(lf))
    agree += tb == tf
    pb = np.exp(lb - lb.max()); pb /= pb.sum()
    pf = np.exp(lf - lf.max()); pf /= pf.sum()
    kl_sum += float(np.sum(pb * (np.log(pb + 1e-12) - np.log(pf + 1e-12))))
    lb = bf.forward(np.asarray([tb]), cb, cb.seq_len)[0]
    lf = f8.forward(np.asarray([tb]), cf, cf.seq_len)[0]
print(f"greedy token agreement over {n_steps} teacher-forced steps: "
      f"{agree}/{n_steps}")
print(f"mean KL(bf16 || {qdtype}) per step: {kl_sum/n_steps:.5f} nats")
