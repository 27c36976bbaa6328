"""LoRA merge-at-load (PEFT adapter format): W' = W + (alpha/r) B@A
folded into the base weights before upload, so adapted models run at
full native speed on either engine (beyond-parity capability; the
reference has no adapter support)."""

import json
import os

import numpy as np
import pytest
import torch

import llm_np_cp_amd as L
from llm_np_cp_amd.io.loader import (apply_lora, load_lora,
                                     random_weights,
                                     write_synthetic_checkpoint)
from llm_np_cp_amd.models.numpy_ref import NumpyKVCache, NumpyModel


def _write_adapter(d, cfg, r=4, alpha=8.0, seed=0,
                   targets=("q_proj", "v_proj")):
    """Synthetic PEFT-format adapter dir targeting attn projections of
    every layer; returns {base_key: (A, B)} for the reference merge."""
    from safetensors.torch import save_file

    rng = np.random.default_rng(seed)
    os.makedirs(d, exist_ok=True)
    with open(os.path.join(d, "adapter_config.json"), "w") as f:
        json.dump({"r": r, "lora_alpha": alpha, "peft_type": "LORA",
                   "target_modules": list(targets)}, f)
    tensors = {}
    ref = {}
    for i in range(cfg.num_hidden_layers):
        for t in targets:
            base = f"model.layers.{i}.self_attn.{t}"
            out_dim = (cfg.num_attention_heads * cfg.head_dim
                       if t == "q_proj"
                       else cfg.num_key_value_heads * cfg.head_dim)
            A = (0.1 * rng.standard_normal((r, cfg.hidden_size))
                 ).astype(np.float32)
            B = (0.1 * rng.standard_normal((out_dim, r))
                 ).astype(np.float32)
            peft = f"base_model.model.{base}"
            tensors[f"{peft}.lora_A.weight"] = torch.from_numpy(A)
            tensors[f"{peft}.lora_B.weight"] = torch.from_numpy(B)
            ref[f"{base}.weight"] = (A, B)
    save_file(tensors, os.path.join(d, "adapter_model.safetensors"))
    return ref, alpha / r


def test_apply_lora_matches_manual_merge(tmp_path):
    cfg = L.preset_config("tiny-llama")
    ref, scaling = _write_adapter(str(tmp_path), cfg, seed=3)

    w = random_weights(cfg, seed=1)
    manual = {k: v.copy() for k, v in w.items()}
    for key, (A, B) in ref.items():
        manual[key] = manual[key] + np.float32(scaling) * (B @ A)

    n = apply_lora(w, str(tmp_path))
    assert n == len(ref) == 2 * cfg.num_hidden_layers
    for k in w:
        np.testing.assert_allclose(w[k], manual[k], rtol=1e-6, atol=1e-7)

    # the merged model's logits differ from base but equal the manual
    # merge's logits exactly
    ids = np.arange(1, 9)
    base = NumpyModel(cfg, random_weights(cfg, seed=1)).forward(
        ids, NumpyKVCache(cfg, 16), 0)
    merged = NumpyModel(cfg, w).forward(ids, NumpyKVCache(cfg, 16), 0)
    man = NumpyModel(cfg, manual).forward(ids, NumpyKVCache(cfg, 16), 0)
    assert not np.allclose(base, merged)
    np.testing.assert_allclose(merged, man, rtol=1e-6, atol=1e-6)


def test_load_model_with_lora_end_to_end(tmp_path):
    ck = str(tmp_path / "ckpt")
    ad = str(tmp_path / "adapter")
    cfg = write_synthetic_checkpoint(ck, "tiny-llama", seed=5)
    _write_adapter(ad, cfg, seed=7)

    tok, base, _ = L.load_model(ck, backend="numpy")
    tok, tuned, _ = L.load_model(ck, backend="numpy", lora=ad)
    p = L.SamplingParams(strategy="greedy")
    a = L.generate("Once upon a time", tok, base, max_tokens=8,
                   stream=False, params=p, stop_on_eos=False)
    b = L.generate("Once upon a time", tok, tuned, max_tokens=8,
                   stream=False, params=p, stop_on_eos=False)
    assert len(b.token_ids) == 8
    # different weights => (almost surely) different greedy logits path
    lb = base.forward(np.arange(1, 6), NumpyKVCache(cfg, 16), 0)
    lt = tuned.forward(np.arange(1, 6), NumpyKVCache(cfg, 16), 0)
    assert not np.allclose(lb, lt)


def test_lora_error_paths(tmp_path):
    cfg = L.preset_config("tiny-llama")
    ad = str(tmp_path / "a")
    _write_adapter(ad, cfg, seed=1)
    w = random_weights(cfg, seed=0)

    # shape mismatch: adapter built for a different architecture
    cfg2 = L.preset_config("tiny-mistral")
    w2 = random_weights(cfg2, seed=0)
    with pytest.raises(ValueError, match="missing base tensor|shape"):
        apply_lora(w2, ad)

    # incomplete adapter (A without B)
    from safetensors.torch import save_file
    bad = str(tmp_path / "bad")
    os.makedirs(bad, exist_ok=True)
    with open(os.path.join(bad, "adapter_config.json"), "w") as f:
        json.dump({"r": 2, "lora_alpha": 4}, f)
    save_file({"base_model.model.model.layers.0.self_attn.q_proj"
               ".lora_A.weight": torch.zeros(2, cfg.hidden_size)},
              os.path.join(bad, "adapter_model.safetensors"))
    with pytest.raises(ValueError, match="incomplete"):
        load_lora(bad)
