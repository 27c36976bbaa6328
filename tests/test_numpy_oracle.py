"""NumPy reference model vs HF transformers on tiny random-init configs.

This is the oracle-validation layer: the reference validated
itself interactively against ``transformers``; we make that comparison an
automated test, then every later (HIP) component is tested against the
NumPy model.
"""

import numpy as np
import pytest
import torch

from llm_np_cp_amd.core.config import preset_config
from llm_np_cp_amd.models.numpy_ref import NumpyKVCache, NumpyModel


def hf_llama(cfg):
    from transformers import LlamaConfig, LlamaForCausalLM

    hf_cfg = LlamaConfig(
        vocab_size=cfg.vocab_size, hidden_size=cfg.hidden_size,
        intermediate_size=cfg.intermediate_size,
        num_hidden_layers=cfg.num_hidden_layers,
        num_attention_heads=cfg.num_attention_heads,
        num_key_value_heads=cfg.num_key_value_heads,
        head_dim=cfg.head_dim, rms_norm_eps=cfg.rms_norm_eps,
        rope_theta=cfg.rope_theta,
        max_position_embeddings=cfg.max_position_embeddings,
        tie_word_embeddings=cfg.tie_word_embeddings,
        attn_implementation="eager",
    )
    torch.manual_seed(7)
    return LlamaForCausalLM(hf_cfg).eval()


def hf_gemma2(cfg):
    from transformers import Gemma2Config, Gemma2ForCausalLM

    hf_cfg = Gemma2Config(
        vocab_size=cfg.vocab_size, hidden_size=cfg.hidden_size,
        intermediate_size=cfg.intermediate_size,
        num_hidden_layers=cfg.num_hidden_layers,
        num_attention_heads=cfg.num_attention_heads,
        num_key_value_heads=cfg.num_key_value_heads,
        head_dim=cfg.head_dim, rms_norm_eps=cfg.rms_norm_eps,
        rope_theta=cfg.rope_theta,
        query_pre_attn_scalar=cfg.query_pre_attn_scalar,
        sliding_window=cfg.sliding_window,
        attn_logit_softcapping=cfg.attn_logit_softcapping,
        final_logit_softcapping=cfg.final_logit_softcapping,
        tie_word_embeddings=True,
        attn_implementation="eager",
    )
    torch.manual_seed(9)
    return Gemma2ForCausalLM(hf_cfg).eval()


def hf_qwen2(cfg):
    from transformers import Qwen2Config, Qwen2ForCausalLM

    hf_cfg = Qwen2Config(
        vocab_size=cfg.vocab_size, hidden_size=cfg.hidden_size,
        intermediate_size=cfg.intermediate_size,
        num_hidden_layers=cfg.num_hidden_layers,
        num_attention_heads=cfg.num_attention_heads,
        num_key_value_heads=cfg.num_key_value_heads,
        rms_norm_eps=cfg.rms_norm_eps, rope_theta=cfg.rope_theta,
        max_position_embeddings=cfg.max_position_embeddings,
        tie_word_embeddings=cfg.tie_word_embeddings,
        attn_implementation="eager",
    )
    torch.manual_seed(3)
    return Qwen2ForCausalLM(hf_cfg).eval()   # qkv biases ARE random-init


def hf_mistral(cfg):
    from transformers import MistralConfig, MistralForCausalLM

    hf_cfg = MistralConfig(
        vocab_size=cfg.vocab_size, hidden_size=cfg.hidden_size,
        intermediate_size=cfg.intermediate_size,
        num_hidden_layers=cfg.num_hidden_layers,
        num_attention_heads=cfg.num_attention_heads,
        num_key_value_heads=cfg.num_key_value_heads,
        head_dim=cfg.head_dim, rms_norm_eps=cfg.rms_norm_eps,
        rope_theta=cfg.rope_theta,
        max_position_embeddings=cfg.max_position_embeddings,
        tie_word_embeddings=cfg.tie_word_embeddings,
        sliding_window=cfg.sliding_window,   # 8 < test seq: window active
        attn_implementation="eager",
    )
    torch.manual_seed(5)
    return MistralForCausalLM(hf_cfg).eval()


def np_weights_from_hf(hf_model):
    return {k: v.detach().to(torch.float32).numpy()
            for k, v in hf_model.state_dict().items()}


@pytest.mark.parametrize("preset,builder", [
    ("tiny-llama", hf_llama),
    ("tiny-gemma2", hf_gemma2),
    ("tiny-qwen2", hf_qwen2),
    ("tiny-mistral", hf_mistral),
])
def test_forward_matches_transformers(preset, builder):
    cfg = preset_config(preset)
    hf = builder(cfg)
    model = NumpyModel(cfg, np_weights_from_hf(hf))

    rng = np.random.default_rng(0)
    ids = rng.integers(0, cfg.vocab_size, size=12)
    with torch.no_grad():
        ref = hf(torch.tensor(ids[None])).logits[0].numpy()

    cache = NumpyKVCache(cfg, 64)
    got = model.forward(ids, cache, 0)

    np.testing.assert_allclose(got, ref, rtol=2e-4, atol=2e-4)


@pytest.mark.parametrize("preset,builder", [
    ("tiny-llama", hf_llama),
    ("tiny-gemma2", hf_gemma2),
    ("tiny-qwen2", hf_qwen2),
    ("tiny-mistral", hf_mistral),
])
def test_incremental_decode_matches_prefill(preset, builder):
    """KV-cache decode path == full-prefill logits (cache correctness)."""
    cfg = preset_config(preset)
    hf = builder(cfg)
    model = NumpyModel(cfg, np_weights_from_hf(hf))

    rng = np.random.default_rng(1)
    ids = rng.integers(0, cfg.vocab_size, size=10)

    cache_full = NumpyKVCache(cfg, 64)
    full = model.forward(ids, cache_full, 0)

    cache_inc = NumpyKVCache(cfg, 64)
    model.forward(ids[:4], cache_inc, 0)
    out = None
    for t in range(4, 10):
        out = model.forward(ids[t:t + 1], cache_inc, t)
    np.testing.assert_allclose(out[0], full[-1], rtol=1e-4, atol=1e-4)


def test_sliding_window_limits_attention():
    """Tokens beyond the sliding window must not influence sliding layers.

    Uses a 1-layer sliding-only Gemma config: logits at the last position
    must be identical whether the far-past tokens differ, once they fall
    outside the window."""
    from llm_np_cp_amd.core.config import ModelConfig
    from llm_np_cp_amd.io.loader import random_weights

    cfg = ModelConfig(
        model_type="gemma2", vocab_size=128, hidden_size=32,
        intermediate_size=64, num_hidden_layers=1, num_attention_heads=2,
        num_key_value_heads=1, head_dim=16, rope_theta=10000.0,
        hidden_act="gelu_pytorch_tanh", query_pre_attn_scalar=16,
        sliding_window=4, attn_logit_softcapping=50.0,
        final_logit_softcapping=30.0,
        layer_types=["sliding_attention"],
    )
    w = random_weights(cfg, seed=3)
    model = NumpyModel(cfg, w)

    rng = np.random.default_rng(4)
    tail = rng.integers(0, 128, size=4)
    a = np.concatenate([rng.integers(0, 128, size=6), tail])
    b = np.concatenate([rng.integers(0, 128, size=6), tail])

    la = model.forward(a, NumpyKVCache(cfg, 32), 0)
    lb = model.forward(b, NumpyKVCache(cfg, 32), 0)
    # last query position attends only to the last 4 keys -> identical
    np.testing.assert_allclose(la[-1], lb[-1], rtol=1e-6, atol=1e-6)
    # sanity: an in-window difference does change logits
    assert not np.allclose(la[-2], lb[-2])


def test_llama3_rope_scaling_matches_transformers():
    """llama3 rope_scaling (which the reference ignored) vs HF at
    positions beyond original_max_position_embeddings/factor."""
    from transformers import LlamaConfig, LlamaForCausalLM

    cfg = preset_config("tiny-llama")
    cfg.rope_theta = 500000.0
    cfg.rope_scaling = {"rope_type": "llama3", "factor": 4.0,
                        "low_freq_factor": 1.0, "high_freq_factor": 4.0,
                        "original_max_position_embeddings": 16}
    hf_cfg = LlamaConfig(
        vocab_size=cfg.vocab_size, hidden_size=cfg.hidden_size,
        intermediate_size=cfg.intermediate_size,
        num_hidden_layers=cfg.num_hidden_layers,
        num_attention_heads=cfg.num_attention_heads,
        num_key_value_heads=cfg.num_key_value_heads,
        head_dim=cfg.head_dim, rms_norm_eps=cfg.rms_norm_eps,
        rope_theta=cfg.rope_theta, rope_scaling=dict(cfg.rope_scaling),
        max_position_embeddings=cfg.max_position_embeddings,
        tie_word_embeddings=True, attn_implementation="eager",
    )
    torch.manual_seed(11)
    hf = LlamaForCausalLM(hf_cfg).eval()
    model = NumpyModel(cfg, np_weights_from_hf(hf))

    rng = np.random.default_rng(12)
    ids = rng.integers(0, cfg.vocab_size, size=48)  # > 16/4 and > 16
    with torch.no_grad():
        ref = hf(torch.tensor(ids[None])).logits[0].numpy()
    got = model.forward(ids, NumpyKVCache(cfg, 64), 0)
    np.testing.assert_allclose(got, ref, rtol=3e-4, atol=3e-4)


def test_qwen2_bias_affects_oracle():
    """Qwen-2 family: qkv biases are loaded and change the forward."""
    import llm_np_cp_amd as L
    from llm_np_cp_amd.io.loader import random_weights
    from llm_np_cp_amd.models.numpy_ref import NumpyModel, NumpyKVCache

    cfg = L.preset_config("tiny-qwen2")
    w = random_weights(cfg, seed=1)
    assert "model.layers.0.self_attn.q_proj.bias" in w
    m = NumpyModel(cfg, dict(w))
    ids = np.arange(1, 6)
    a = m.forward(ids, NumpyKVCache(cfg, 16), 0)

    w2 = dict(w)
    for i in range(cfg.num_hidden_layers):
        for n in ("q", "k", "v"):
            key = f"model.layers.{i}.self_attn.{n}_proj.bias"
            w2[key] = np.zeros_like(w2[key])
    b = NumpyModel(cfg, w2).forward(ids, NumpyKVCache(cfg, 16), 0)
    assert not np.allclose(a, b)


def test_mistral_all_layer_sliding_window():
    """Mistral: sliding window active on EVERY layer (vs Gemma's
    alternating); the window changes long-prefix logits."""
    import llm_np_cp_amd as L
    from llm_np_cp_amd.io.loader import random_weights
    from llm_np_cp_amd.models.numpy_ref import NumpyModel, NumpyKVCache

    cfg = L.preset_config("tiny-mistral")
    assert all(cfg.is_sliding(i) for i in range(cfg.num_hidden_layers))
    w = random_weights(cfg, seed=2)
    m = NumpyModel(cfg, dict(w))
    rng = np.random.default_rng(3)
    ids = rng.integers(0, cfg.vocab_size, size=20)  # > window 8
    a = m.forward(ids, NumpyKVCache(cfg, 32), 0)

    cfg2 = L.preset_config("tiny-mistral")
    cfg2.sliding_window = None
    b = NumpyModel(cfg2, dict(w)).forward(ids, NumpyKVCache(cfg2, 32), 0)
    # early positions (inside the window) agree; late ones differ
    np.testing.assert_allclose(a[:8], b[:8], rtol=1e-5, atol=1e-6)
    assert not np.allclose(a[-1], b[-1])


def test_sliding_window_geq_seq_equals_full_attention():
    """window >= sequence length must degenerate to full attention
    (guards the windowed-scan bounds in the oracle and, transitively,
    the GPU kernels validated against it)."""
    import numpy as np
    from llm_np_cp_amd.core.config import preset_config
    from llm_np_cp_amd.io.loader import random_weights
    from llm_np_cp_amd.models.numpy_ref import NumpyModel, NumpyKVCache

    base = preset_config("tiny-mistral")
    w = random_weights(base, seed=5)
    ids = np.arange(1, 13)

    wide = preset_config("tiny-mistral")
    wide.sliding_window = 64          # > seq: no position masked out
    l_wide = NumpyModel(wide, w).forward(ids, NumpyKVCache(wide, 32), 0)

    off = preset_config("tiny-mistral")
    off.sliding_window = None         # layer_types already frozen in
    off.layer_types = ["full_attention"] * off.num_hidden_layers
    l_full = NumpyModel(off, w).forward(ids, NumpyKVCache(off, 32), 0)

    assert np.allclose(l_wide, l_full, atol=1e-5)

    tight = preset_config("tiny-mistral")  # window 8 < 12 really masks
    l_tight = NumpyModel(tight, w).forward(ids, NumpyKVCache(tight, 32), 0)
    assert not np.allclose(l_tight, l_full, atol=1e-3)


def hf_mixtral(cfg):
    from transformers import MixtralConfig, MixtralForCausalLM

    hf_cfg = MixtralConfig(
        vocab_size=cfg.vocab_size, hidden_size=cfg.hidden_size,
        intermediate_size=cfg.intermediate_size,
        num_hidden_layers=cfg.num_hidden_layers,
        num_attention_heads=cfg.num_attention_heads,
        num_key_value_heads=cfg.num_key_value_heads,
        head_dim=cfg.head_dim, rms_norm_eps=cfg.rms_norm_eps,
        rope_theta=cfg.rope_theta,
        max_position_embeddings=cfg.max_position_embeddings,
        num_local_experts=cfg.num_local_experts,
        num_experts_per_tok=cfg.num_experts_per_tok,
        tie_word_embeddings=cfg.tie_word_embeddings,
        attn_implementation="eager",
    )
    torch.manual_seed(11)
    return MixtralForCausalLM(hf_cfg).eval()


def np_weights_from_hf_mixtral(hf_model, cfg):
    """transformers>=5 stores fused per-expert tensors
    (mlp.experts.gate_up_proj (E,2I,H), .down_proj (E,H,I)); Mixtral hub
    checkpoints (and this repo) use block_sparse_moe.experts.{e}.w1/w3/w2
    — split them back out."""
    I = cfg.intermediate_size
    w = {}
    for k, v in hf_model.state_dict().items():
        a = v.detach().to(torch.float32).numpy()
        if ".mlp.gate.weight" in k:
            w[k.replace(".mlp.gate.", ".block_sparse_moe.gate.")] = a
        elif ".mlp.experts.gate_up_proj" in k:
            p = k.replace(".mlp.experts.gate_up_proj",
                          ".block_sparse_moe.experts")
            for e in range(a.shape[0]):
                w[f"{p}.{e}.w1.weight"] = a[e, :I]
                w[f"{p}.{e}.w3.weight"] = a[e, I:]
        elif ".mlp.experts.down_proj" in k:
            p = k.replace(".mlp.experts.down_proj",
                          ".block_sparse_moe.experts")
            for e in range(a.shape[0]):
                w[f"{p}.{e}.w2.weight"] = a[e]
        else:
            w[k] = a
    return w


def test_mixtral_moe_forward_matches_transformers():
    cfg = preset_config("tiny-mixtral")
    hf = hf_mixtral(cfg)
    model = NumpyModel(cfg, np_weights_from_hf_mixtral(hf, cfg))

    rng = np.random.default_rng(3)
    ids = rng.integers(0, cfg.vocab_size, size=12)
    with torch.no_grad():
        ref = hf(torch.tensor(ids[None])).logits[0].numpy()
    got = model.forward(ids, NumpyKVCache(cfg, 64), 0)
    np.testing.assert_allclose(got, ref, rtol=3e-4, atol=3e-4)


def test_mixtral_incremental_decode_matches_prefill():
    cfg = preset_config("tiny-mixtral")
    hf = hf_mixtral(cfg)
    model = NumpyModel(cfg, np_weights_from_hf_mixtral(hf, cfg))
    rng = np.random.default_rng(4)
    ids = rng.integers(0, cfg.vocab_size, size=10)
    full = model.forward(ids, NumpyKVCache(cfg, 64), 0)
    inc = NumpyKVCache(cfg, 64)
    model.forward(ids[:4], inc, 0)
    out = None
    for t in range(4, 10):
        out = model.forward(ids[t:t + 1], inc, t)
    np.testing.assert_allclose(out[0], full[-1], rtol=1e-4, atol=1e-4)


def test_mixtral_routing_is_sparse():
    """Different tokens route to different experts; zeroing an UNROUTED
    expert's weights must not change that token's output."""
    from llm_np_cp_amd.io.loader import random_weights
    cfg = preset_config("tiny-mixtral")
    w = random_weights(cfg, seed=8)
    model = NumpyModel(cfg, dict(w))
    h = np.asarray(np.random.default_rng(0).standard_normal(
        (1, cfg.hidden_size)), dtype=np.float32)
    p = "model.layers.0.block_sparse_moe"
    logits = h @ w[f"{p}.gate.weight"].T
    order = np.argsort(-logits[0])
    unused = order[-1]  # least-likely expert: not in top-2 (E=4)
    out0 = model._moe_mlp(0, h)
    w2 = dict(w)
    for nm in ("w1", "w2", "w3"):
        w2[f"{p}.experts.{unused}.{nm}.weight"] = \
            np.zeros_like(w[f"{p}.experts.{unused}.{nm}.weight"])
    out1 = NumpyModel(cfg, w2)._moe_mlp(0, h)
    np.testing.assert_allclose(out0, out1)
