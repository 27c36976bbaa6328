"""Config round-trip, sharded-checkpoint, and sampler-distribution tests."""

import json
import os

import numpy as np
import pytest

import llm_np_cp_amd as L
from llm_np_cp_amd.core.config import ModelConfig, preset_config


def test_config_json_roundtrip(tmp_path):
    for name in ("llama-3.2-1b", "gemma-2-9b"):
        cfg = preset_config(name)
        p = tmp_path / f"{name}.json"
        p.write_text(json.dumps(cfg.to_hf_dict()))
        back = ModelConfig.from_json(str(p))
        assert back.to_hf_dict() == cfg.to_hf_dict()
        assert back.attn_scale == cfg.attn_scale
        assert back.layer_types == cfg.layer_types


def test_preset_shapes_match_hf_published():
    """Spot-check the preset architecture tables."""
    c = preset_config("llama-3.2-1b")
    assert (c.vocab_size, c.hidden_size, c.num_hidden_layers) == \
        (128256, 2048, 16)
    assert (c.num_attention_heads, c.num_key_value_heads, c.head_dim) == \
        (32, 8, 64)
    g = preset_config("gemma-2-9b")
    assert (g.vocab_size, g.hidden_size, g.num_hidden_layers) == \
        (256000, 3584, 42)
    assert g.head_dim == 256 and g.sliding_window == 4096
    assert g.layer_types[0] == "sliding_attention"
    assert g.layer_types[1] == "full_attention"


def test_sharded_safetensors_checkpoint(tmp_path):
    """Two-shard checkpoint with an index.json loads identically to the
    single-file path (reference weight_map capability)."""
    import torch
    from safetensors.torch import save_file

    from llm_np_cp_amd.io.loader import load_weights_numpy, random_weights

    cfg = preset_config("tiny-llama")
    w = random_weights(cfg, seed=77)
    d = str(tmp_path / "ck")
    os.makedirs(d)
    names = sorted(w)
    half = len(names) // 2
    save_file({k: torch.from_numpy(w[k]) for k in names[:half]},
              os.path.join(d, "model-00001-of-00002.safetensors"))
    save_file({k: torch.from_numpy(w[k]) for k in names[half:]},
              os.path.join(d, "model-00002-of-00002.safetensors"))
    weight_map = {k: ("model-00001-of-00002.safetensors" if i < half else
                      "model-00002-of-00002.safetensors")
                  for i, k in enumerate(names)}
    with open(os.path.join(d, "model.safetensors.index.json"), "w") as f:
        json.dump({"weight_map": weight_map}, f)
    with open(os.path.join(d, "config.json"), "w") as f:
        json.dump(cfg.to_hf_dict(), f)

    got = load_weights_numpy(d)
    assert set(got) == set(w)
    for k in names:
        np.testing.assert_array_equal(got[k], w[k])


def test_missing_checkpoint_raises(tmp_path):
    from llm_np_cp_amd.io.loader import iter_safetensors

    with pytest.raises(FileNotFoundError):
        list(iter_safetensors(str(tmp_path)))


def test_min_p_sampling_distribution():
    """Sampled frequencies track the renormalized min-p distribution."""
    from llm_np_cp_amd.runtime.sampling import SamplingParams, sample_token

    logits = np.log(np.array([0.5, 0.3, 0.15, 0.05], dtype=np.float32))
    # p_max=0.5; min_p=0.09 -> threshold 0.045: all four kept (the 0.05
    # token sits exactly on the 0.1 threshold modulo fp rounding)
    rng = np.random.default_rng(0)
    p = SamplingParams(strategy="min_p", min_p=0.09)
    counts = np.zeros(4)
    n = 4000
    for _ in range(n):
        counts[sample_token(logits, p, rng)] += 1
    freq = counts / n
    np.testing.assert_allclose(freq, [0.5, 0.3, 0.15, 0.05], atol=0.03)


def test_lazy_random_weights_match_shapes():
    from llm_np_cp_amd.io.loader import (LazyRandomWeights, hf_weight_shapes)

    cfg = preset_config("tiny-gemma2")
    lw = LazyRandomWeights(cfg, seed=3)
    shapes = hf_weight_shapes(cfg)
    for name, shape in shapes.items():
        assert lw[name].shape == shape
    # norm gammas near gemma's stored-gamma convention (~0)
    g = lw["model.layers.0.input_layernorm.weight"]
    assert abs(g.mean()) < 0.05


def test_additional_presets_construct():
    from llm_np_cp_amd.io.loader import hf_weight_shapes

    for name in ("llama-3.1-8b", "gemma-2-27b", "llama-3.2-3b"):
        c = preset_config(name)
        shapes = hf_weight_shapes(c)
        n = sum(int(np.prod(s)) for s in shapes.values())
        assert n > 1e9
    assert "lm_head.weight" in hf_weight_shapes(preset_config("llama-3.1-8b"))


def test_hf_config_field_variants():
    """Real checkpoint config.json variants: Gemma-2 uses
    'hidden_activation'; older Llama rope_scaling uses 'type'."""
    d = {
        "model_type": "gemma2", "vocab_size": 256, "hidden_size": 64,
        "intermediate_size": 128, "num_hidden_layers": 2,
        "num_attention_heads": 4, "num_key_value_heads": 2,
        "head_dim": 16, "hidden_activation": "gelu_pytorch_tanh",
        "query_pre_attn_scalar": 16, "sliding_window": 8,
    }
    cfg = ModelConfig.from_hf_dict(d)
    assert cfg.hidden_act == "gelu_pytorch_tanh"
    assert cfg.is_sliding(0) and not cfg.is_sliding(1)

    d2 = {
        "model_type": "llama", "vocab_size": 256, "hidden_size": 64,
        "intermediate_size": 128, "num_hidden_layers": 1,
        "num_attention_heads": 4, "rope_theta": 500000.0,
        "rope_scaling": {"type": "llama3", "factor": 8.0,
                         "low_freq_factor": 1.0, "high_freq_factor": 4.0,
                         "original_max_position_embeddings": 8192},
    }
    cfg2 = ModelConfig.from_hf_dict(d2)
    assert cfg2.head_dim == 16  # derived hidden/heads
    plain = ModelConfig.from_hf_dict({**d2, "rope_scaling": None})
    assert np.any(cfg2.rope_inv_freq() < plain.rope_inv_freq())


def test_all_presets_hf_dict_roundtrip():
    """Every preset survives to_hf_dict -> from_hf_dict (config drift
    guard across llama/gemma2/qwen2/mistral families)."""
    import llm_np_cp_amd as L
    from llm_np_cp_amd.core.config import PRESETS, ModelConfig

    for name in PRESETS:
        a = L.preset_config(name)
        b = ModelConfig.from_hf_dict(a.to_hf_dict())
        for f in ("model_type", "vocab_size", "hidden_size",
                  "intermediate_size", "num_hidden_layers",
                  "num_attention_heads", "num_key_value_heads", "head_dim",
                  "rms_norm_eps", "rope_theta", "hidden_act",
                  "tie_word_embeddings", "attention_bias",
                  "query_pre_attn_scalar", "sliding_window",
                  "attn_logit_softcapping", "final_logit_softcapping"):
            assert getattr(a, f) == getattr(b, f), (name, f)
        assert a.layer_types == b.layer_types, name


def test_real_presets_tp_divisibility():
    """The TP sharding preconditions (engine asserts) hold for the real
    model presets at their natural TP degrees."""
    import llm_np_cp_amd as L

    expect = {
        "llama-3.2-1b": [1, 2, 4, 8],
        "llama-3.2-3b": [1, 2, 4, 8],
        "llama-3.1-8b": [1, 2, 4, 8],
        "gemma-2-2b": [1, 2, 4],
        "gemma-2-9b": [1, 2, 4, 8],
        "gemma-2-27b": [1, 2, 4, 8],
        "qwen2.5-7b": [1, 2, 4],
        "mistral-7b": [1, 2, 4, 8],
    }
    for name, tps in expect.items():
        cfg = L.preset_config(name)
        for tp in tps:
            assert cfg.num_attention_heads % tp == 0, (name, tp)
            assert cfg.num_key_value_heads % tp == 0, (name, tp)
            assert cfg.intermediate_size % tp == 0, (name, tp)
            assert cfg.vocab_size % tp == 0, (name, tp)


# ---------- real hub config.json fixtures (verbatim field sets) ----------

LLAMA_32_1B_CONFIG = {
    "architectures": ["LlamaForCausalLM"], "attention_bias": False,
    "attention_dropout": 0.0, "bos_token_id": 128000,
    "eos_token_id": 128001, "head_dim": 64, "hidden_act": "silu",
    "hidden_size": 2048, "initializer_range": 0.02,
    "intermediate_size": 8192, "max_position_embeddings": 131072,
    "mlp_bias": False, "model_type": "llama", "num_attention_heads": 32,
    "num_hidden_layers": 16, "num_key_value_heads": 8,
    "pretraining_tp": 1, "rms_norm_eps": 1e-05,
    "rope_scaling": {"factor": 32.0, "high_freq_factor": 4.0,
                     "low_freq_factor": 1.0,
                     "original_max_position_embeddings": 8192,
                     "rope_type": "llama3"},
    "rope_theta": 500000.0, "tie_word_embeddings": True,
    "torch_dtype": "bfloat16", "use_cache": True, "vocab_size": 128256,
}

GEMMA_2_9B_CONFIG = {
    "architectures": ["Gemma2ForCausalLM"],
    "attention_bias": False, "attention_dropout": 0.0,
    "attn_logit_softcapping": 50.0, "bos_token_id": 2,
    "cache_implementation": "hybrid", "eos_token_id": 1,
    "final_logit_softcapping": 30.0, "head_dim": 256,
    "hidden_act": "gelu_pytorch_tanh",
    "hidden_activation": "gelu_pytorch_tanh", "hidden_size": 3584,
    "initializer_range": 0.02, "intermediate_size": 14336,
    "max_position_embeddings": 8192, "model_type": "gemma2",
    "num_attention_heads": 16, "num_hidden_layers": 42,
    "num_key_value_heads": 8, "pad_token_id": 0,
    "query_pre_attn_scalar": 256, "rms_norm_eps": 1e-06,
    "rope_theta": 10000.0, "sliding_window": 4096,
    "torch_dtype": "float32", "use_cache": True, "vocab_size": 256000,
}

MIXTRAL_8X7B_CONFIG = {
    "architectures": ["MixtralForCausalLM"], "attention_dropout": 0.0,
    "bos_token_id": 1, "eos_token_id": 2, "hidden_act": "silu",
    "hidden_size": 4096, "initializer_range": 0.02,
    "intermediate_size": 14336, "max_position_embeddings": 32768,
    "model_type": "mixtral", "num_attention_heads": 32,
    "num_experts_per_tok": 2, "num_hidden_layers": 32,
    "num_key_value_heads": 8, "num_local_experts": 8,
    "output_router_logits": False, "rms_norm_eps": 1e-05,
    "rope_theta": 1000000.0, "router_aux_loss_coef": 0.02,
    "sliding_window": None, "tie_word_embeddings": False,
    "torch_dtype": "bfloat16", "use_cache": True, "vocab_size": 32000,
}

QWEN_25_7B_CONFIG = {
    "architectures": ["Qwen2ForCausalLM"], "attention_dropout": 0.0,
    "bos_token_id": 151643, "eos_token_id": 151643,
    "hidden_act": "silu", "hidden_size": 3584,
    "initializer_range": 0.02, "intermediate_size": 18944,
    "max_position_embeddings": 131072, "max_window_layers": 28,
    "model_type": "qwen2", "num_attention_heads": 28,
    "num_hidden_layers": 28, "num_key_value_heads": 4,
    "rms_norm_eps": 1e-06, "rope_theta": 1000000.0,
    "sliding_window": None, "tie_word_embeddings": False,
    "torch_dtype": "bfloat16", "use_cache": True,
    "use_sliding_window": False, "vocab_size": 152064,
}


def test_real_llama32_config_parses():
    """An actual Llama-3.2-1B hub config.json (extra keys and all)
    parses to the right architecture, and llama3 rope_scaling survives."""
    from llm_np_cp_amd.core.config import ModelConfig

    cfg = ModelConfig.from_hf_dict(LLAMA_32_1B_CONFIG)
    assert (cfg.hidden_size, cfg.num_hidden_layers) == (2048, 16)
    assert (cfg.num_attention_heads, cfg.num_key_value_heads) == (32, 8)
    assert cfg.head_dim == 64 and cfg.vocab_size == 128256
    assert cfg.tie_word_embeddings and not cfg.attention_bias
    assert cfg.rope_scaling["rope_type"] == "llama3"
    f = cfg.rope_inv_freq()
    assert f.shape == (32,)
    assert not cfg.is_moe and cfg.attn_scale == 64 ** -0.5


def test_real_gemma2_config_parses():
    from llm_np_cp_amd.core.config import ModelConfig

    cfg = ModelConfig.from_hf_dict(GEMMA_2_9B_CONFIG)
    assert cfg.model_type == "gemma2" and cfg.head_dim == 256
    assert cfg.attn_logit_softcapping == 50.0
    assert cfg.final_logit_softcapping == 30.0
    assert cfg.attn_scale == 256 ** -0.5  # query_pre_attn_scalar
    assert cfg.embed_scale == pytest.approx(3584 ** 0.5)
    assert cfg.hidden_act == "gelu_pytorch_tanh"
    # alternating sliding layers, layer 0 sliding
    assert cfg.is_sliding(0) and not cfg.is_sliding(1)
    assert sum(cfg.is_sliding(i) for i in range(42)) == 21


def test_real_mixtral_config_parses():
    from llm_np_cp_amd.core.config import ModelConfig

    cfg = ModelConfig.from_hf_dict(MIXTRAL_8X7B_CONFIG)
    assert cfg.is_moe
    assert (cfg.num_local_experts, cfg.num_experts_per_tok) == (8, 2)
    assert not cfg.tie_word_embeddings
    assert cfg.sliding_window is None
    assert not any(cfg.is_sliding(i) for i in range(32))
    assert cfg.head_dim == 4096 // 32  # derived (no head_dim key)


def test_real_qwen25_config_parses():
    from llm_np_cp_amd.core.config import ModelConfig

    cfg = ModelConfig.from_hf_dict(QWEN_25_7B_CONFIG)
    assert cfg.attention_bias  # HF Qwen2Attention hardwires qkv bias
    assert cfg.head_dim == 3584 // 28
    assert cfg.num_kv_groups == 7
    # use_sliding_window false => sliding_window None => no windowing
    assert not any(cfg.is_sliding(i) for i in range(28))


def test_validate_weights_actionable_errors():
    """Missing tensors and shape mismatches produce a clear ValueError
    naming the offenders (vs a KeyError deep in the forward)."""
    import numpy as np
    from llm_np_cp_amd.core.config import preset_config
    from llm_np_cp_amd.io.loader import (LazyRandomWeights,
                                         random_weights, validate_weights)
    from llm_np_cp_amd.models.numpy_ref import NumpyModel

    cfg = preset_config("tiny-llama")
    w = random_weights(cfg, seed=0)
    validate_weights(cfg, w)  # complete: no raise
    validate_weights(cfg, LazyRandomWeights(cfg))  # lazy table trusted

    w2 = dict(w)
    del w2["model.layers.1.mlp.down_proj.weight"]
    with pytest.raises(ValueError, match="missing.*down_proj"):
        NumpyModel(cfg, w2)

    w3 = dict(w)
    w3["model.norm.weight"] = np.zeros(7, dtype=np.float32)
    with pytest.raises(ValueError, match="shape mismatch.*model.norm"):
        validate_weights(cfg, w3)

    # extra tensors are fine (real checkpoints ship rotary buffers)
    w4 = dict(w)
    w4["model.rotary_emb.inv_freq"] = np.zeros(8, dtype=np.float32)
    validate_weights(cfg, w4)

    # mixtral naming goes through the same table
    mcfg = preset_config("tiny-mixtral")
    mw = random_weights(mcfg, seed=1)
    validate_weights(mcfg, mw)
    del mw["model.layers.0.block_sparse_moe.experts.2.w3.weight"]
    with pytest.raises(ValueError, match="experts.2.w3"):
        validate_weights(mcfg, mw)


def test_llama_70b_preset_shapes_and_memory_math():
    """The 288 GB sizing preset: weight-byte math confirms fp8 70B fits
    TP=1 with KV headroom (and bf16 fits at all — the MI355X story)."""
    from llm_np_cp_amd.core.config import preset_config
    from llm_np_cp_amd.io.loader import hf_weight_shapes

    cfg = preset_config("llama-3.1-70b")
    assert (cfg.hidden_size, cfg.num_hidden_layers) == (8192, 80)
    assert cfg.num_kv_groups == 8
    shapes = hf_weight_shapes(cfg)
    n_params = sum(int(np.prod(s)) for s in shapes.values())
    assert 69e9 < n_params < 72e9          # ~70.6B
    assert n_params * 1 / 2**30 < 70       # fp8 bytes fit easily
    assert n_params * 2 / 2**30 < 288      # bf16 fits in HBM too
    # every TP degree the node offers divides the head/vocab dims
    for tp in (1, 2, 4, 8):
        assert cfg.num_key_value_heads % tp == 0
        assert cfg.vocab_size % tp == 0
        assert cfg.intermediate_size % tp == 0


def test_from_hf_dict_rejects_foreign_configs():
    """A non-decoder or unknown-architecture config.json fails with an
    actionable error instead of silently parsing as Llama."""
    from llm_np_cp_amd.core.config import ModelConfig

    with pytest.raises(ValueError, match="unsupported model_type 'gpt2'"):
        ModelConfig.from_hf_dict({"model_type": "gpt2",
                                  "vocab_size": 50257, "hidden_size": 768,
                                  "intermediate_size": 3072,
                                  "num_hidden_layers": 12,
                                  "num_attention_heads": 12})
    with pytest.raises(ValueError, match="missing required fields"):
        ModelConfig.from_hf_dict({"model_type": "llama",
                                  "vocab_size": 128256})
