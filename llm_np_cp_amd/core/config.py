"""Model configuration.

Capability parity with the reference's ``AttributeDict`` over raw HF
``config.json`` (``/root/reference/llama3.2_model.py:204-207,1068-1073``), but
typed, validated, and decoupled from any process-global state.  The HF
``config.json`` stays the architecture source of truth: any Llama-3.2 /
Gemma-2 checkpoint directory loads by name.

Unlike the reference we also honor fields it ignored:
``rope_scaling`` (llama3 long-context scaling, reference ignores it —
``llama3.2_model.py:39``), Gemma-2 ``sliding_window`` and
``attn_logit_softcapping`` (reference has them only as comments,
``gemma2_model.py:48,109``).
"""

from __future__ import annotations

import json
import math
import os
from dataclasses import dataclass, field
from typing import Optional


@dataclass
class ModelConfig:
    model_type: str  # "llama" | "gemma2"
    vocab_size: int
    hidden_size: int
    intermediate_size: int
    num_hidden_layers: int
    num_attention_heads: int
    num_key_value_heads: int
    head_dim: int
    rms_norm_eps: float = 1e-5
    rope_theta: float = 10000.0
    rope_scaling: Optional[dict] = None
    max_position_embeddings: int = 8192
    hidden_act: str = "silu"  # "silu" (Llama SwiGLU) | "gelu_pytorch_tanh" (Gemma GeGLU)
    tie_word_embeddings: bool = True
    bos_token_id: int = 1
    eos_token_id: int = 2
    attention_bias: bool = False  # Qwen-2(.5): bias on q/k/v projections
    # Mixtral sparse MoE: E experts per MLP, top-k routed per token
    # (0 = dense MLP)
    num_local_experts: int = 0
    num_experts_per_tok: int = 2
    # Gemma-2 specifics
    query_pre_attn_scalar: Optional[float] = None
    sliding_window: Optional[int] = None
    attn_logit_softcapping: Optional[float] = None
    final_logit_softcapping: Optional[float] = None
    # derived
    layer_types: list = field(default_factory=list)

    def __post_init__(self):
        if self.head_dim is None:
            self.head_dim = self.hidden_size // self.num_attention_heads
        if not self.layer_types:
            if self.model_type == "gemma2":
                # Gemma-2: even layers sliding, odd layers global (HF layer_types)
                self.layer_types = [
                    "sliding_attention" if i % 2 == 0 else "full_attention"
                    for i in range(self.num_hidden_layers)
                ]
            elif self.model_type == "mistral" and self.sliding_window:
                # Mistral v0.1/v0.2: sliding window on EVERY layer
                self.layer_types = (["sliding_attention"]
                                    * self.num_hidden_layers)
            else:
                self.layer_types = ["full_attention"] * self.num_hidden_layers

    @property
    def num_kv_groups(self) -> int:
        return self.num_attention_heads // self.num_key_value_heads

    @property
    def attn_scale(self) -> float:
        """Softmax scale. Gemma-2 uses query_pre_attn_scalar**-0.5 (the
        reference computed it but never used it, ``gemma2_model.py:434``;
        we apply it, matching HF semantics)."""
        if self.model_type == "gemma2" and self.query_pre_attn_scalar is not None:
            return self.query_pre_attn_scalar ** -0.5
        return self.head_dim ** -0.5

    @property
    def is_moe(self) -> bool:
        return self.num_local_experts > 0

    def is_sliding(self, layer_idx: int) -> bool:
        return (
            self.sliding_window is not None
            and self.layer_types[layer_idx] == "sliding_attention"
        )

    @property
    def embed_scale(self) -> float:
        """Gemma scales embeddings by sqrt(hidden) (``gemma2_model.py:738-739``)."""
        if self.model_type == "gemma2":
            return math.sqrt(self.hidden_size)
        return 1.0

    # ------------------------------------------------------------------
    SUPPORTED_FAMILIES = ("llama", "gemma2", "qwen2", "mistral",
                          "mixtral")

    @classmethod
    def from_hf_dict(cls, d: dict) -> "ModelConfig":
        model_type = d.get("model_type", "llama")
        if model_type not in cls.SUPPORTED_FAMILIES:
            raise ValueError(
                f"unsupported model_type {model_type!r}; this framework "
                f"implements the {'/'.join(cls.SUPPORTED_FAMILIES)} "
                f"decoder families (treating an unknown architecture as "
                f"Llama would silently produce wrong results)")
        required = ("vocab_size", "hidden_size", "intermediate_size",
                    "num_hidden_layers", "num_attention_heads")
        missing = [k for k in required if k not in d]
        if missing:
            raise ValueError(
                f"config.json is missing required fields {missing} — "
                f"not a decoder-LM config this framework can load")
        heads = d["num_attention_heads"]
        cfg = cls(
            model_type=model_type,
            vocab_size=d["vocab_size"],
            hidden_size=d["hidden_size"],
            intermediate_size=d["intermediate_size"],
            num_hidden_layers=d["num_hidden_layers"],
            num_attention_heads=heads,
            num_key_value_heads=d.get("num_key_value_heads", heads),
            head_dim=d.get("head_dim") or d["hidden_size"] // heads,
            rms_norm_eps=d.get("rms_norm_eps", 1e-5),
            rope_theta=d.get("rope_theta", 10000.0),
            rope_scaling=d.get("rope_scaling"),
            max_position_embeddings=d.get("max_position_embeddings", 8192),
            hidden_act=d.get("hidden_act", d.get("hidden_activation", "silu")),
            tie_word_embeddings=d.get("tie_word_embeddings", True),
            bos_token_id=d.get("bos_token_id", 1),
            eos_token_id=d.get("eos_token_id", 2),
            # HF Qwen2Attention hardwires qkv bias=True (no config key)
            attention_bias=d.get("attention_bias",
                                 model_type == "qwen2"),
            num_local_experts=d.get("num_local_experts", 0),
            num_experts_per_tok=d.get("num_experts_per_tok", 2),
            query_pre_attn_scalar=d.get("query_pre_attn_scalar"),
            sliding_window=d.get("sliding_window"),
            attn_logit_softcapping=d.get("attn_logit_softcapping"),
            final_logit_softcapping=d.get("final_logit_softcapping"),
            layer_types=d.get("layer_types", []),
        )
        return cfg

    @classmethod
    def from_json(cls, path: str) -> "ModelConfig":
        with open(path) as f:
            return cls.from_hf_dict(json.load(f))

    def to_hf_dict(self) -> dict:
        d = {
            "model_type": self.model_type,
            "vocab_size": self.vocab_size,
            "hidden_size": self.hidden_size,
            "intermediate_size": self.intermediate_size,
            "num_hidden_layers": self.num_hidden_layers,
            "num_attention_heads": self.num_attention_heads,
            "num_key_value_heads": self.num_key_value_heads,
            "head_dim": self.head_dim,
            "rms_norm_eps": self.rms_norm_eps,
            "rope_theta": self.rope_theta,
            "max_position_embeddings": self.max_position_embeddings,
            "hidden_act": self.hidden_act,
            "tie_word_embeddings": self.tie_word_embeddings,
            "bos_token_id": self.bos_token_id,
            "eos_token_id": self.eos_token_id,
        }
        if self.attention_bias:
            d["attention_bias"] = True
        if self.num_local_experts:
            d["num_local_experts"] = self.num_local_experts
            d["num_experts_per_tok"] = self.num_experts_per_tok
        if self.rope_scaling is not None:
            d["rope_scaling"] = self.rope_scaling
        for k in ("query_pre_attn_scalar", "sliding_window",
                  "attn_logit_softcapping", "final_logit_softcapping"):
            v = getattr(self, k)
            if v is not None:
                d[k] = v
        return d

    # ------------------------------------------------------------------
    def rope_inv_freq(self):
        """Inverse frequencies, honoring llama3 rope_scaling (which the
        reference ignored — SURVEY §2.4).  Returns a numpy fp64 array of
        length head_dim//2."""
        import numpy as np

        dim = self.head_dim
        inv_freq = 1.0 / (
            self.rope_theta ** (np.arange(0, dim, 2, dtype=np.float64) / dim)
        )
        rs = self.rope_scaling
        if rs and rs.get("rope_type", rs.get("type")) == "llama3":
            factor = rs["factor"]
            low = rs["low_freq_factor"]
            high = rs["high_freq_factor"]
            old_ctx = rs["original_max_position_embeddings"]
            low_wl = old_ctx / low
            high_wl = old_ctx / high
            wavelen = 2 * math.pi / inv_freq
            scaled = np.where(wavelen > low_wl, inv_freq / factor, inv_freq)
            smooth = (old_ctx / wavelen - low) / (high - low)
            smoothed = (1 - smooth) * scaled / factor + smooth * scaled
            mid = (wavelen >= high_wl) & (wavelen <= low_wl)
            inv_freq = np.where(mid, smoothed, scaled)
        return inv_freq


# ----------------------------------------------------------------------
# Preset architectures for synthetic/random-init runs (no network: shapes
# transcribed from the public HF config.json of each checkpoint).
# ----------------------------------------------------------------------

PRESETS = {
    "llama-3.2-1b": dict(
        model_type="llama", vocab_size=128256, hidden_size=2048,
        intermediate_size=8192, num_hidden_layers=16, num_attention_heads=32,
        num_key_value_heads=8, head_dim=64, rms_norm_eps=1e-5,
        rope_theta=500000.0,
        rope_scaling={"rope_type": "llama3", "factor": 32.0,
                      "low_freq_factor": 1.0, "high_freq_factor": 4.0,
                      "original_max_position_embeddings": 8192},
        max_position_embeddings=131072, hidden_act="silu",
        tie_word_embeddings=True, bos_token_id=128000, eos_token_id=128001,
    ),
    "llama-3.2-3b": dict(
        model_type="llama", vocab_size=128256, hidden_size=3072,
        intermediate_size=8192, num_hidden_layers=28, num_attention_heads=24,
        num_key_value_heads=8, head_dim=128, rms_norm_eps=1e-5,
        rope_theta=500000.0,
        rope_scaling={"rope_type": "llama3", "factor": 32.0,
                      "low_freq_factor": 1.0, "high_freq_factor": 4.0,
                      "original_max_position_embeddings": 8192},
        max_position_embeddings=131072, hidden_act="silu",
        tie_word_embeddings=True, bos_token_id=128000, eos_token_id=128001,
    ),
    "gemma-2-2b": dict(
        model_type="gemma2", vocab_size=256000, hidden_size=2304,
        intermediate_size=9216, num_hidden_layers=26, num_attention_heads=8,
        num_key_value_heads=4, head_dim=256, rms_norm_eps=1e-6,
        rope_theta=10000.0, max_position_embeddings=8192,
        hidden_act="gelu_pytorch_tanh", tie_word_embeddings=True,
        bos_token_id=2, eos_token_id=1,
        query_pre_attn_scalar=256, sliding_window=4096,
        attn_logit_softcapping=50.0, final_logit_softcapping=30.0,
    ),
    "gemma-2-9b": dict(
        model_type="gemma2", vocab_size=256000, hidden_size=3584,
        intermediate_size=14336, num_hidden_layers=42, num_attention_heads=16,
        num_key_value_heads=8, head_dim=256, rms_norm_eps=1e-6,
        rope_theta=10000.0, max_position_embeddings=8192,
        hidden_act="gelu_pytorch_tanh", tie_word_embeddings=True,
        bos_token_id=2, eos_token_id=1,
        query_pre_attn_scalar=256, sliding_window=4096,
        attn_logit_softcapping=50.0, final_logit_softcapping=30.0,
    ),
    "llama-3.1-8b": dict(
        model_type="llama", vocab_size=128256, hidden_size=4096,
        intermediate_size=14336, num_hidden_layers=32,
        num_attention_heads=32, num_key_value_heads=8, head_dim=128,
        rms_norm_eps=1e-5, rope_theta=500000.0,
        rope_scaling={"rope_type": "llama3", "factor": 8.0,
                      "low_freq_factor": 1.0, "high_freq_factor": 4.0,
                      "original_max_position_embeddings": 8192},
        max_position_embeddings=131072, hidden_act="silu",
        tie_word_embeddings=False, bos_token_id=128000,
        eos_token_id=128001,
    ),
    # Llama-3.1-70B: the 288 GB HBM3E sizing case — fp8 weights are
    # ~70 GB, so TP=1 fits with >200 GB left for KV; bf16 (141 GB)
    # also fits on ONE MI355X (impossible on 80-192 GB parts)
    "llama-3.1-70b": dict(
        model_type="llama", vocab_size=128256, hidden_size=8192,
        intermediate_size=28672, num_hidden_layers=80,
        num_attention_heads=64, num_key_value_heads=8, head_dim=128,
        rms_norm_eps=1e-5, rope_theta=500000.0,
        rope_scaling={"rope_type": "llama3", "factor": 8.0,
                      "low_freq_factor": 1.0, "high_freq_factor": 4.0,
                      "original_max_position_embeddings": 8192},
        max_position_embeddings=131072, hidden_act="silu",
        tie_word_embeddings=False, bos_token_id=128000,
        eos_token_id=128001,
    ),
    # Qwen-2.5: Llama-family arch + qkv bias + untied lm_head
    # (shape table from the public Qwen/Qwen2.5-7B config.json)
    "qwen2.5-7b": dict(
        model_type="qwen2", vocab_size=152064, hidden_size=3584,
        intermediate_size=18944, num_hidden_layers=28,
        num_attention_heads=28, num_key_value_heads=4, head_dim=128,
        rms_norm_eps=1e-6, rope_theta=1000000.0,
        max_position_embeddings=32768, hidden_act="silu",
        tie_word_embeddings=False, bos_token_id=151643,
        eos_token_id=151643, attention_bias=True,
    ),
    # Mistral: Llama-family arch + sliding window on every layer
    # (shape table from the public mistralai/Mistral-7B-v0.1 config.json)
    "mistral-7b": dict(
        model_type="mistral", vocab_size=32000, hidden_size=4096,
        intermediate_size=14336, num_hidden_layers=32,
        num_attention_heads=32, num_key_value_heads=8, head_dim=128,
        rms_norm_eps=1e-5, rope_theta=10000.0,
        max_position_embeddings=32768, hidden_act="silu",
        tie_word_embeddings=False, bos_token_id=1, eos_token_id=2,
        sliding_window=4096,
    ),
    "gemma-2-27b": dict(
        model_type="gemma2", vocab_size=256000, hidden_size=4608,
        intermediate_size=36864, num_hidden_layers=46,
        num_attention_heads=32, num_key_value_heads=16, head_dim=128,
        rms_norm_eps=1e-6, rope_theta=10000.0,
        max_position_embeddings=8192, hidden_act="gelu_pytorch_tanh",
        tie_word_embeddings=True, bos_token_id=2, eos_token_id=1,
        query_pre_attn_scalar=144, sliding_window=4096,
        attn_logit_softcapping=50.0, final_logit_softcapping=30.0,
    ),
    # Mixtral 8x7B sparse MoE: 8 experts, top-2 routing, GQA
    # (shape table from the public mistralai/Mixtral-8x7B-v0.1 config.json)
    "mixtral-8x7b": dict(
        model_type="mixtral", vocab_size=32000, hidden_size=4096,
        intermediate_size=14336, num_hidden_layers=32,
        num_attention_heads=32, num_key_value_heads=8, head_dim=128,
        rms_norm_eps=1e-5, rope_theta=1000000.0,
        max_position_embeddings=32768, hidden_act="silu",
        tie_word_embeddings=False, bos_token_id=1, eos_token_id=2,
        num_local_experts=8, num_experts_per_tok=2,
    ),
    # tiny configs for tests
    "tiny-mixtral": dict(       # sparse MoE: 4 experts, top-2
        model_type="mixtral", vocab_size=512, hidden_size=64,
        intermediate_size=128, num_hidden_layers=2, num_attention_heads=4,
        num_key_value_heads=2, head_dim=16, rms_norm_eps=1e-5,
        rope_theta=10000.0, max_position_embeddings=512, hidden_act="silu",
        tie_word_embeddings=True, num_local_experts=4,
        num_experts_per_tok=2,
    ),
    "tiny-llama-hd64": dict(    # exercises the hd=64 MFMA prefill path
        model_type="llama", vocab_size=512, hidden_size=128,
        intermediate_size=256, num_hidden_layers=2, num_attention_heads=2,
        num_key_value_heads=1, head_dim=64, rms_norm_eps=1e-5,
        rope_theta=10000.0, max_position_embeddings=512, hidden_act="silu",
        tie_word_embeddings=True,
    ),
    "tiny-llama-tp4": dict(  # TP=4-friendly (kvh=4, sharded K %64)
        model_type="llama", vocab_size=512, hidden_size=256,
        intermediate_size=512, num_hidden_layers=2, num_attention_heads=8,
        num_key_value_heads=4, head_dim=64, rms_norm_eps=1e-5,
        rope_theta=10000.0, max_position_embeddings=512, hidden_act="silu",
        tie_word_embeddings=True,
    ),
    "tiny-llama-tp": dict(   # TP=2-friendly: sharded o/down K stays %64
        model_type="llama", vocab_size=512, hidden_size=256,
        intermediate_size=256, num_hidden_layers=2, num_attention_heads=4,
        num_key_value_heads=2, head_dim=64, rms_norm_eps=1e-5,
        rope_theta=10000.0, max_position_embeddings=512, hidden_act="silu",
        tie_word_embeddings=True,
    ),
    "tiny-llama": dict(
        model_type="llama", vocab_size=512, hidden_size=64,
        intermediate_size=128, num_hidden_layers=2, num_attention_heads=4,
        num_key_value_heads=2, head_dim=16, rms_norm_eps=1e-5,
        rope_theta=10000.0, max_position_embeddings=512, hidden_act="silu",
        tie_word_embeddings=True,
    ),
    "tiny-qwen2": dict(         # llama arch + qkv bias
        model_type="qwen2", vocab_size=512, hidden_size=64,
        intermediate_size=128, num_hidden_layers=2, num_attention_heads=4,
        num_key_value_heads=2, head_dim=16, rms_norm_eps=1e-6,
        rope_theta=10000.0, max_position_embeddings=512, hidden_act="silu",
        tie_word_embeddings=True, attention_bias=True,
    ),
    "tiny-mistral": dict(       # llama arch + ALL-layer sliding window
        model_type="mistral", vocab_size=512, hidden_size=128,
        intermediate_size=256, num_hidden_layers=2, num_attention_heads=2,
        num_key_value_heads=1, head_dim=64, rms_norm_eps=1e-5,
        rope_theta=10000.0, max_position_embeddings=512, hidden_act="silu",
        tie_word_embeddings=True, sliding_window=8,
    ),
    "tiny-gemma2-hd64": dict(   # exercises gemma + MFMA prefill (hd=64)
        model_type="gemma2", vocab_size=512, hidden_size=128,
        intermediate_size=256, num_hidden_layers=4, num_attention_heads=2,
        num_key_value_heads=1, head_dim=64, rms_norm_eps=1e-6,
        rope_theta=10000.0, max_position_embeddings=512,
        hidden_act="gelu_pytorch_tanh", tie_word_embeddings=True,
        query_pre_attn_scalar=64, sliding_window=8,
        attn_logit_softcapping=50.0, final_logit_softcapping=30.0,
    ),
    "tiny-gemma2": dict(
        model_type="gemma2", vocab_size=512, hidden_size=64,
        intermediate_size=128, num_hidden_layers=4, num_attention_heads=4,
        num_key_value_heads=2, head_dim=16, rms_norm_eps=1e-6,
        rope_theta=10000.0, max_position_embeddings=512,
        hidden_act="gelu_pytorch_tanh", tie_word_embeddings=True,
        query_pre_attn_scalar=16, sliding_window=8,
        attn_logit_softcapping=50.0, final_logit_softcapping=30.0,
    ),
}


def preset_config(name: str) -> ModelConfig:
    key = name.lower()
    if key not in PRESETS:
        raise KeyError(f"unknown preset {name!r}; have {sorted(PRESETS)}")
    return ModelConfig(**PRESETS[key])
