"""Pure-NumPy reference implementation of the decoder-LM forward pass
(Llama-3.2 / Gemma-2 / Qwen-2 / Mistral / Mixtral sparse-MoE).

This is the in-repo oracle (the role HF ``transformers`` played for the
reference) and BASELINE config 0 (the reference's NumPy path,
``/root/reference/llama3.2_model_numpy.py``).  Deliberate fixes over the
reference (SURVEY §7 stage 1):

- numerically **stable** softmax everywhere (the reference's live NumPy
  softmax was unstabilized, ``llama3.2_model_numpy.py:915-919``);
- causal mask applied for every q_len (reference skipped it at q_len<=2,
  ``llama3.2_model.py:471``);
- llama3 ``rope_scaling`` honored (reference ignored it);
- Gemma-2 sliding-window attention and attention-logit soft-capping
  implemented (reference omitted both);
- decode feeds back token *ids*, not re-tokenized text (generate loop).

All math is fp32 (weights are loaded/cast to fp32), accumulation in fp32.
Weights are HF-state-dict-keyed, matching the reference's capability of
building the model directly from an HF checkpoint (``llama3.2_model.py:
156-160,369-372``) without the reference's module-global ``weights`` dict.
"""

from __future__ import annotations

import math
from typing import Dict, Optional

import numpy as np

from ..core.config import ModelConfig


def softmax(x: np.ndarray, axis: int = -1) -> np.ndarray:
    m = np.max(x, axis=axis, keepdims=True)
    e = np.exp(x - m)
    return e / np.sum(e, axis=axis, keepdims=True)


def silu(x: np.ndarray) -> np.ndarray:
    return x / (1.0 + np.exp(-x))


def gelu_tanh(x: np.ndarray) -> np.ndarray:
    return 0.5 * x * (1.0 + np.tanh(
        math.sqrt(2.0 / math.pi) * (x + 0.044715 * x ** 3)))


ACT2FN = {"silu": silu, "gelu_pytorch_tanh": gelu_tanh, "gelu": gelu_tanh}


class NumpyKVCache:
    """Preallocated per-layer KV cache, layout (kv_heads, max_seq, head_dim).

    Replaces the reference's O(T^2) concat cache (``llama3.2_model.py:
    303-332``) with in-place writes into a fixed pool."""

    def __init__(self, config: ModelConfig, max_seq: int = 2048):
        self.max_seq = max_seq
        kvh, hd = config.num_key_value_heads, config.head_dim
        L = config.num_hidden_layers
        self.k = np.zeros((L, kvh, max_seq, hd), dtype=np.float32)
        self.v = np.zeros((L, kvh, max_seq, hd), dtype=np.float32)
        self.seq_len = 0  # filled length (same for all layers)

    def update(self, layer: int, k: np.ndarray, v: np.ndarray, pos: int):
        """k, v: (kv_heads, q_len, head_dim) written at [pos, pos+q_len)."""
        q_len = k.shape[1]
        if pos + q_len > self.max_seq:
            raise ValueError(f"KV cache overflow: {pos}+{q_len} > {self.max_seq}")
        self.k[layer][:, pos:pos + q_len] = k
        self.v[layer][:, pos:pos + q_len] = v


class NumpyModel:
    """Decoder-only causal LM: Llama-3.2 (pre-norm, SwiGLU) or Gemma-2
    (sandwich norm, GeGLU, soft-caps, sliding window), selected by config.

    ``weights``: dict of HF-named fp32 numpy arrays (``model.layers.{i}.*``).
    """

    def __init__(self, config: ModelConfig, weights: Dict[str, np.ndarray]):
        from ..io.loader import validate_weights

        validate_weights(config, weights)  # actionable error vs KeyError
        self.config = config
        self.w = weights
        if "lm_head.weight" not in self.w and config.tie_word_embeddings:
            self.w["lm_head.weight"] = self.w["model.embed_tokens.weight"]
        self.inv_freq = config.rope_inv_freq()  # fp64, (head_dim//2,)

    # -- building blocks -------------------------------------------------
    def _rmsnorm(self, x: np.ndarray, key: str) -> np.ndarray:
        g = self.w[key].astype(np.float64)
        xx = x.astype(np.float64)
        var = np.mean(xx * xx, axis=-1, keepdims=True)
        n = xx / np.sqrt(var + self.config.rms_norm_eps)
        if self.config.model_type == "gemma2":
            n = n * (1.0 + g)  # Gemma gamma+1 (gemma2_model.py:334)
        else:
            n = n * g
        return n.astype(np.float32)

    def _rope(self, x: np.ndarray, pos0: int) -> np.ndarray:
        """x: (heads, q_len, head_dim); rotate-half RoPE at positions
        pos0..pos0+q_len."""
        q_len, hd = x.shape[1], x.shape[2]
        t = np.arange(pos0, pos0 + q_len, dtype=np.float64)
        freqs = np.outer(t, self.inv_freq)              # (q_len, hd/2)
        emb = np.concatenate([freqs, freqs], axis=-1)   # (q_len, hd)
        cos = np.cos(emb)[None, :, :]
        sin = np.sin(emb)[None, :, :]
        x1, x2 = x[..., : hd // 2], x[..., hd // 2:]
        rot = np.concatenate([-x2, x1], axis=-1)
        return (x * cos + rot * sin).astype(np.float32)

    def _attention(self, layer: int, h: np.ndarray, cache: NumpyKVCache,
                   pos0: int) -> np.ndarray:
        cfg = self.config
        q_len = h.shape[0]
        p = f"model.layers.{layer}.self_attn"
        q = h @ self.w[f"{p}.q_proj.weight"].T
        k = h @ self.w[f"{p}.k_proj.weight"].T
        v = h @ self.w[f"{p}.v_proj.weight"].T
        if f"{p}.q_proj.bias" in self.w:  # Qwen-2 family
            q = q + self.w[f"{p}.q_proj.bias"]
            k = k + self.w[f"{p}.k_proj.bias"]
            v = v + self.w[f"{p}.v_proj.bias"]
        nh, kvh, hd = cfg.num_attention_heads, cfg.num_key_value_heads, cfg.head_dim
        q = q.reshape(q_len, nh, hd).transpose(1, 0, 2)    # (nh, q, hd)
        k = k.reshape(q_len, kvh, hd).transpose(1, 0, 2)   # (kvh, q, hd)
        v = v.reshape(q_len, kvh, hd).transpose(1, 0, 2)
        q = self._rope(q, pos0)
        k = self._rope(k, pos0)
        cache.update(layer, k, v, pos0)
        T = pos0 + q_len
        K = cache.k[layer][:, :T]                          # (kvh, T, hd)
        V = cache.v[layer][:, :T]
        groups = cfg.num_kv_groups
        out = np.empty((nh, q_len, hd), dtype=np.float32)
        scale = cfg.attn_scale
        # causal mask: query at absolute position pos0+i attends to <= that
        qpos = np.arange(pos0, T)[:, None]                 # (q_len, 1)
        kpos = np.arange(T)[None, :]                       # (1, T)
        mask = kpos > qpos
        if cfg.is_sliding(layer):
            mask = mask | (kpos <= qpos - cfg.sliding_window)
        for hh in range(nh):
            kv = hh // groups
            scores = (q[hh].astype(np.float64) @ K[kv].astype(np.float64).T) * scale
            if cfg.attn_logit_softcapping:
                c = cfg.attn_logit_softcapping
                scores = c * np.tanh(scores / c)
            scores = np.where(mask, -np.inf, scores)
            probs = softmax(scores, axis=-1)
            out[hh] = (probs @ V[kv].astype(np.float64)).astype(np.float32)
        out = out.transpose(1, 0, 2).reshape(q_len, nh * hd)
        return out @ self.w[f"{p}.o_proj.weight"].T

    def _mlp(self, layer: int, h: np.ndarray) -> np.ndarray:
        if self.config.is_moe:
            return self._moe_mlp(layer, h)
        p = f"model.layers.{layer}.mlp"
        act = ACT2FN[self.config.hidden_act]
        gate = act(h @ self.w[f"{p}.gate_proj.weight"].T)
        up = h @ self.w[f"{p}.up_proj.weight"].T
        return (gate * up) @ self.w[f"{p}.down_proj.weight"].T

    def _moe_mlp(self, layer: int, h: np.ndarray) -> np.ndarray:
        """Mixtral sparse MoE (HF semantics, modeling_mixtral
        MixtralTopKRouter/MixtralExperts): softmax over ALL experts ->
        top-k probs renormalized by their own sum -> weighted sum of
        the routed experts' SwiGLU outputs."""
        cfg = self.config
        p = f"model.layers.{layer}.block_sparse_moe"
        act = ACT2FN[cfg.hidden_act]
        E, K = cfg.num_local_experts, cfg.num_experts_per_tok
        logits = h @ self.w[f"{p}.gate.weight"].T          # (M, E)
        probs = softmax(logits.astype(np.float64), axis=-1)
        top = np.argsort(-probs, axis=-1)[:, :K]           # (M, K)
        out = np.zeros_like(h)
        for m in range(h.shape[0]):
            pe = probs[m, top[m]]
            pe = pe / pe.sum()
            for j, e in enumerate(top[m]):
                q = f"{p}.experts.{e}"
                gate = act(h[m] @ self.w[f"{q}.w1.weight"].T)
                up = h[m] @ self.w[f"{q}.w3.weight"].T
                out[m] += np.float32(pe[j]) * (
                    (gate * up) @ self.w[f"{q}.w2.weight"].T)
        return out

    # -- forward ---------------------------------------------------------
    def forward(self, input_ids: np.ndarray, cache: NumpyKVCache,
                pos0: Optional[int] = None,
                collect_hidden: Optional[list] = None) -> np.ndarray:
        """input_ids: (q_len,) int array.  Returns logits (q_len, vocab).
        Writes K/V at positions [pos0, pos0+q_len) and sets cache.seq_len.
        ``collect_hidden``: optional list that receives the embedding
        output plus each layer's output (L+1 arrays — the reference's
        ``all_hidden_states``, llama3.2_model.py:682-716)."""
        cfg = self.config
        if pos0 is None:
            pos0 = cache.seq_len
        h = self.w["model.embed_tokens.weight"][input_ids].astype(np.float32)
        h = h * np.float32(cfg.embed_scale)
        if collect_hidden is not None:
            collect_hidden.append(h.copy())
        gemma = cfg.model_type == "gemma2"
        for i in range(cfg.num_hidden_layers):
            pl = f"model.layers.{i}"
            res = h
            x = self._rmsnorm(h, f"{pl}.input_layernorm.weight")
            x = self._attention(i, x, cache, pos0)
            if gemma:
                x = self._rmsnorm(x, f"{pl}.post_attention_layernorm.weight")
                h = res + x
                res = h
                x = self._rmsnorm(h, f"{pl}.pre_feedforward_layernorm.weight")
                x = self._mlp(i, x)
                x = self._rmsnorm(x, f"{pl}.post_feedforward_layernorm.weight")
                h = res + x
            else:
                h = res + x
                res = h
                x = self._rmsnorm(h, f"{pl}.post_attention_layernorm.weight")
                h = res + self._mlp(i, x)
            if collect_hidden is not None:
                collect_hidden.append(h.copy())
        h = self._rmsnorm(h, "model.norm.weight")
        logits = h @ self.w["lm_head.weight"].T
        if cfg.final_logit_softcapping:
            c = cfg.final_logit_softcapping
            logits = c * np.tanh(logits / c)
        cache.seq_len = pos0 + len(input_ids)
        return logits

    def forward_hf(self, input_ids: np.ndarray, cache: NumpyKVCache = None):
        """Reference-parity output tuple ``(loss, logits, kv_cache,
        hidden_states, attentions)`` (llama3.2_model.py:726-822).
        ``attentions`` is None, mirroring the GPU engine's surface
        (its fused online-softmax kernels never materialize the
        probability matrix; the oracle keeps the same tuple shape)."""
        cache = cache or NumpyKVCache(self.config,
                                      len(np.ravel(input_ids)) + 1)
        hidden = []
        logits = self.forward(np.ravel(input_ids), cache, 0,
                              collect_hidden=hidden)
        return None, logits, cache, hidden, None
