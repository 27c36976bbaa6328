"""Token samplers.

Reference parity: the live sampler in the reference is min-p with
p_base=0.1 over an (unstabilized) softmax followed by ``torch.multinomial``
(``/root/reference/llama3.2_model.py:1000-1013``).  Here: stabilized
softmax, explicit generator for reproducibility, plus greedy /
temperature / top-k / top-p which the reference only had commented out
(``llama3.2_model.py:895-896``).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import numpy as np


@dataclass
class SamplingParams:
    strategy: str = "min_p"   # "min_p" | "greedy" | "top_k" | "top_p" | "temperature"
    min_p: float = 0.1
    temperature: float = 1.0
    top_k: int = 50
    top_p: float = 0.9
    seed: Optional[int] = None
    # OpenAI-style per-token logit offsets {token_id: bias}; applied to
    # the raw logits before temperature/filtering (+-100 ~ ban/force)
    logit_bias: Optional[dict] = None


def _apply_bias(logits: np.ndarray, params: "SamplingParams") -> np.ndarray:
    if not params.logit_bias:
        return logits
    logits = np.array(logits, dtype=np.float32, copy=True)
    for tid, b in params.logit_bias.items():
        t = int(tid)
        if 0 <= t < len(logits):
            logits[t] += float(b)
    return logits


def _softmax(x: np.ndarray) -> np.ndarray:
    x = x.astype(np.float64)
    m = x.max()
    e = np.exp(x - m)
    return e / e.sum()


def filter_probs(logits: np.ndarray, params: SamplingParams) -> np.ndarray:
    """The sampling DISTRIBUTION the strategy defines: temperature-scaled
    softmax with the strategy's support filter applied, renormalized.
    (fp64, sums to 1.)  ``sample_token`` draws from this; speculative
    sampling (runtime/speculative.py) needs the vector itself to compute
    accept ratios p(x)/q(x) and the residual distribution.  logit_bias
    applies first, so every consumer (host sampler, speculative
    accept/reject) sees the same biased distribution."""
    logits = _apply_bias(logits, params)
    if params.strategy == "greedy":
        p = np.zeros(len(logits), dtype=np.float64)
        p[int(np.argmax(logits))] = 1.0
        return p
    if params.temperature != 1.0:
        logits = logits / max(params.temperature, 1e-6)
    probs = _softmax(logits)
    if params.strategy == "min_p":
        # keep tokens with p >= min_p * p_max (reference: 0.1 * p_max,
        # llama3.2_model.py:1004-1008)
        keep = probs >= params.min_p * probs.max()
        probs = np.where(keep, probs, 0.0)
    elif params.strategy == "top_k":
        k = min(params.top_k, len(probs))
        thresh = np.partition(probs, -k)[-k]
        probs = np.where(probs >= thresh, probs, 0.0)
    elif params.strategy == "top_p":
        order = np.argsort(-probs)
        csum = np.cumsum(probs[order])
        cutoff = np.searchsorted(csum, params.top_p) + 1
        mask = np.zeros_like(probs, dtype=bool)
        mask[order[:cutoff]] = True
        probs = np.where(mask, probs, 0.0)
    elif params.strategy != "temperature":
        raise ValueError(f"unknown sampling strategy {params.strategy!r}")
    return probs / probs.sum()


def sample_token(logits: np.ndarray, params: SamplingParams,
                 rng: Optional[np.random.Generator] = None) -> int:
    """logits: (vocab,) fp32. Returns a token id (int)."""
    if params.strategy == "greedy":
        return int(np.argmax(_apply_bias(logits, params)))
    if rng is None:
        rng = np.random.default_rng(params.seed)
    probs = filter_probs(logits, params)
    return int(rng.choice(len(probs), p=probs))
