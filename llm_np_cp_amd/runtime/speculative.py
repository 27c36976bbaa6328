"""Greedy speculative decoding: a small DRAFT model proposes k tokens,
the TARGET verifies them in ONE prefill-shaped pass and accepts the
longest matching prefix, emitting m accepted drafts plus one target
token per verify pass (1..k+1 tokens per target forward).

The output is IDENTICAL to the target's own greedy decode under the
verify pass's numerics: every emitted token is an argmax of target
logits (accepted drafts matched it; the correction/bonus token IS it).

MI355X-native fit: both engines keep preallocated KV pools whose live
length is a device scalar, so rejecting draft tokens is an O(1) length
rewind (``GPUModel.rewind`` / ``NumpyKVCache.seq_len``) — never a
KV copy, unlike cache-concat designs (the reference's concat cache,
``/root/reference/llama3.2_model.py:303-332``, would have to slice and
re-materialize every rejection).  The verify pass reuses the fp8/bf16
MFMA prefill GEMMs (``forward_positions``), so verifying k tokens
streams the weights ONCE instead of k times — the same
bandwidth argument that makes batched decode pay.

The reference has no speculative decoding (SURVEY §2 — single-model
greedy/min-p only); this is a beyond-parity capability (ROADMAP §5).
"""

from __future__ import annotations

import time
from typing import List, Optional

import numpy as np

from .generate import GenerateResult


def _cap(model, cache) -> int:
    """Usable KV length for (model, cache)."""
    n = getattr(cache, "max_seq", None)
    return int(n if n is not None else model.max_seq)


def _all_logits(model, cache, ids: List[int], pos0: int) -> np.ndarray:
    """(len(ids), V) fp32 logits, KV written at [pos0, pos0+M)."""
    if hasattr(model, "forward_positions"):
        out = model.forward_positions(np.asarray(ids, dtype=np.int32), pos0)
    else:
        out = model.forward(np.asarray(ids, dtype=np.int64), cache, pos0)
    cache.seq_len = pos0 + len(ids)
    return out


def _rewind(model, cache, n: int) -> None:
    if hasattr(model, "rewind"):
        model.rewind(n)
    cache.seq_len = n


def generate_speculative(prompt: str, tokenizer, draft, target,
                         max_tokens: int = 200, k: int = 4,
                         stop_on_eos: bool = True,
                         on_token=None) -> GenerateResult:
    """Greedy decode of ``target`` accelerated by ``draft`` proposals.

    ``draft`` and ``target`` must share the tokenizer/vocab (standard
    speculative-decoding requirement).  Returns a GenerateResult; the
    extra attribute ``spec_stats`` holds ``{proposed, accepted,
    verify_passes}`` (acceptance rate = accepted / proposed).
    """
    if k < 1:
        raise ValueError("k must be >= 1")
    prompt_ids = [int(t) for t in tokenizer.encode(prompt)]
    P0 = len(prompt_ids)
    eos = getattr(target.config, "eos_token_id", None)
    eos_set = (set() if eos is None or not stop_on_eos else
               {int(eos)} if np.isscalar(eos) else {int(e) for e in eos})

    t_cache = target.make_cache(min(getattr(target, "max_seq",
                                            P0 + max_tokens + k + 2),
                                    P0 + max_tokens + k + 2))
    d_cache = draft.make_cache(min(getattr(draft, "max_seq",
                                           P0 + max_tokens + k + 2),
                                   P0 + max_tokens + k + 2))
    t_max, d_max = _cap(target, t_cache), _cap(draft, d_cache)
    if P0 + 1 > min(t_max, d_max):
        raise ValueError(f"prompt {P0} fills the KV pool "
                         f"({min(t_max, d_max)})")

    t0 = time.perf_counter()
    vl = _all_logits(target, t_cache, prompt_ids, 0)
    pending = int(np.argmax(vl[-1]))      # first output token
    _all_logits(draft, d_cache, prompt_ids, 0)
    t_prefill = time.perf_counter() - t0

    out: List[int] = [pending]
    if on_token is not None:
        on_token(tokenizer.decode([pending]))
    all_tokens = prompt_ids + out         # context + pending
    n = P0                                # verified context length
    d_len = P0                            # draft-processed length
    stats = {"proposed": 0, "accepted": 0, "verify_passes": 0}

    t1 = time.perf_counter()
    while len(out) < max_tokens and pending not in eos_set:
        kk = min(k, t_max - n - 1, d_max - n)
        if kk < 1:
            if n + 1 < t_max:             # no draft room: plain step
                vl = _all_logits(target, t_cache, [pending], n)
                pending = int(np.argmax(vl[-1]))
                out.append(pending)
                all_tokens.append(pending)
                n += 1
                if on_token is not None:
                    on_token(tokenizer.decode([pending]))
                continue
            break                         # target pool exhausted too
        # draft proposes kk tokens (first call also catches the draft
        # cache up on tokens it skipped when a verify pass over-ran it)
        dl = _all_logits(draft, d_cache, all_tokens[d_len:], d_len)
        d_len = len(all_tokens)
        drafts = [int(np.argmax(dl[-1]))]
        for _ in range(kk - 1):
            dl = _all_logits(draft, d_cache, [drafts[-1]], d_len)
            d_len += 1
            drafts.append(int(np.argmax(dl[-1])))
        # target verifies pending + kk drafts in one pass
        vl = _all_logits(target, t_cache, [pending] + drafts, n)
        m = 0
        while m < kk and int(np.argmax(vl[m])) == drafts[m]:
            m += 1
        pending = int(np.argmax(vl[m]))   # correction (m<kk) or bonus
        emitted = drafts[:m] + [pending]
        out.extend(emitted)
        all_tokens.extend(emitted)
        n += m + 1
        _rewind(target, t_cache, n)
        _rewind(draft, d_cache, min(d_len, n))
        d_len = min(d_len, n)
        stats["proposed"] += kk
        stats["accepted"] += m
        stats["verify_passes"] += 1
        if on_token is not None:
            on_token(tokenizer.decode(emitted))
        if eos_set and any(t in eos_set for t in emitted):
            break
    t_decode = time.perf_counter() - t1

    if eos_set:
        hit = next((i for i, t in enumerate(out) if t in eos_set), None)
        if hit is not None:
            out = out[:hit + 1]
    out = out[:max_tokens]
    res = GenerateResult(text=tokenizer.decode(out), token_ids=out,
                         prefill_time_s=t_prefill, decode_time_s=t_decode)
    res.spec_stats = stats
    return res
