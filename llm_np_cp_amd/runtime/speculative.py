"""Speculative decoding: a small DRAFT model proposes k tokens, the
TARGET verifies them in ONE prefill-shaped pass, emitting the accepted
drafts plus one target token per verify pass (1..k+1 tokens per target
forward).

Two modes through ONE accept/reject machine (``spec_accept``):

- **Stochastic** (``params`` with a sampling strategy): the standard
  speculative-sampling rule — draft token x ~ q is accepted with
  probability min(1, p(x)/q(x)); on the first rejection the replacement
  is drawn from the residual distribution norm(max(p - q, 0)); if all k
  survive, a bonus token is drawn from the target's next-position p.
  Every emitted token is distributed EXACTLY as the target's own
  (filtered) sampling distribution p — the draft changes speed, never
  statistics (chi-square-verified in tests/test_speculative.py).
  p and q are both warped by the SAME SamplingParams
  (temperature + min-p/top-k/top-p filter, ``sampling.filter_probs``).

- **Greedy** (default / ``strategy="greedy"``): the same machine with
  one-hot p and q degenerates to longest-matching-prefix acceptance,
  so the output is IDENTICAL to the target's own greedy decode: every
  emitted token is an argmax of target logits (accepted drafts matched
  it; the correction/bonus token IS it).

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
from .sampling import SamplingParams, filter_probs


def spec_accept(target_probs: np.ndarray, draft_probs: np.ndarray,
                drafts: List[int], rng: np.random.Generator):
    """One verify pass of the speculative accept/reject rule.

    ``target_probs``: (k+1, V) filtered target distributions — row i is
    p_i, the target's distribution for draft slot i; the last row is
    the bonus-position distribution.  ``draft_probs``: (k, V) filtered
    draft distributions q_i.  ``drafts``: the k proposed ids, with
    drafts[i] drawn from q_i.

    Returns ``(m, next_token)``: the first m drafts are accepted and
    ``next_token`` is drawn from the residual norm(max(p_m - q_m, 0))
    (m < k, rejection) or from the bonus row (m == k).  Marginally each
    emitted token ~ p exactly (Leviathan/Chen speculative sampling);
    with one-hot p/q (greedy) this reduces to longest-matching-prefix
    + argmax correction, bit-equal to target-only greedy decode.
    """
    k = len(drafts)
    for i in range(k):
        x = int(drafts[i])
        p, q = target_probs[i], draft_probs[i]
        qx = float(q[x])
        px = float(p[x])
        # accept with prob min(1, p(x)/q(x)); qx > 0 because x ~ q
        if qx > 0.0 and rng.random() * qx < px:
            continue
        resid = np.maximum(p - q, 0.0)
        s = float(resid.sum())
        if s <= 0.0:          # p <= q everywhere at fp precision ⇒ p == q
            resid, s = p, float(p.sum())
        return i, int(rng.choice(len(resid), p=resid / s))
    p = target_probs[k]
    return k, int(rng.choice(len(p), p=p / float(p.sum())))


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
                         params: Optional[SamplingParams] = None,
                         on_token=None) -> GenerateResult:
    """Decode ``target`` accelerated by ``draft`` proposals.

    ``params=None`` (or strategy "greedy") reproduces target-only
    greedy decode token-identically; any other SamplingParams runs
    stochastic speculative sampling whose per-token distribution equals
    target-only sampling under the same params (see module docstring).

    ``draft`` and ``target`` must share the tokenizer/vocab (standard
    speculative-decoding requirement).  Returns a GenerateResult; the
    extra attribute ``spec_stats`` holds ``{proposed, accepted,
    verify_passes}`` (acceptance rate = accepted / proposed).
    """
    if k < 1:
        raise ValueError("k must be >= 1")
    if params is None:
        params = SamplingParams(strategy="greedy")
    rng = np.random.default_rng(params.seed)

    def _pick(logits_row: np.ndarray) -> int:
        # route greedy through filter_probs too so logit_bias applies
        # identically here and in the accept/reject distributions
        p = filter_probs(logits_row, params)
        if params.strategy == "greedy":
            return int(np.argmax(p))
        return int(rng.choice(len(p), p=p))
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
    pending = _pick(vl[-1])               # first output token
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
                pending = _pick(vl[-1])
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
        q_rows = [filter_probs(dl[-1], params)]
        drafts = [int(rng.choice(len(q_rows[0]), p=q_rows[0]))]
        for _ in range(kk - 1):
            dl = _all_logits(draft, d_cache, [drafts[-1]], d_len)
            d_len += 1
            q_rows.append(filter_probs(dl[-1], params))
            drafts.append(int(rng.choice(len(q_rows[-1]), p=q_rows[-1])))
        # target verifies pending + kk drafts in one pass; rows 0..kk of
        # vl are the target distributions at each draft slot + bonus
        vl = _all_logits(target, t_cache, [pending] + drafts, n)
        p_rows = np.stack([filter_probs(vl[i], params)
                           for i in range(kk + 1)])
        m, pending = spec_accept(p_rows, np.stack(q_rows), drafts, rng)
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
