"""Multi-turn sessions with KV reuse.

A ``ChatSession`` keeps one KV cache alive across turns: each
``send()`` prefills ONLY the new tokens (user text appended at the
current sequence position) instead of re-prefilling the whole
transcript — turn latency stays O(new tokens), not O(conversation).

Engine-agnostic by construction: both engines expose the same
continuation primitives (``forward(ids, cache, pos0)`` writes KV at
[pos0, pos0+n) and returns last-position logits; the GPU engine
additionally runs its device-side hipGraph ``decode`` loop from that
state).  The reference re-prefills from scratch every call
(``llama3.2_model.py:865-902`` — and when caching, feeds back
re-tokenized text); this is a beyond-parity capability.
"""

from __future__ import annotations

from typing import List, Optional

import numpy as np

from .generate import GenerateResult
from .sampling import SamplingParams, sample_token


class ChatSession:
    """Incremental multi-turn generation over one persistent KV cache.

    >>> s = ChatSession(tok, model)
    >>> a = s.send("Hello!", max_tokens=32)          # prefills prompt
    >>> b = s.send("Tell me more.", max_tokens=32)   # prefills ONLY this
    ``s.token_ids`` holds the full transcript (prompt + generated turns).
    ``reset()`` starts over without reallocating the pool.
    """

    def __init__(self, tokenizer, model,
                 params: Optional[SamplingParams] = None,
                 max_seq: Optional[int] = None):
        self.tokenizer = tokenizer
        self.model = model
        self.params = params or SamplingParams(strategy="greedy")
        n = max_seq or getattr(model, "max_seq", 4096)
        self.cache = model.make_cache(n)
        self.max_seq = int(getattr(self.cache, "max_seq", n))
        self.token_ids: List[int] = []
        eos = getattr(model.config, "eos_token_id", None)
        self.eos_set = (set() if eos is None else
                        {int(eos)} if np.isscalar(eos) else
                        {int(e) for e in eos})
        self._rng = np.random.default_rng(self.params.seed)

    def reset(self):
        self.cache = self.model.make_cache(self.max_seq)
        self.token_ids = []
        self._rng = np.random.default_rng(self.params.seed)

    @property
    def seq_len(self) -> int:
        return len(self.token_ids)

    def send(self, text: str, max_tokens: int = 128,
             stop_on_eos: bool = True) -> GenerateResult:
        """Append ``text`` to the transcript and generate a reply.
        Only the new text's tokens are prefilled (KV for everything
        before is already resident)."""
        import time

        new_ids = [int(t) for t in self.tokenizer.encode(text)]
        if not new_ids:
            raise ValueError("empty text")
        pos0 = len(self.token_ids)
        if pos0 + len(new_ids) + 1 > self.max_seq:
            raise ValueError(
                f"session full: {pos0}+{len(new_ids)} tokens in a "
                f"{self.max_seq} pool (reset() or raise max_seq)")
        max_tokens = min(max_tokens,
                         self.max_seq - pos0 - len(new_ids) - 1)

        t0 = time.perf_counter()
        logits = self.model.forward(np.asarray(new_ids, dtype=np.int64),
                                    self.cache, pos0)
        t_prefill = time.perf_counter() - t0
        self.token_ids.extend(new_ids)

        out: List[int] = []
        hit_eos = False
        t1 = time.perf_counter()
        greedy_or_minp = self.params.strategy in ("greedy", "min_p")
        if (hasattr(self.model, "decode") and greedy_or_minp
                and not self.params.logit_bias):
            # GPU engine: continue with the device-side hipGraph loop
            # (the prefill logits above are already in b_logits)
            produced = 0
            first = True
            while produced < max_tokens and not hit_eos:
                n = min(16, max_tokens - produced)
                ids = self.model.decode(
                    n, greedy=self.params.strategy == "greedy",
                    min_p=self.params.min_p, use_graph=True,
                    first_from_logits=first,
                    temperature=self.params.temperature)
                first = False
                produced += n
                take = [int(t) for t in ids]
                hit = [j for j, t in enumerate(take)
                       if stop_on_eos and t in self.eos_set]
                if hit:
                    take = take[:hit[0] + 1]
                    hit_eos = True
                out.extend(take)
            if out:
                # the device loop leaves the LAST sampled token pending
                # (its KV is written by the step that would consume it)
                # — settle it so the next send() attends over complete
                # KV: O(1) rewind + one M=1 forward
                self.model.rewind(pos0 + len(new_ids) + len(out) - 1)
                self.cache.seq_len = pos0 + len(new_ids) + len(out) - 1
                self.model.forward(
                    np.asarray([out[-1]], dtype=np.int64), self.cache,
                    self.cache.seq_len)
            self.cache.seq_len = len(self.token_ids) + len(out)
        else:
            for _ in range(max_tokens):
                nid = sample_token(np.asarray(logits[-1],
                                              dtype=np.float32),
                                   self.params, self._rng)
                out.append(nid)
                hit_eos = stop_on_eos and nid in self.eos_set
                # forward EVERY sampled token (the eos included) so the
                # cache holds KV for the whole transcript — the next
                # turn attends over all of it
                logits = self.model.forward(
                    np.asarray([nid], dtype=np.int64), self.cache,
                    self.cache.seq_len)
                if hit_eos:
                    break
        t_decode = time.perf_counter() - t1

        self.token_ids.extend(out)
        self.cache.seq_len = len(self.token_ids)
        return GenerateResult(
            text=self.tokenizer.decode(out), token_ids=out,
            prefill_time_s=t_prefill, decode_time_s=t_decode,
            finish_reason="stop" if hit_eos else "length")
