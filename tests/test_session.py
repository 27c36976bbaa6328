"""ChatSession multi-turn KV reuse: incremental turns must be
token-identical to recomputing the whole transcript from scratch."""

import numpy as np
import pytest

import llm_np_cp_amd as L
from llm_np_cp_amd.models.numpy_ref import NumpyKVCache


def _greedy_cont(model, cfg, ids, k):
    """Reference: greedy continuation of `ids` with a FRESH cache."""
    cache = NumpyKVCache(cfg, 512)
    logits = model.forward(np.asarray(ids, dtype=np.int64), cache, 0)
    out = []
    for _ in range(k):
        nid = int(np.argmax(logits[-1]))
        out.append(nid)
        logits = model.forward(np.asarray([nid], dtype=np.int64), cache,
                               cache.seq_len)
    return out


def test_session_turns_match_from_scratch():
    tok, model, cfg = L.load_model("tiny-llama", backend="numpy", seed=0)
    s = L.ChatSession(tok, model,
                      params=L.SamplingParams(strategy="greedy"),
                      max_seq=256)
    a_ids = tok.encode("Hello there.")
    r1 = s.send("Hello there.", max_tokens=10, stop_on_eos=False)
    ref1 = _greedy_cont(model, cfg, a_ids, 10)
    assert r1.token_ids == ref1

    b_ids = tok.encode(" And then?")
    r2 = s.send(" And then?", max_tokens=10, stop_on_eos=False)
    ref2 = _greedy_cont(model, cfg, a_ids + ref1 + b_ids, 10)
    assert r2.token_ids == ref2
    assert s.token_ids == a_ids + ref1 + b_ids + ref2
    assert s.seq_len == len(s.token_ids)


def test_session_reset_and_pool_guard():
    tok, model, cfg = L.load_model("tiny-llama", backend="numpy", seed=1)
    s = L.ChatSession(tok, model,
                      params=L.SamplingParams(strategy="greedy"),
                      max_seq=32)
    r = s.send("abcdef", max_tokens=100, stop_on_eos=False)
    assert len(r.token_ids) <= 32 - 6 - 1   # clamped to pool room
    with pytest.raises(ValueError, match="session full"):
        s.send("x" * 40)
    first = list(s.token_ids)
    s.reset()
    assert s.seq_len == 0
    r2 = s.send("abcdef", max_tokens=100, stop_on_eos=False)
    assert s.token_ids == first  # deterministic replay after reset


def test_session_eos_stops_turn():
    tok, model, cfg = L.load_model("tiny-llama", backend="numpy", seed=0)
    probe = L.ChatSession(tok, model,
                          params=L.SamplingParams(strategy="greedy"),
                          max_seq=128)
    r = probe.send("Once upon", max_tokens=8, stop_on_eos=False)
    eos = int(r.token_ids[3])
    first = r.token_ids.index(eos)   # repetitive greedy text: the
    cfg.eos_token_id = eos           # FIRST occurrence may precede [3]
    try:
        s = L.ChatSession(tok, model,
                          params=L.SamplingParams(strategy="greedy"),
                          max_seq=128)
        r2 = s.send("Once upon", max_tokens=8, stop_on_eos=True)
        assert r2.token_ids == r.token_ids[:first + 1]
        assert r2.finish_reason == "stop"
        # the eos token's KV must be resident so the NEXT turn attends
        # over the complete transcript
        kc = s.cache.k[:, :, :s.seq_len]
        norms = np.abs(kc).sum(axis=(0, 1, 3))
        assert (norms > 0).all(), np.where(norms == 0)
        r3 = s.send(" next turn", max_tokens=4, stop_on_eos=False)
        assert len(r3.token_ids) == 4
    finally:
        cfg.eos_token_id = None


class DeviceLoopAdapter:
    """Reproduces GPUModel's decode-loop CONTRACT over the NumPy oracle
    so ChatSession's GPU branch is CPU-testable: forward() leaves
    last-position logits; decode(n, first_from_logits) samples token 0
    from them, then each further step PROCESSES the pending token
    (writing its KV / advancing the cache) and samples the next — the
    final sampled token is left pending with NO KV written, exactly
    like the hipGraph loop.  rewind() is an O(1) length set."""

    def __init__(self, cfg, ref, max_seq=256):
        self.config = cfg
        self.ref = ref
        self.max_seq = max_seq
        self._logits = None
        self._pending = None

    def make_cache(self, n):
        self._cache = NumpyKVCache(self.config, n)
        return self._cache

    def forward(self, ids, cache, pos0):
        self._logits = self.ref.forward(np.asarray(ids), cache, pos0)
        return self._logits

    def rewind(self, n):
        self._cache.seq_len = n

    def decode(self, n, greedy=True, min_p=0.1, use_graph=True,
               first_from_logits=True, temperature=1.0):
        assert greedy, "adapter models the greedy device loop"
        out = []
        steps = n
        if first_from_logits:
            self._pending = int(np.argmax(self._logits[-1]))
            out.append(self._pending)
            steps -= 1
        for _ in range(steps):
            self._logits = self.ref.forward(
                np.asarray([self._pending]), self._cache,
                self._cache.seq_len)
            self._pending = int(np.argmax(self._logits[-1]))
            out.append(self._pending)
        return np.asarray(out, dtype=np.int32)


def test_session_device_loop_branch_matches_plain_oracle():
    """The GPU branch of ChatSession (device decode loop + pending-token
    settle) must be turn-identical to the plain oracle branch — in
    particular turn 2 attends over the KV of turn 1's LAST token, which
    the device loop leaves unwritten until settled."""
    tok, model, cfg = L.load_model("tiny-llama", backend="numpy", seed=0)
    from llm_np_cp_amd.models.numpy_ref import NumpyModel
    ref2 = NumpyModel(cfg, dict(model.w))
    dev = DeviceLoopAdapter(cfg, ref2, max_seq=256)

    p = L.SamplingParams(strategy="greedy")
    s_plain = L.ChatSession(tok, model, params=p, max_seq=256)
    s_dev = L.ChatSession(tok, dev, params=p, max_seq=256)
    for text, k in (("Hello there.", 9), (" And then?", 9),
                    (" The end?", 6)):
        a = s_plain.send(text, max_tokens=k, stop_on_eos=False)
        b = s_dev.send(text, max_tokens=k, stop_on_eos=False)
        assert a.token_ids == b.token_ids, text
        # the settle must leave KV WRITTEN for every resident position
        # (without it the last token of each turn has a zeroed KV row —
        # argmax can mask that on tiny models, the cache cannot)
        kc = dev._cache.k[:, :, :s_dev.seq_len]
        norms = np.abs(kc).sum(axis=(0, 1, 3))
        assert (norms > 0).all(), np.where(norms == 0)
    assert s_plain.token_ids == s_dev.token_ids


def test_session_min_p_seeded_reset_determinism():
    """Stochastic sessions: reset() restores the RNG stream, so the
    same seed replays the same multi-turn sampling."""
    tok, model, cfg = L.load_model("tiny-llama", backend="numpy", seed=2)
    p = L.SamplingParams(strategy="min_p", min_p=0.05, seed=77)
    s = L.ChatSession(tok, model, params=p, max_seq=128)
    a1 = s.send("hello", max_tokens=6, stop_on_eos=False).token_ids
    a2 = s.send(" more", max_tokens=6, stop_on_eos=False).token_ids
    s.reset()
    b1 = s.send("hello", max_tokens=6, stop_on_eos=False).token_ids
    b2 = s.send(" more", max_tokens=6, stop_on_eos=False).token_ids
    assert (a1, a2) == (b1, b2)
