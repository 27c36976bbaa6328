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
    finally:
        cfg.eos_token_id = None
