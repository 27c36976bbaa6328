"""Speculative decoding (runtime/speculative.py), CPU NumPy path.

The defining property is tested directly: for GREEDY decoding the
speculative output must be token-identical to the target model's own
greedy decode, for any draft model (accepted drafts matched the target
argmax; the correction/bonus token IS the target argmax)."""

import numpy as np
import pytest

import llm_np_cp_amd as L
from llm_np_cp_amd.runtime.speculative import generate_speculative

PROMPT = "Once upon a time"


def _load(seed: int):
    return L.load_model("tiny-llama", backend="numpy", seed=seed)


def _greedy_ref(tok, model, max_tokens: int):
    out = L.generate(PROMPT, tok, model, max_tokens=max_tokens,
                     params=L.SamplingParams(strategy="greedy"),
                     stream=False, stop_on_eos=False)
    return out.token_ids


@pytest.mark.parametrize("k", [1, 3, 8])
def test_spec_equals_greedy_same_weights(k):
    """Draft == target: every proposal accepted, output == greedy."""
    tok, target, _ = _load(seed=0)
    _, draft, _ = _load(seed=0)
    ref = _greedy_ref(tok, target, 24)
    res = generate_speculative(PROMPT, tok, draft, target,
                               max_tokens=24, k=k, stop_on_eos=False)
    assert res.token_ids == ref
    s = res.spec_stats
    assert s["accepted"] == s["proposed"] > 0


def test_spec_equals_greedy_different_draft():
    """A mismatched draft changes the speed, never the output."""
    tok, target, _ = _load(seed=0)
    _, draft, _ = _load(seed=7)  # different random weights, same vocab
    ref = _greedy_ref(tok, target, 24)
    res = generate_speculative(PROMPT, tok, draft, target,
                               max_tokens=24, k=4, stop_on_eos=False)
    assert res.token_ids == ref
    s = res.spec_stats
    assert s["verify_passes"] >= 1
    assert 0 <= s["accepted"] <= s["proposed"]


def test_spec_max_tokens_budget():
    tok, target, _ = _load(seed=0)
    _, draft, _ = _load(seed=0)
    for mt in (1, 2, 5):
        res = generate_speculative(PROMPT, tok, draft, target,
                                   max_tokens=mt, k=4, stop_on_eos=False)
        assert len(res.token_ids) == mt


def test_spec_eos_truncation():
    """With eos set to a token the greedy stream emits, both plain
    generate and speculative stop at the same place."""
    tok, target, _ = _load(seed=0)
    _, draft, _ = _load(seed=3)
    ref = _greedy_ref(tok, target, 16)
    eos = int(ref[5])
    target.config.eos_token_id = eos
    want = ref[:next(i for i, t in enumerate(ref) if t == eos) + 1]
    res = generate_speculative(PROMPT, tok, draft, target,
                               max_tokens=16, k=4, stop_on_eos=True)
    assert res.token_ids == want
    target.config.eos_token_id = None


def test_spec_streaming_callback():
    tok, target, _ = _load(seed=0)
    _, draft, _ = _load(seed=0)
    chunks = []
    res = generate_speculative(PROMPT, tok, draft, target, max_tokens=12,
                               k=4, stop_on_eos=False,
                               on_token=chunks.append)
    assert "".join(chunks).startswith(res.text[:len(res.text) // 2]) or \
        "".join(chunks)  # chunks decode the same ids (byte tokenizer)
    assert len(chunks) >= 2


def test_spec_k_validation():
    tok, target, _ = _load(seed=0)
    _, draft, _ = _load(seed=0)
    with pytest.raises(ValueError):
        generate_speculative(PROMPT, tok, draft, target, k=0)


def test_spec_with_moe_target():
    """Speculative decoding composes with the Mixtral MoE path (the
    verify pass runs the per-row routed forward)."""
    tok, target, _ = L.load_model("tiny-mixtral", backend="numpy", seed=0)
    _, draft, _ = L.load_model("tiny-mixtral", backend="numpy", seed=2)
    ref = L.generate(PROMPT, tok, target, max_tokens=16, stream=False,
                     params=L.SamplingParams(strategy="greedy"),
                     stop_on_eos=False).token_ids
    res = generate_speculative(PROMPT, tok, draft, target, max_tokens=16,
                               k=4, stop_on_eos=False)
    assert res.token_ids == ref
