"""GPU engine integration tests (1 GPU): parity vs the NumPy oracle,
cache-path equivalence, and hipGraph decode correctness."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module", autouse=True)
def _build():
    from csrc.build import ensure_built
    ensure_built()


def softmax_np(x, axis=-1):
    x = np.asarray(x, dtype=np.float64)
    m = x.max(axis=axis, keepdims=True)
    e = np.exp(x - m)
    return e / e.sum(axis=axis, keepdims=True)


def kl_bits(p_logits, q_logits, axis=-1):
    """KL(p || q) in bits per position — catches kernel regressions that
    argmax-only checks miss (VERDICT r1 'tighten verification')."""
    p = softmax_np(p_logits, axis)
    q = softmax_np(q_logits, axis)
    return (p * (np.log2(p + 1e-30) - np.log2(q + 1e-30))).sum(axis)


def topk_overlap(a_logits, b_logits, k=8):
    a = set(np.argsort(a_logits)[-k:].tolist())
    b = set(np.argsort(b_logits)[-k:].tolist())
    return len(a & b) / k


def make_pair(preset, seed=0, max_seq=256):
    import llm_np_cp_amd as L
    from llm_np_cp_amd.io.loader import random_weights
    from llm_np_cp_amd.models.engine import GPUModel
    from llm_np_cp_amd.models.numpy_ref import NumpyModel, NumpyKVCache

    cfg = L.preset_config(preset)
    w = random_weights(cfg, seed=seed)
    gpu = GPUModel(cfg, w, max_seq=max_seq)
    ref = NumpyModel(cfg, dict(w))
    ref.make_cache = lambda n: NumpyKVCache(cfg, n)
    return cfg, gpu, ref


@pytest.mark.parametrize("preset", ["tiny-llama", "tiny-gemma2",
                                    "tiny-qwen2", "tiny-mistral"])
def test_prefill_logits_match_oracle(preset):
    cfg, gpu, ref = make_pair(preset)
    rng = np.random.default_rng(0)
    ids = rng.integers(0, cfg.vocab_size, size=13)

    from llm_np_cp_amd.models.numpy_ref import NumpyKVCache
    ref_logits = ref.forward(ids, NumpyKVCache(cfg, 64), 0)

    cache = gpu.make_cache(64)
    got = gpu.forward(ids, cache, 0)[0]

    # bf16 forward vs fp32 oracle: compare softmax-relevant structure
    np.testing.assert_allclose(got, ref_logits[-1], rtol=0.15, atol=0.15)
    assert np.argmax(got) == np.argmax(ref_logits[-1])
    # distributional bounds (sharper than argmax): next-token KL and
    # top-k set agreement vs the fp32 oracle
    assert kl_bits(ref_logits[-1], got) < 0.02
    assert topk_overlap(got, ref_logits[-1], k=8) >= 0.875


@pytest.mark.parametrize("preset", ["tiny-llama", "tiny-gemma2",
                                    "tiny-qwen2", "tiny-mistral"])
def test_greedy_decode_matches_oracle(preset):
    """Token-id equality over a greedy rollout (SURVEY §4 integration)."""
    import llm_np_cp_amd as L

    cfg, gpu, ref = make_pair(preset, seed=1)
    tok = L.ByteTokenizer()
    p = L.SamplingParams(strategy="greedy")
    r_ref = L.generate("Once upon", tok, ref, max_tokens=12, stream=False,
                       params=p, stop_on_eos=False)
    r_gpu = L.generate("Once upon", tok, gpu, max_tokens=12, stream=False,
                       params=p, stop_on_eos=False)
    assert r_ref.token_ids == r_gpu.token_ids


def test_graph_decode_matches_eager():
    cfg, gpu, ref = make_pair("tiny-llama", seed=2)
    prompt = np.arange(1, 9)

    gpu.prefill(prompt)
    ids_eager = gpu.decode(10, greedy=True, use_graph=False)

    gpu.prefill(prompt)
    ids_graph = gpu.decode(10, greedy=True, use_graph=True)
    np.testing.assert_array_equal(ids_eager, ids_graph)


def test_fast_decode_matches_oracle_greedy():
    import llm_np_cp_amd as L

    cfg, gpu, ref = make_pair("tiny-llama", seed=3)
    rng = np.random.default_rng(5)
    prompt = rng.integers(0, cfg.vocab_size, size=7)

    # oracle rollout
    from llm_np_cp_amd.models.numpy_ref import NumpyKVCache
    cache = NumpyKVCache(cfg, 128)
    logits = ref.forward(prompt, cache, 0)
    want = []
    for _ in range(8):
        t = int(np.argmax(logits[-1]))
        want.append(t)
        logits = ref.forward(np.asarray([t]), cache, cache.seq_len)

    gpu.prefill(prompt)
    got = gpu.decode(8, greedy=True, use_graph=True)
    assert got.tolist() == want


def test_long_decode_past_prefill_chunk():
    """Prompt longer than one prefill chunk exercises chunked prefill."""
    cfg, gpu, ref = make_pair("tiny-llama", seed=4, max_seq=1200)
    rng = np.random.default_rng(6)
    ids = rng.integers(0, cfg.vocab_size, size=700)  # > PC=512

    from llm_np_cp_amd.models.numpy_ref import NumpyKVCache
    ref_logits = ref.forward(ids, NumpyKVCache(cfg, 1200), 0)
    got = gpu.forward(ids, gpu.make_cache(1200), 0)[0]
    assert np.argmax(got) == np.argmax(ref_logits[-1])


def test_min_p_decode_stays_in_support():
    """min-p sampled tokens must be in the oracle's min-p support set."""
    cfg, gpu, ref = make_pair("tiny-llama", seed=7)
    prompt = np.arange(1, 6)
    from llm_np_cp_amd.models.numpy_ref import NumpyKVCache

    gpu.prefill(prompt)
    ids = gpu.decode(1, greedy=False, min_p=0.1, use_graph=False)

    ref_logits = ref.forward(prompt, NumpyKVCache(cfg, 64), 0)[-1]
    probs = np.exp(ref_logits - ref_logits.max())
    probs /= probs.sum()
    support = set(np.where(probs >= 0.05 * probs.max())[0].tolist())
    assert int(ids[-1]) in support


def test_fp8_engine_decode_close_to_bf16():
    """fp8-weight decode must track the bf16 engine closely on logits and
    produce a plausible greedy rollout (per-channel e4m3 quantization)."""
    import llm_np_cp_amd as L
    from llm_np_cp_amd.io.loader import random_weights
    from llm_np_cp_amd.models.engine import GPUModel

    cfg = L.preset_config("tiny-llama")
    w = random_weights(cfg, seed=9)
    bf = GPUModel(cfg, w, max_seq=128)
    f8 = GPUModel(cfg, dict(w), max_seq=128, dtype="fp8")
    prompt = np.arange(1, 9)

    _, logits_bf = bf.prefill(prompt)
    _, logits_f8 = f8.prefill(prompt)
    # single-copy fp8: prefill now runs fp8 MFMA (both operands
    # quantized per-row) -> differences bounded by e4m3 grid noise
    np.testing.assert_allclose(logits_f8, logits_bf, rtol=5e-2, atol=8e-2)
    assert np.argmax(logits_f8) == np.argmax(logits_bf)

    ids_bf = bf.decode(6, greedy=True, use_graph=False)
    f8.prefill(prompt)
    ids_f8 = f8.decode(6, greedy=True, use_graph=False)
    # fp8 rollout may diverge late; first tokens should agree on tiny cfg
    assert ids_f8[0] == ids_bf[0]


def test_fp8_graph_decode_matches_eager():
    import llm_np_cp_amd as L
    from llm_np_cp_amd.io.loader import random_weights
    from llm_np_cp_amd.models.engine import GPUModel

    cfg = L.preset_config("tiny-llama")
    w = random_weights(cfg, seed=10)
    m = GPUModel(cfg, w, max_seq=128, dtype="fp8")
    prompt = np.arange(1, 9)
    m.prefill(prompt)
    a = m.decode(8, greedy=True, use_graph=False)
    m.prefill(prompt)
    b = m.decode(8, greedy=True, use_graph=True)
    np.testing.assert_array_equal(a, b)


def test_tp_branch_matches_plain_on_one_gpu():
    """The TP code path (partial GEMV -> all-reduce -> add, vocab shard +
    gather) must produce identical greedy ids at world=1 where the
    collectives are no-ops — catches TP-branch buffer/shape bugs before
    a real multi-GPU run."""
    import llm_np_cp_amd as L
    from llm_np_cp_amd.io.loader import random_weights
    from llm_np_cp_amd.models.engine import GPUModel

    cfg = L.preset_config("tiny-llama")
    w = random_weights(cfg, seed=21)
    plain = GPUModel(cfg, dict(w), max_seq=128)
    tp = GPUModel(cfg, dict(w), max_seq=128, force_tp_path=True)
    prompt = np.arange(1, 10)

    plain.prefill(prompt)
    a = plain.decode(8, greedy=True, use_graph=False)
    tp.prefill(prompt)
    b = tp.decode(8, greedy=True, use_graph=False)
    np.testing.assert_array_equal(a, b)

    # and through the graph path
    tp.prefill(prompt)
    c = tp.decode(8, greedy=True, use_graph=True)
    np.testing.assert_array_equal(a, c)


def test_tp_branch_gemma():
    import llm_np_cp_amd as L
    from llm_np_cp_amd.io.loader import random_weights
    from llm_np_cp_amd.models.engine import GPUModel

    cfg = L.preset_config("tiny-gemma2")
    w = random_weights(cfg, seed=22)
    plain = GPUModel(cfg, dict(w), max_seq=128)
    tp = GPUModel(cfg, dict(w), max_seq=128, force_tp_path=True)
    prompt = np.arange(1, 8)
    plain.prefill(prompt)
    a = plain.decode(6, greedy=True, use_graph=False)
    tp.prefill(prompt)
    b = tp.decode(6, greedy=True, use_graph=False)
    np.testing.assert_array_equal(a, b)


def test_forward_full_matches_oracle_all_positions():
    """All-positions logits parity (the reference's HF-tuple shape)."""
    cfg, gpu, ref = make_pair("tiny-llama", seed=30)
    rng = np.random.default_rng(31)
    ids = rng.integers(0, cfg.vocab_size, size=11)

    from llm_np_cp_amd.models.numpy_ref import NumpyKVCache
    want = ref.forward(ids, NumpyKVCache(cfg, 64), 0)
    got = gpu.forward_full(ids)
    assert got.shape == want.shape
    # bf16 gemm logits vs fp32 oracle: argmax agreement per position
    assert (got.argmax(1) == want.argmax(1)).mean() > 0.9
    # per-position KL bound (mean over positions) — a ±1-ulp-class
    # kernel regression shows up here long before argmax flips
    kl = kl_bits(want, got, axis=1)
    assert kl.mean() < 0.02 and kl.max() < 0.08, (kl.mean(), kl.max())


def test_forward_full_gemma_softcap():
    cfg, gpu, ref = make_pair("tiny-gemma2", seed=32)
    rng = np.random.default_rng(33)
    ids = rng.integers(0, cfg.vocab_size, size=9)
    from llm_np_cp_amd.models.numpy_ref import NumpyKVCache
    want = ref.forward(ids, NumpyKVCache(cfg, 64), 0)
    got = gpu.forward_full(ids)
    np.testing.assert_allclose(got[-1], want[-1], rtol=0.2, atol=0.2)
    assert np.abs(got).max() <= cfg.final_logit_softcapping + 1e-3


def test_hd64_engine_matches_oracle():
    """hd=64 tiny config: the engine's MFMA flash-prefill + decode path
    (the one real presets use) vs the NumPy oracle."""
    import llm_np_cp_amd as L

    cfg, gpu, ref = make_pair("tiny-llama-hd64", seed=40)
    rng = np.random.default_rng(41)
    ids = rng.integers(0, cfg.vocab_size, size=23)

    from llm_np_cp_amd.models.numpy_ref import NumpyKVCache
    ref_logits = ref.forward(ids, NumpyKVCache(cfg, 64), 0)
    got = gpu.forward(ids, gpu.make_cache(64), 0)[0]
    assert np.argmax(got) == np.argmax(ref_logits[-1])

    # greedy rollout equality
    tok = L.ByteTokenizer()
    p = L.SamplingParams(strategy="greedy")
    a = L.generate("Once upon", tok, ref, max_tokens=10, stream=False,
                   params=p, stop_on_eos=False)
    b = L.generate("Once upon", tok, gpu, max_tokens=10, stream=False,
                   params=p, stop_on_eos=False)
    assert a.token_ids == b.token_ids


def test_decode_overflow_guard():
    import llm_np_cp_amd as L
    import pytest as _pytest

    cfg, gpu, ref = make_pair("tiny-llama", seed=50, max_seq=64)
    gpu.prefill(np.arange(1, 33))
    with _pytest.raises(ValueError, match="overflow"):
        gpu.decode(100, greedy=True, use_graph=False)


def test_gemma_hd64_mfma_prefill_matches_oracle():
    """Gemma semantics (sliding window + softcaps + sandwich norms)
    through the MFMA prefill path (hd=64) vs the NumPy oracle."""
    import llm_np_cp_amd as L

    cfg, gpu, ref = make_pair("tiny-gemma2-hd64", seed=60)
    rng = np.random.default_rng(61)
    ids = rng.integers(0, cfg.vocab_size, size=21)  # > sliding_window=8

    from llm_np_cp_amd.models.numpy_ref import NumpyKVCache
    ref_logits = ref.forward(ids, NumpyKVCache(cfg, 64), 0)
    got = gpu.forward(ids, gpu.make_cache(64), 0)[0]
    assert np.argmax(got) == np.argmax(ref_logits[-1])

    tok = L.ByteTokenizer()
    p = L.SamplingParams(strategy="greedy")
    a = L.generate("Hello there", tok, ref, max_tokens=8, stream=False,
                   params=p, stop_on_eos=False)
    b = L.generate("Hello there", tok, gpu, max_tokens=8, stream=False,
                   params=p, stop_on_eos=False)
    assert a.token_ids == b.token_ids


def test_fast_generate_eos_stop():
    """generate() fast path stops at EOS."""
    import llm_np_cp_amd as L
    from llm_np_cp_amd.io.loader import random_weights
    from llm_np_cp_amd.models.engine import GPUModel

    cfg = L.preset_config("tiny-llama")
    w = random_weights(cfg, seed=70)
    m = GPUModel(cfg, w, max_seq=128)
    tok = L.ByteTokenizer()
    p = L.SamplingParams(strategy="greedy")
    # learn what greedy produces, then declare token #3 the EOS
    r = L.generate("abc", tok, m, max_tokens=8, stream=False, params=p,
                   stop_on_eos=False)
    assert len(r.token_ids) == 8
    eos = r.token_ids[2]
    cfg.eos_token_id = eos
    first = r.token_ids.index(eos)  # greedy can repeat tokens
    r2 = L.generate("abc", tok, m, max_tokens=8, stream=False, params=p,
                    stop_on_eos=True)
    assert r2.token_ids == r.token_ids[:first + 1]


def test_gpu_forward_hf_hidden_states_match_oracle():
    """GPU forward_hf: reference 5-tuple surface with per-layer hidden
    states matching the NumPy oracle."""
    cfg, gpu, ref = make_pair("tiny-llama", seed=40)
    ids = np.random.default_rng(41).integers(0, cfg.vocab_size, size=9)
    loss, logits, cache, hidden, attn = gpu.forward_hf(ids)
    assert loss is None and attn is None
    assert cache.seq_len == len(ids)
    assert len(hidden) == cfg.num_hidden_layers + 1
    r_loss, r_logits, r_cache, r_hidden, r_attn = ref.forward_hf(ids)
    for i, (a, b) in enumerate(zip(hidden, r_hidden)):
        assert a.shape == b.shape
        scale = np.abs(b).max() + 1e-6
        assert np.abs(a - b).max() < 0.05 * scale, f"layer {i}"
    kl = kl_bits(r_logits, logits, axis=1)
    assert kl.mean() < 0.02, kl.mean()


@pytest.mark.parametrize("preset", ["tiny-llama", "tiny-gemma2"])
def test_fp8_kv_cache_close_to_bf16_kv(preset):
    """e4m3 KV pool (per-head-position scales) vs the bf16 pool: prefill
    logits within quantization noise, greedy rollouts agree early, and
    the graph decode path works (quantized write inside k_attn_dec)."""
    import llm_np_cp_amd as L
    from llm_np_cp_amd.io.loader import random_weights
    from llm_np_cp_amd.models.engine import GPUModel

    cfg = L.preset_config(preset)
    w = random_weights(cfg, seed=50)
    ref = GPUModel(cfg, dict(w), max_seq=128)
    kv8 = GPUModel(cfg, dict(w), max_seq=128, kv_dtype="fp8")
    prompt = np.random.default_rng(51).integers(0, cfg.vocab_size, size=11)

    _, logits_ref = ref.prefill(prompt)
    _, logits_kv8 = kv8.prefill(prompt)
    assert np.argmax(logits_kv8) == np.argmax(logits_ref)
    assert kl_bits(logits_ref.ravel(), logits_kv8.ravel()) < 0.05

    ids_ref = ref.decode(6, greedy=True, use_graph=False)
    kv8.prefill(prompt)
    a = kv8.decode(6, greedy=True, use_graph=False)
    kv8.prefill(prompt)
    b = kv8.decode(6, greedy=True, use_graph=True)
    np.testing.assert_array_equal(a, b)  # graph == eager with fp8 KV
    assert a[0] == ids_ref[0]


def test_fp8_kv_with_fp8_weights():
    """Full fp8 stack: fp8 weights + fp8 KV, decode still tracks the
    all-bf16 engine's first greedy tokens."""
    import llm_np_cp_amd as L
    from llm_np_cp_amd.io.loader import random_weights
    from llm_np_cp_amd.models.engine import GPUModel

    cfg = L.preset_config("tiny-llama")
    w = random_weights(cfg, seed=52)
    ref = GPUModel(cfg, dict(w), max_seq=128)
    f8 = GPUModel(cfg, dict(w), max_seq=128, dtype="fp8", kv_dtype="fp8")
    prompt = np.arange(1, 9)
    _, lr = ref.prefill(prompt)
    _, l8 = f8.prefill(prompt)
    assert np.argmax(l8) == np.argmax(lr)
    ref_ids = ref.decode(4, greedy=True, use_graph=False)
    f8_ids = f8.decode(4, greedy=True, use_graph=True)
    assert f8_ids[0] == ref_ids[0]


def test_auto_max_seq_sizes_from_free_hbm():
    """max_seq=None sizes the KV pool from free HBM (288 GB story):
    tiny model on a 288 GB part must hit the max_position_embeddings
    cap, and the pool must actually be usable."""
    import llm_np_cp_amd as L
    from llm_np_cp_amd.io.loader import random_weights
    from llm_np_cp_amd.models.engine import GPUModel

    cfg = L.preset_config("tiny-llama")
    w = random_weights(cfg, seed=60)
    m = GPUModel(cfg, w, max_seq=None)
    assert m.max_seq == cfg.max_position_embeddings  # tiny model, vast HBM
    m.prefill(np.arange(1, 9))
    ids = m.decode(4, greedy=True)
    assert len(ids) == 4


@pytest.mark.parametrize("preset", ["tiny-llama", "tiny-gemma2"])
def test_batch_decode_rows_match_single_sequence(preset):
    """Lockstep batched decode: every batch row must reproduce the
    single-sequence greedy rollout for its own prompt (per-sequence KV
    pools + batched attention/sampler), including through the graph."""
    import llm_np_cp_amd as L
    from llm_np_cp_amd.io.loader import random_weights
    from llm_np_cp_amd.models.engine import GPUModel

    cfg = L.preset_config(preset)
    w = random_weights(cfg, seed=70)
    rng = np.random.default_rng(71)
    B, P, N = 3, 9, 6
    prompts = rng.integers(0, cfg.vocab_size, size=(B, P))

    singles = []
    single = GPUModel(cfg, dict(w), max_seq=64)
    for b in range(B):
        single.prefill(prompts[b])
        singles.append(single.decode(N, greedy=True, use_graph=False))
    del single
    torch.cuda.empty_cache()

    m = GPUModel(cfg, dict(w), max_seq=64, max_batch=4)
    ids = m.generate_tokens_batch(prompts, N, greedy=True)
    assert ids.shape == (B, N)
    for b in range(B):
        np.testing.assert_array_equal(ids[b], singles[b]), b

    # again without the graph: identical
    m.prefill_batch(prompts)
    ids2 = m.decode_batch(N, greedy=True, use_graph=False)
    np.testing.assert_array_equal(ids, ids2)


def test_batch_decode_fp8_weights_and_kv():
    import llm_np_cp_amd as L
    from llm_np_cp_amd.io.loader import random_weights
    from llm_np_cp_amd.models.engine import GPUModel

    cfg = L.preset_config("tiny-llama")
    w = random_weights(cfg, seed=72)
    rng = np.random.default_rng(73)
    prompts = rng.integers(0, cfg.vocab_size, size=(4, 8))
    m = GPUModel(cfg, w, max_seq=64, max_batch=4, dtype="fp8",
                 kv_dtype="fp8")
    ids = m.generate_tokens_batch(prompts, 5, greedy=True)
    assert ids.shape == (4, 5)
    assert (ids >= 0).all() and (ids < cfg.vocab_size).all()
    # deterministic across runs
    ids2 = m.generate_tokens_batch(prompts, 5, greedy=True)
    np.testing.assert_array_equal(ids, ids2)


def test_batch_decode_ragged_lengths():
    """RAGGED batch: rows prefilled at different lengths decode in
    lockstep from their own device positions and each reproduces its
    single-sequence rollout."""
    import llm_np_cp_amd as L
    from llm_np_cp_amd.io.loader import random_weights
    from llm_np_cp_amd.models.engine import GPUModel

    cfg = L.preset_config("tiny-llama")
    w = random_weights(cfg, seed=80)
    rng = np.random.default_rng(81)
    lens = [5, 11, 8]
    prompts = [rng.integers(0, cfg.vocab_size, size=p) for p in lens]
    N = 6

    singles = []
    single = GPUModel(cfg, dict(w), max_seq=64)
    for p in prompts:
        single.prefill(p)
        singles.append(single.decode(N, greedy=True, use_graph=False))
    del single
    torch.cuda.empty_cache()

    m = GPUModel(cfg, dict(w), max_seq=64, max_batch=4)
    ids = m.generate_tokens_batch(prompts, N, greedy=True)
    assert ids.shape == (3, N)
    for b in range(3):
        np.testing.assert_array_equal(ids[b], singles[b])


@pytest.mark.parametrize("preset,B,force_mx", [
    ("tiny-llama", 2, False), ("tiny-gemma2", 2, False),
    ("tiny-llama", 4, False), ("tiny-gemma2", 4, False),
    ("tiny-llama", 7, False),
    ("tiny-llama", 2, True), ("tiny-gemma2", 4, True)])
def test_batch_decode_fused_paths_match_single_fp8(preset, B, force_mx,
                                                   monkeypatch):
    """The fused fp8 batch paths (skinny MFMA GEMM default; the multi-x
    GEMV reference path via LLM_BATCH_MX_MAX) must reproduce each row's
    single-sequence fp8 rollout (ragged lengths)."""
    if force_mx:
        monkeypatch.setenv("LLM_BATCH_MX_MAX", "16")
    import llm_np_cp_amd as L
    from llm_np_cp_amd.io.loader import random_weights
    from llm_np_cp_amd.models.engine import GPUModel

    cfg = L.preset_config(preset)
    w = random_weights(cfg, seed=90)
    rng = np.random.default_rng(91)
    lens = ([6, 10, 7, 9, 5, 8, 11])[:B]
    prompts = [rng.integers(0, cfg.vocab_size, size=p) for p in lens]
    N = 5

    singles = []
    single = GPUModel(cfg, dict(w), max_seq=64, dtype="fp8")
    for p in prompts:
        single.prefill(p)
        singles.append(single.decode(N, greedy=True, use_graph=False))
    del single
    torch.cuda.empty_cache()

    m = GPUModel(cfg, dict(w), max_seq=64, max_batch=B, dtype="fp8")
    ids = m.generate_tokens_batch(prompts, N, greedy=True)
    for b in range(B):
        np.testing.assert_array_equal(ids[b], singles[b]), b


def test_continuous_batching_primitives():
    """prefill_row / decode_rows / compact_row: rows join a live group,
    decode in lockstep from their own positions, retire and compact —
    surviving rows must continue their exact single-sequence rollout."""
    import llm_np_cp_amd as L
    from llm_np_cp_amd.io.loader import random_weights
    from llm_np_cp_amd.models.engine import GPUModel

    cfg = L.preset_config("tiny-llama")
    w = random_weights(cfg, seed=95)
    rng = np.random.default_rng(96)
    p0 = rng.integers(0, cfg.vocab_size, size=6)
    p1 = rng.integers(0, cfg.vocab_size, size=9)
    p2 = rng.integers(0, cfg.vocab_size, size=4)

    # single-sequence references
    single = GPUModel(cfg, dict(w), max_seq=64, dtype="fp8")
    refs = {}
    for name, p, n in (("p0", p0, 4), ("p1", p1, 9), ("p2", p2, 6)):
        single.prefill(p)
        refs[name] = list(single.decode(n, greedy=True, use_graph=False))
    del single
    torch.cuda.empty_cache()

    m = GPUModel(cfg, dict(w), max_seq=64, max_batch=4, dtype="fp8")
    m.reset()
    m.bt_nout.zero_()
    m._host_lens = [0] * 4
    m._batch_n = 0
    ids = {0: [], 1: []}
    m.prefill_row(0, p0, greedy=True)
    m.prefill_row(1, p1, greedy=True)
    ids[0].append(int(m.bt_ring[0, 0].item()))
    ids[1].append(int(m.bt_ring[1, 0].item()))
    out = m.decode_rows(2, 3, greedy=True)
    ids[0] += [int(t) for t in out[0]]
    ids[1] += [int(t) for t in out[1]]
    assert ids[0] == refs["p0"][:4]
    assert ids[1] == refs["p1"][:4]
    # retire row 0; row 1 compacts into slot 0; a NEW row joins slot 1
    m.compact_row(0, 1)
    ids = {0: ids[1], 1: []}
    m.prefill_row(1, p2, greedy=True)
    ids[1].append(int(m.bt_ring[1, 0].item()))
    out = m.decode_rows(2, 5, greedy=True)
    ids[0] += [int(t) for t in out[0]]
    ids[1] += [int(t) for t in out[1]]
    assert ids[0] == refs["p1"][:9], (ids[0], refs["p1"])
    assert ids[1] == refs["p2"][:6], (ids[1], refs["p2"])


def test_fp4_engine_decode_on_grid_weights():
    """dtype='fp4' (MXFP4 decode weights): with projection weights drawn
    from the exactly-representable e2m1 grid the fp4 rollout must match
    the bf16 engine's greedy ids."""
    import llm_np_cp_amd as L
    from llm_np_cp_amd.io.loader import random_weights
    from llm_np_cp_amd.models.engine import GPUModel

    cfg = L.preset_config("tiny-llama")
    w = random_weights(cfg, seed=97)
    # scale must be a POWER OF TWO (e8m0): grid/32 keeps every value
    # exactly on the e2m1-times-2^-5 lattice -> lossless quantization
    grid = np.array([0.0, 0.5, 1.0, 1.5, 2.0, 3.0, 4.0, 6.0,
                     -0.5, -1.0, -1.5, -2.0, -3.0, -4.0, -6.0],
                    dtype=np.float32) / 32.0
    rng = np.random.default_rng(98)
    for k in w:
        # embed_tokens included: it is the TIED lm_head, which the fp4
        # engine quantizes
        if ("proj.weight" in k or "embed_tokens" in k
                or k in ("lm_head.weight",)):
            a = grid[rng.integers(0, len(grid), size=w[k].shape)]
            a.reshape(-1, 32)[:, 0] = 6.0 / 32.0  # pin block scales
            w[k] = a.astype(np.float32)
    bf = GPUModel(cfg, dict(w), max_seq=128)
    f4 = GPUModel(cfg, dict(w), max_seq=128, dtype="fp4")
    assert not hasattr(f4, "lm_head_q"), "max_batch=1 fp4 is single-copy"
    prompt = np.arange(1, 9)
    _, la = bf.prefill(prompt)
    a = bf.decode(6, greedy=True, use_graph=False)
    _, lb = f4.prefill(prompt)
    b = f4.decode(6, greedy=True, use_graph=True)
    # grid weights quantize losslessly: the fp4w bf16-MFMA prefill must
    # track the bf16 engine tightly
    np.testing.assert_allclose(lb, la, rtol=5e-2, atol=5e-2)
    assert a[0] == b[0], (a, b)
    # at least the early steps agree (kernel rounding may diverge later)
    assert list(a[:3]) == list(b[:3]), (a, b)


def test_fp4_batch_decode_keeps_fp8_copy():
    """fp4 + max_batch>1: batch pools require the fp8 skinny path, so
    the fp8 copy is retained and batched decode works."""
    import llm_np_cp_amd as L
    from llm_np_cp_amd.io.loader import random_weights
    from llm_np_cp_amd.models.engine import GPUModel

    cfg = L.preset_config("tiny-llama")
    w = random_weights(cfg, seed=99)
    m = GPUModel(cfg, w, max_seq=64, max_batch=4, dtype="fp4")
    assert hasattr(m, "lm_head_q")
    ids = m.generate_tokens_batch(
        [np.arange(1, 7), np.arange(2, 9)], 4, greedy=True)
    assert ids.shape == (2, 4)


def test_speculative_decode_gpu_matches_target_greedy():
    """Greedy speculative output == the target's own greedy chain under
    the verify pass's numerics (forward_positions single steps), with a
    mismatched draft; a same-weights draft gets 100% acceptance."""
    import llm_np_cp_amd as L
    from llm_np_cp_amd.io.loader import random_weights
    from llm_np_cp_amd.models.engine import GPUModel
    from llm_np_cp_amd.runtime.generate import ByteTokenizer
    from llm_np_cp_amd.runtime.speculative import generate_speculative

    cfg = L.preset_config("tiny-llama")
    target = GPUModel(cfg, random_weights(cfg, seed=0), max_seq=256)
    draft = GPUModel(cfg, random_weights(cfg, seed=9), max_seq=256)
    tok = ByteTokenizer()
    prompt = "Once upon a time"
    pid = tok.encode(prompt)

    # reference: greedy chain via the same verify-pass kernels
    target.reset()
    vl = target.forward_positions(np.asarray(pid, np.int32), 0)
    ref = [int(np.argmax(vl[-1]))]
    pos = len(pid)
    for _ in range(19):
        vl = target.forward_positions(np.asarray([ref[-1]], np.int32), pos)
        pos += 1
        ref.append(int(np.argmax(vl[-1])))

    res = generate_speculative(prompt, tok, draft, target, max_tokens=20,
                               k=4, stop_on_eos=False)
    assert res.token_ids == ref
    assert res.spec_stats["verify_passes"] >= 1

    same = GPUModel(cfg, random_weights(cfg, seed=0), max_seq=256)
    res2 = generate_speculative(prompt, tok, same, target, max_tokens=20,
                                k=4, stop_on_eos=False)
    assert res2.token_ids == ref
    s = res2.spec_stats
    assert s["accepted"] == s["proposed"] > 0


# ---------------------------------------------------------------------
# Mixtral sparse MoE (router + expert-indexed GEMVs / dense prefill loop)
# ---------------------------------------------------------------------

def _bf16_round(w):
    """Round fp32 weights through bf16 so the oracle makes the SAME
    discrete routing decisions as the bf16 engine (router top-k is a
    step function; raw-fp32-vs-bf16 near-ties would flip experts)."""
    return {k: torch.from_numpy(v).to(torch.bfloat16).float().numpy()
            for k, v in w.items()}


def _mixtral_pair(seed=4):
    import llm_np_cp_amd as L
    from llm_np_cp_amd.io.loader import random_weights
    from llm_np_cp_amd.models.engine import GPUModel
    from llm_np_cp_amd.models.numpy_ref import NumpyModel, NumpyKVCache

    cfg = L.preset_config("tiny-mixtral")
    w = _bf16_round(random_weights(cfg, seed=seed))
    gpu = GPUModel(cfg, w, max_seq=256)
    ref = NumpyModel(cfg, dict(w))
    ref.make_cache = lambda n: NumpyKVCache(cfg, n)
    return cfg, gpu, ref


def test_mixtral_prefill_logits_match_oracle():
    from llm_np_cp_amd.models.numpy_ref import NumpyKVCache
    cfg, gpu, ref = _mixtral_pair(seed=4)
    rng = np.random.default_rng(0)
    ids = rng.integers(0, cfg.vocab_size, size=13)
    ref_logits = ref.forward(ids, NumpyKVCache(cfg, 64), 0)
    got = gpu.forward(ids, gpu.make_cache(64), 0)[0]
    np.testing.assert_allclose(got, ref_logits[-1], rtol=0.15, atol=0.15)
    assert np.argmax(got) == np.argmax(ref_logits[-1])
    assert kl_bits(ref_logits[-1], got) < 0.02
    assert topk_overlap(got, ref_logits[-1], k=8) >= 0.875


def test_mixtral_greedy_decode_matches_oracle():
    import llm_np_cp_amd as L
    cfg, gpu, ref = _mixtral_pair(seed=5)
    tok = L.ByteTokenizer()
    p = L.SamplingParams(strategy="greedy")
    r_ref = L.generate("Once upon", tok, ref, max_tokens=12, stream=False,
                       params=p, stop_on_eos=False)
    r_gpu = L.generate("Once upon", tok, gpu, max_tokens=12, stream=False,
                       params=p, stop_on_eos=False)
    assert r_ref.token_ids == r_gpu.token_ids


def test_mixtral_graph_decode_matches_eager():
    cfg, gpu, _ = _mixtral_pair(seed=6)
    prompt = np.arange(1, 9)
    gpu.prefill(prompt)
    a = gpu.decode(10, greedy=True, use_graph=False)
    gpu.prefill(prompt)
    b = gpu.decode(10, greedy=True, use_graph=True)
    np.testing.assert_array_equal(a, b)


def test_mixtral_fp8_decode_close_to_bf16():
    import llm_np_cp_amd as L
    from llm_np_cp_amd.io.loader import random_weights
    from llm_np_cp_amd.models.engine import GPUModel

    cfg = L.preset_config("tiny-mixtral")
    w = random_weights(cfg, seed=7)
    bf = GPUModel(cfg, w, max_seq=128)
    f8 = GPUModel(cfg, dict(w), max_seq=128, dtype="fp8")
    prompt = np.arange(1, 9)
    _, logits_bf = bf.prefill(prompt)
    _, logits_f8 = f8.prefill(prompt)
    assert np.argmax(logits_f8) == np.argmax(logits_bf)
    assert topk_overlap(logits_f8[0], logits_bf[0], k=8) >= 0.75
    ids_bf = bf.decode(6, greedy=True, use_graph=False)
    f8.prefill(prompt)
    ids_f8 = f8.decode(6, greedy=True, use_graph=True)  # graph too
    assert ids_f8[0] == ids_bf[0]


def test_mixtral_batch_rows_match_b1():
    """Batched MoE decode (generic path: per-row routing + dense expert
    loop) must reproduce the B=1 rollout of the same machinery."""
    import llm_np_cp_amd as L
    from llm_np_cp_amd.io.loader import random_weights
    from llm_np_cp_amd.models.engine import GPUModel

    cfg = L.preset_config("tiny-mixtral")
    w = _bf16_round(random_weights(cfg, seed=8))
    rng = np.random.default_rng(9)
    B, P, N = 3, 9, 6
    prompts = rng.integers(0, cfg.vocab_size, size=(B, P))
    m = GPUModel(cfg, w, max_seq=64, max_batch=4)
    singles = [m.generate_tokens_batch(prompts[b:b + 1], N)[0]
               for b in range(B)]
    ids = m.generate_tokens_batch(prompts, N, greedy=True)
    for b in range(B):
        np.testing.assert_array_equal(ids[b], singles[b])

