"""Autoregressive generation loop with streaming output.

Reference parity: ``generate(prompt, tokenizer, model, max_tokens,
kv_cache, config) -> str`` with per-token streaming prints
(``/root/reference/llama3.2_model.py:865-902``).  Deliberate fix: later
steps feed back the sampled token *id* — the reference re-tokenized the
decoded text of the previous token (``llama3.2_model.py:874-878``), which
corrupts the stream for any token whose text does not round-trip.

Works with both the NumPy oracle model and the GPU engine: a model here
is anything with ``.forward(ids, cache, pos0) -> logits`` and
``.make_cache(max_seq)``.  Cache-less mode (``use_cache=False``) re-runs
the full prefix each step — a correctness/debug mode, as in the reference
(SURVEY §3.3).
"""

from __future__ import annotations

import sys
import time
from dataclasses import dataclass
from typing import Callable, List, Optional

import numpy as np

from .sampling import SamplingParams, sample_token


@dataclass
class GenerateResult:
    text: str
    token_ids: List[int]
    prefill_time_s: float = 0.0
    decode_time_s: float = 0.0
    finish_reason: str = "length"   # "stop" (EOS / stop string) | "length"
    # per emitted token, when requested: {"id", "token", "logprob",
    # "top": [{"id", "token", "logprob"}, ...]} — log-softmax of the RAW
    # model logits (pre-temperature/filter, OpenAI semantics)
    logprobs: Optional[List[dict]] = None

    @property
    def decode_tokens_per_s(self) -> float:
        n = len(self.token_ids)
        return n / self.decode_time_s if self.decode_time_s > 0 else 0.0


class _StopScan:
    """Incremental stop-sequence scanner for streaming: emits only text
    that can no longer become part of a stop string (holds back
    max(len(stop))-1 chars), cuts the stream at the earliest match."""

    def __init__(self, stops: List[str], emit):
        self.stops = stops
        self.emit = emit
        self.hold = max(len(s) for s in stops) - 1
        self.buf = ""
        self.hit = False

    def feed(self, piece: str) -> bool:
        """Returns True once a stop string appeared (emission stops
        BEFORE it)."""
        if self.hit:
            return True
        self.buf += piece
        found = [i for i in (self.buf.find(s) for s in self.stops)
                 if i >= 0]
        if found:
            cut = min(found)
            if cut and self.emit:
                self.emit(self.buf[:cut])
            self.hit = True
            return True
        if self.hold == 0 or len(self.buf) > self.hold:
            safe = self.buf if self.hold == 0 else self.buf[:-self.hold]
            if safe and self.emit:
                self.emit(safe)
            self.buf = self.buf[len(safe):]
        return False

    def flush(self):
        if not self.hit and self.buf and self.emit:
            self.emit(self.buf)
            self.buf = ""


def _logprob_entry(logits: np.ndarray, chosen: int, top_n: int,
                   tokenizer) -> dict:
    """Stable log-softmax of the raw logits row -> chosen-token logprob
    + the top_n most likely alternatives."""
    z = logits.astype(np.float64)
    z = z - z.max()
    lse = np.log(np.exp(z).sum())
    lp = z - lse
    top_ids = np.argsort(-lp)[:max(top_n, 0)]
    return {
        "id": int(chosen),
        "token": tokenizer.decode([int(chosen)]),
        "logprob": float(lp[chosen]),
        "top": [{"id": int(i), "token": tokenizer.decode([int(i)]),
                 "logprob": float(lp[i])} for i in top_ids],
    }


def _truncate_at_stop(ids: List[int], tokenizer, stops: List[str]):
    """Earliest stop occurrence in decode(ids) -> (ids', text',
    hit).  The stop string itself is excluded (OpenAI semantics); ids'
    is the minimal token prefix covering text'."""
    text = tokenizer.decode(ids)
    found = [i for i in (text.find(s) for s in stops) if i >= 0]
    if not found:
        return ids, text, False
    keep = text[:min(found)]
    out: List[int] = []
    for t in ids:
        if len(tokenizer.decode(out)) >= len(keep):
            break
        out.append(int(t))
    return out, keep, True


def generate(prompt: str, tokenizer, model, max_tokens: int = 200,
             kv_cache=None, params: Optional[SamplingParams] = None,
             use_cache: bool = True, stream: bool = True,
             stop_on_eos: bool = True,
             stop: Optional[List[str]] = None,
             logprobs: Optional[int] = None,
             on_token: Optional[Callable[[str], None]] = None) -> GenerateResult:
    """Generate up to max_tokens continuation tokens of prompt.

    ``stop``: optional stop strings — generation ends at the earliest
    occurrence in the decoded text, which is excluded from the result
    (OpenAI semantics).  Streaming holds back max(len(stop))-1 chars so
    a stop string spanning token boundaries is never emitted.

    ``logprobs``: when an int N >= 0, GenerateResult.logprobs carries
    per-token raw-model logprobs with the top-N alternatives (takes the
    logits-visible step loop — the device-side sampler never surfaces
    the full distribution).

    Seeds: the GPU fast path's device sampler derives its stream from
    the ENGINE seed + a device counter (baked into the captured graph),
    so ``params.seed`` is honored on the logits-visible paths (CPU
    oracle; logprobs / logit_bias / top-k / top-p requests) but not by
    the in-graph device sampler."""
    params = params or SamplingParams()
    rng = np.random.default_rng(params.seed)
    prompt_ids = list(tokenizer.encode(prompt))
    stops = [s for s in (stop or []) if s]
    # HF configs store eos_token_id as an int OR a list (e.g.
    # Llama-3.2-Instruct: [128001, 128008, 128009]) — normalize to a set
    eos = getattr(model.config, "eos_token_id", None)
    eos_set = (set() if eos is None else
               {int(eos)} if np.isscalar(eos) else {int(e) for e in eos})

    if kv_cache is None:
        kv_cache = model.make_cache(len(prompt_ids) + max_tokens + 1)

    emit = on_token
    if emit is None and stream:
        emit = lambda s: (sys.stdout.write(s), sys.stdout.flush())
    scan = _StopScan(stops, emit) if stops else None

    def _finish(ids: List[int], hit_eos: bool):
        """Apply stop-string truncation; build the result text."""
        ids = [int(i) for i in ids]
        if stops:
            ids, text, hit = _truncate_at_stop(ids, tokenizer, stops)
            if scan is not None:
                scan.flush()
            return ids, text, "stop" if (hit or hit_eos) else "length"
        return ids, tokenizer.decode(ids), "stop" if hit_eos else "length"

    # fast path: device-side chunked decode loop (GPU engine, greedy or
    # min-p, cached mode) — same sampler semantics, device RNG
    if (use_cache and logprobs is None and not params.logit_bias
            and hasattr(model, "generate_tokens")
            and params.strategy in ("greedy", "min_p")):
        t0 = time.perf_counter()

        def _emit(ids_chunk):
            piece = tokenizer.decode(list(ids_chunk))
            if scan is not None:
                scan.feed(piece)
            elif emit:
                emit(piece)

        def _stop_fn(all_ids):
            return any(s in tokenizer.decode(list(all_ids)) for s in stops)

        ids = model.generate_tokens(
            prompt_ids, max_tokens,
            greedy=params.strategy == "greedy", min_p=params.min_p,
            eos_id=eos_set if stop_on_eos else None, on_ids=_emit,
            temperature=params.temperature,
            stop_fn=_stop_fn if stops else None)
        dt = time.perf_counter() - t0
        tp = getattr(model, "last_prefill_time_s", 0.0)
        hit_eos = bool(ids) and stop_on_eos and int(ids[-1]) in eos_set
        ids, text, reason = _finish(ids, hit_eos)
        if not stops and len(ids) >= max_tokens and not hit_eos:
            reason = "length"
        return GenerateResult(text=text, token_ids=ids,
                              prefill_time_s=tp,
                              decode_time_s=max(dt - tp, 1e-9),
                              finish_reason=reason)

    out_ids: List[int] = []
    t0 = time.perf_counter()
    logits = model.forward(np.asarray(prompt_ids, dtype=np.int64), kv_cache, 0)
    t_prefill = time.perf_counter() - t0

    hit_eos = False
    lp_entries: List[dict] = [] if logprobs is not None else None
    t1 = time.perf_counter()
    for _ in range(max_tokens):
        row = np.asarray(logits[-1], dtype=np.float32)
        next_id = sample_token(row, params, rng)
        out_ids.append(next_id)
        if lp_entries is not None:
            lp_entries.append(_logprob_entry(row, next_id, logprobs,
                                             tokenizer))
        piece = tokenizer.decode([next_id])
        if scan is not None:
            if scan.feed(piece):
                break
        elif emit:
            emit(piece)
        if stop_on_eos and next_id in eos_set:
            hit_eos = True
            break
        if use_cache:
            logits = model.forward(np.asarray([next_id], dtype=np.int64),
                                   kv_cache, kv_cache.seq_len)
        else:
            full = prompt_ids + out_ids
            fresh = model.make_cache(len(full) + 1)
            logits = model.forward(np.asarray(full, dtype=np.int64), fresh, 0)
    t_decode = time.perf_counter() - t1

    out_ids, text, reason = _finish(out_ids, hit_eos)
    if lp_entries is not None:
        lp_entries = lp_entries[:len(out_ids)]  # stop truncation
    return GenerateResult(
        text=text,
        token_ids=out_ids,
        prefill_time_s=t_prefill,
        decode_time_s=t_decode,
        finish_reason=reason,
        logprobs=lp_entries,
    )


def _hub_download(repo_id: str) -> str:
    """Fetch a checkpoint from the HF hub (reference parity:
    ``snapshot_download``, llama3.2_model.py:1090).  Raises a clear
    FileNotFoundError when offline or the repo is unavailable."""
    try:
        from huggingface_hub import snapshot_download
    except ImportError as e:
        raise FileNotFoundError(
            f"{repo_id!r} looks like an HF repo id but huggingface_hub "
            f"is not installed") from e
    try:
        return snapshot_download(
            repo_id,
            allow_patterns=["*.json", "*.safetensors", "tokenizer*",
                            "*.model"])
    except Exception as e:
        raise FileNotFoundError(
            f"could not download {repo_id!r} from the HF hub (no network "
            f"access, auth, or unknown repo): {type(e).__name__}: {e}"
        ) from e


class ByteTokenizer:
    """Self-contained fallback tokenizer (byte-level) for environments
    without HF tokenizer files; real checkpoints use ``AutoTokenizer``."""

    vocab_size = 256

    def encode(self, text: str) -> List[int]:
        return list(text.encode("utf-8"))

    def decode(self, ids) -> str:
        return bytes(int(i) % 256 for i in ids).decode("utf-8", errors="replace")


def load_model(model_dir_or_preset: str, backend: str = "auto",
               device: str = "cuda", dtype: str = "bf16",
               max_seq: int = 4096, seed: int = 0,
               kv_dtype: str = "bf16", max_batch: int = 1,
               lora: Optional[str] = None):
    """Reference-parity entry (``load_model`` -> (tokenizer, model, config),
    ``llama3.2_model.py:1082-1099``).

    ``model_dir_or_preset``: a checkpoint directory (config.json +
    safetensors), a preset name (random-init synthetic weights), or an
    HF hub repo id like ``"meta-llama/Llama-3.2-1B"`` — the hub path
    mirrors the reference's ``snapshot_download`` flow
    (``llama3.2_model.py:1090``) and fails with a clear error when the
    machine has no network access.
    ``backend``: "numpy" (CPU oracle), "gpu" (HIP engine), or "auto".
    ``max_batch``: >1 allocates per-sequence KV pools on the GPU engine
    (lockstep batched decode / continuous-batching server); ignored by
    the NumPy oracle.
    ``lora``: optional PEFT adapter directory (adapter_config.json +
    adapter_model.safetensors) merged into the base weights at load
    (W' = W + (alpha/r)·B@A) — the adapted model then runs at full
    native speed on either engine.
    """
    import os

    from ..core.config import preset_config, PRESETS
    from ..io.loader import load_config, load_weights_numpy, random_weights

    weights_dir = None
    if os.path.isdir(model_dir_or_preset):
        config = load_config(model_dir_or_preset)
        weights_dir = tok_dir = model_dir_or_preset
    elif model_dir_or_preset.lower() in PRESETS:
        config = preset_config(model_dir_or_preset)
        weights = random_weights(config, seed=seed)
        tok_dir = None
    elif "/" in model_dir_or_preset:
        local = _hub_download(model_dir_or_preset)
        config = load_config(local)
        weights_dir = tok_dir = local
    else:
        raise FileNotFoundError(
            f"{model_dir_or_preset!r} is neither a directory, a preset "
            f"({sorted(PRESETS)}), nor an HF repo id")

    if backend == "auto":
        try:
            import torch
            backend = "gpu" if torch.cuda.is_available() else "numpy"
        except ImportError:
            backend = "numpy"

    if weights_dir is not None:
        if backend == "gpu" and lora is None:
            # one-tensor host-memory peak: the upload converts each
            # tensor to bf16/fp8 on arrival, so an eager fp32 dict
            # (~280 GB host RAM for 70B) is pure waste
            from ..io.loader import LazyCheckpointWeights
            weights = LazyCheckpointWeights(weights_dir)
        else:
            # the oracle computes from (and mutates) the host dict;
            # LoRA merge also needs a writable dict
            weights = load_weights_numpy(weights_dir)

    if lora is not None:
        from ..io.loader import apply_lora
        apply_lora(weights, lora)

    tokenizer = None
    if tok_dir is not None:
        try:
            from transformers import AutoTokenizer
            tokenizer = AutoTokenizer.from_pretrained(tok_dir)
            # a dir with config.json but no tokenizer files can yield a
            # "working" object that encodes everything to [] — verify
            if not tokenizer.encode("test"):
                tokenizer = None
        except Exception:
            tokenizer = None
    if tokenizer is None:
        if tok_dir is not None:
            print(f"# no usable tokenizer files in {tok_dir}; using the "
                  f"byte-level fallback tokenizer", file=sys.stderr)
        tokenizer = ByteTokenizer()

    if backend == "numpy":
        from ..models.numpy_ref import NumpyModel, NumpyKVCache

        model = NumpyModel(config, weights)
        model.make_cache = lambda n: NumpyKVCache(config, n)
    elif backend == "gpu":
        from ..models.engine import GPUModel

        model = GPUModel(config, weights, dtype=dtype, max_seq=max_seq,
                         kv_dtype=kv_dtype, max_batch=max_batch)
    else:
        raise ValueError(f"unknown backend {backend!r}")
    return tokenizer, model, config
