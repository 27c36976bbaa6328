#!/usr/bin/env python3
"""Speculative-decoding benchmark: target-only greedy vs draft+verify.

With synthetic random weights the only honest pair is a QUANTIZED
SELF-DRAFT (same weights, cheaper dtype) — independent random models
never agree, so acceptance ~0 and speculation cannot win.
Real speedups need a trained draft/target pair; this harness is the
measurement tool for that day (and a correctness soak meanwhile: the
speculative output must equal the greedy chain).

    python tools/bench_spec.py --model llama-3.1-8b \
        --target-dtype bf16 --draft-dtype fp4 --max-tokens 64 --k 4
    python tools/bench_spec.py --model tiny-llama --backend numpy
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama-3.1-8b")
    ap.add_argument("--draft-model", default=None,
                    help="draft preset (default: same model = self-draft)")
    ap.add_argument("--target-dtype", default="bf16",
                    choices=["bf16", "fp8", "fp4"])
    ap.add_argument("--draft-dtype", default="fp4",
                    choices=["bf16", "fp8", "fp4"])
    ap.add_argument("--backend", default="gpu", choices=["gpu", "numpy"])
    ap.add_argument("--max-tokens", type=int, default=64)
    ap.add_argument("--k", type=int, default=4)
    ap.add_argument("--prompt", default="Once upon a time")
    ap.add_argument("--max-seq", type=int, default=512)
    ap.add_argument("--strategy", default="greedy",
                    choices=["greedy", "min_p", "top_k", "top_p",
                             "temperature"],
                    help="greedy: token-identical verify; others: "
                         "stochastic speculative sampling")
    ap.add_argument("--min-p", type=float, default=0.1)
    ap.add_argument("--temperature", type=float, default=1.0)
    ap.add_argument("--seed", type=int, default=0)
    args = ap.parse_args()

    import llm_np_cp_amd as L
    from llm_np_cp_amd.runtime.speculative import generate_speculative

    if args.backend == "gpu":
        from csrc.build import ensure_built
        ensure_built()
    tok, target, cfg = L.load_model(args.model, backend=args.backend,
                                    dtype=args.target_dtype,
                                    max_seq=args.max_seq, seed=0)
    _, draft, _ = L.load_model(args.draft_model or args.model,
                               backend=args.backend,
                               dtype=args.draft_dtype,
                               max_seq=args.max_seq, seed=0)

    p = L.SamplingParams(strategy=args.strategy, min_p=args.min_p,
                         temperature=args.temperature, seed=args.seed)
    base = L.generate(args.prompt, tok, target, max_tokens=args.max_tokens,
                      params=p, stream=False, stop_on_eos=False)
    t0 = time.perf_counter()
    base = L.generate(args.prompt, tok, target, max_tokens=args.max_tokens,
                      params=p, stream=False, stop_on_eos=False)
    t_base = time.perf_counter() - t0

    res = generate_speculative(args.prompt, tok, draft, target,
                               max_tokens=args.max_tokens, k=args.k,
                               stop_on_eos=False, params=p)
    t0 = time.perf_counter()
    res = generate_speculative(args.prompt, tok, draft, target,
                               max_tokens=args.max_tokens, k=args.k,
                               stop_on_eos=False, params=p)
    t_spec = time.perf_counter() - t0

    s = res.spec_stats
    acc = s["accepted"] / max(s["proposed"], 1)
    n = len(res.token_ids)
    print(f"baseline greedy : {n / t_base:8.1f} tok/s "
          f"({t_base * 1e3:.1f} ms)")
    print(f"speculative k={args.k}: {n / t_spec:8.1f} tok/s "
          f"({t_spec * 1e3:.1f} ms)  acceptance {acc:.0%} "
          f"({s['accepted']}/{s['proposed']}, "
          f"{s['verify_passes']} verify passes)")
    if args.strategy == "greedy":
        same = res.token_ids == base.token_ids
        note = ("identical" if same else
                "differ — expected when verify-pass numerics (GEMM) and "
                "the decode GEMV round differently")
        print(f"token match vs baseline chain: {same} ({note})")
    else:
        print("stochastic mode: per-token distribution equals target-only "
              "sampling (chain identity not expected)")


if __name__ == "__main__":
    main()
