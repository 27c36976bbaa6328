"""CLI entry — parity with the reference's ``python <file>.py`` scripts
(``/root/reference/llama3.2_model.py:1101-1108``: load model, generate 200
tokens from a prompt with streaming output), plus flags the reference
hard-coded.

    python -m llm_np_cp_amd "Once upon a time" --model llama-3.2-1b \
        --max-tokens 200 --strategy min_p --backend auto
"""

import argparse
import sys
import time


def main(argv=None):
    import llm_np_cp_amd as _pkg

    ap = argparse.ArgumentParser(prog="llm_np_cp_amd")
    ap.add_argument("--version", action="version",
                    version=f"llm_np_cp_amd {_pkg.__version__}")
    ap.add_argument("prompt", nargs="?", default="Once upon a time")
    ap.add_argument("--model", default="llama-3.2-1b",
                    help="checkpoint directory or preset name")
    ap.add_argument("--max-tokens", type=int, default=200)
    ap.add_argument("--backend", default="auto",
                    choices=["auto", "gpu", "numpy"])
    ap.add_argument("--dtype", default="bf16",
                    choices=["bf16", "fp8", "fp4"])
    ap.add_argument("--kv-dtype", default="bf16", choices=["bf16", "fp8"])
    ap.add_argument("--strategy", default="min_p",
                    choices=["min_p", "greedy", "top_k", "top_p",
                             "temperature"])
    ap.add_argument("--min-p", type=float, default=0.1)
    ap.add_argument("--temperature", type=float, default=1.0)
    ap.add_argument("--seed", type=int, default=None)
    ap.add_argument("--max-seq", type=int, default=4096)
    ap.add_argument("--lora", default=None,
                    help="PEFT adapter dir merged into the weights at "
                         "load (W' = W + (alpha/r) B@A)")
    ap.add_argument("--stop", action="append", default=None,
                    help="stop string (repeatable); excluded from output")
    ap.add_argument("--draft", default=None,
                    help="draft model (dir/preset) — enables speculative "
                         "decoding (greedy mode is token-identical to the "
                         "target's own decode)")
    ap.add_argument("--draft-dtype", default=None,
                    help="draft weight dtype (default: same as --dtype)")
    ap.add_argument("--spec-k", type=int, default=4,
                    help="draft tokens per verify pass")
    ap.add_argument("--no-cache", action="store_true",
                    help="stateless re-prefill each step (debug mode, "
                         "reference use_cache=False path)")
    args = ap.parse_args(argv)

    import llm_np_cp_amd as L

    tok, model, cfg = L.load_model(args.model, backend=args.backend,
                                   dtype=args.dtype, max_seq=args.max_seq,
                                   kv_dtype=args.kv_dtype, lora=args.lora)
    params = L.SamplingParams(strategy=args.strategy, min_p=args.min_p,
                              temperature=args.temperature, seed=args.seed)
    t0 = time.time()
    if args.draft is not None:
        _, draft, _ = L.load_model(args.draft, backend=args.backend,
                                   dtype=args.draft_dtype or args.dtype,
                                   max_seq=args.max_seq)
        out = L.generate_speculative(
            args.prompt, tok, draft, model, max_tokens=args.max_tokens,
            k=args.spec_k, params=params,
            on_token=lambda s: (sys.stdout.write(s), sys.stdout.flush()))
        s = out.spec_stats
        print(f"\n[speculative: {s['accepted']}/{s['proposed']} drafts "
              f"accepted over {s['verify_passes']} verify passes]",
              file=sys.stderr)
    else:
        out = L.generate(args.prompt, tok, model,
                         max_tokens=args.max_tokens, params=params,
                         use_cache=not args.no_cache, stream=True,
                         stop=args.stop)
    dt = time.time() - t0
    print(f"\n[{len(out.token_ids)} tokens in {dt:.2f}s; "
          f"prefill {out.prefill_time_s * 1e3:.0f} ms, "
          f"decode {out.decode_tokens_per_s:.1f} tok/s]", file=sys.stderr)


if __name__ == "__main__":
    main()
