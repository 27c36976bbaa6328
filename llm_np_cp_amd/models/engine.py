"""GPU inference engine: Llama-3.2 / Gemma-2 forward on hand-written
CDNA4 HIP kernels, PyTorch tensors as memory containers only.

MI355X-first design (SURVEY §7 stages 2-4):
  - every compute op is a kernel from ``csrc/llm_ops.hip`` — no ATen math
    in the forward pass, no eager fallback (ops fail loudly if the
    extension is missing);
  - preallocated KV pool per layer (in-place writes, no concat);
  - decode state (seq length, sampled token, RNG counter, output ring)
    lives in device memory so a decode step is hipGraph-replayable with
    zero host round-trips (torch.cuda.CUDAGraph == hipGraph on ROCm);
  - tensor parallelism: head/row/col-sharded weights per rank, RCCL
    all-reduce after o_proj and down_proj, all-gather for logits
    (world_size==1 skips collectives — same code path).

Reference parity: this replaces the CuPy model classes
(``/root/reference/llama3.2_model.py:334-822``) — same capability
(forward with KV cache, returns logits), built from the architectural
spec in SURVEY §2.4 rather than translated.
"""

from __future__ import annotations

from typing import Dict, Optional

import numpy as np
import torch

from ..core.config import ModelConfig
from ..ops import hip_ops as ho
from ..parallel import tp as tpu
from ..utils.tracing import trace_range


class DeviceCacheHandle:
    """Host-side mirror of the device seq-length (generate()-compatible)."""

    def __init__(self, engine):
        self.engine = engine
        self.seq_len = 0


class GPUModel:
    def __init__(self, config: ModelConfig, weights: Dict[str, np.ndarray],
                 dtype: str = "bf16", max_seq: Optional[int] = 4096,
                 prefill_chunk: int = 2048, device: Optional[str] = None,
                 seed: int = 0, force_tp_path: bool = False,
                 kv_dtype: str = "bf16", max_batch: int = 1):
        """``max_seq=None`` sizes the KV pool from FREE HBM after the
        weights land (288 GB per MI355X — SURVEY §5 long-context), capped
        at config.max_position_embeddings.  ``max_batch>1`` allocates
        per-sequence KV pools for lockstep batched decode
        (prefill_batch/decode_batch; beyond-parity serving capability)."""
        if not torch.cuda.is_available():
            raise RuntimeError("GPUModel requires a GPU (MI355X)")
        self.config = config
        self.seed = seed
        assert 1 <= max_batch <= 256
        self.max_batch = max_batch
        self._pb = 0  # current sequence row during per-sequence prefill
        self.dtype = dtype
        if dtype not in ("bf16", "fp8", "fp4"):
            raise ValueError(f"unsupported dtype {dtype!r}")
        # fp4 = MXFP4 weights on the single-sequence DECODE GEMVs (half
        # the fp8 stream); prefill and batched decode run on the fp8
        # copy (the fp8 machinery flag covers both quantized modes)
        self.fp8 = dtype in ("fp8", "fp4")
        self.wq4 = dtype == "fp4"
        # fp4 + batch pools: the skinny MFMA GEMM consumes fp8, so keep
        # the fp8 copy only then; max_batch=1 fp4 engines are SINGLE-COPY
        self.keep_fp8 = (dtype == "fp8") or (self.wq4 and max_batch > 1)
        if kv_dtype not in ("bf16", "fp8"):
            raise ValueError(f"unsupported kv_dtype {kv_dtype!r}")
        self.kv_dtype = kv_dtype
        self.kv8 = kv_dtype == "fp8"  # e4m3 KV pool, per-(head,pos) scales
        # Mixtral sparse MoE: per-token top-k routed experts (decode:
        # expert-indexed GEMVs off device idx; prefill: dense per-expert
        # GEMM loop weighted by router probs)
        self.moe = config.is_moe
        self.topk = config.num_experts_per_tok
        if self.moe:
            if self.wq4:
                raise ValueError("MoE + fp4 weights not supported yet")
            if config.model_type == "gemma2":
                raise ValueError("MoE is a llama-family (Mixtral) feature")
        self.rank, self.world = tpu.init_distributed()
        # exercise the TP code path on 1 GPU (collectives no-op at
        # world=1, partial sums are then exact) — used by tests
        self.tp_branch = self.world > 1 or force_tp_path
        if device is None:
            device = f"cuda:{self.rank % max(torch.cuda.device_count(), 1)}"
        self.device = torch.device(device)
        torch.cuda.set_device(self.device)

        cfg = config
        tp = self.world
        assert cfg.num_attention_heads % tp == 0, "nh must divide tp"
        assert cfg.num_key_value_heads % tp == 0, \
            f"kv heads {cfg.num_key_value_heads} not divisible by tp {tp}"
        assert cfg.intermediate_size % tp == 0
        assert cfg.vocab_size % tp == 0
        if self.moe and tp > 1:
            raise ValueError("MoE tensor parallelism (expert parallel) "
                             "is not implemented yet — run TP=1")
        self.nh_l = cfg.num_attention_heads // tp
        self.kvh_l = cfg.num_key_value_heads // tp
        self.inter_l = cfg.intermediate_size // tp
        self.vocab_l = cfg.vocab_size // tp
        self.hd = cfg.head_dim
        self.H = cfg.hidden_size

        # one-shot xGMI collectives for decode-sized payloads (validated
        # on hardware at init; silent RCCL fallback when unavailable)
        if self.world > 1:
            slot_bytes = max(self.H * 2, self.vocab_l * 4, 1 << 16)
            tpu.init_xgmi(self.device, slot_bytes)
        # prefill-sized all-reduces: RCCL async on its own stream,
        # overlapped with the next row-chunk's GEMM (north star: comm
        # overlap on a second HIP stream).  Needs the nccl(=RCCL)
        # backend; gloo (CPU CI) takes the inline path.
        import torch.distributed as dist
        self._async_ar = bool(
            self.world > 1 and dist.is_initialized()
            and "nccl" in str(dist.get_backend()).lower())

        self._upload_weights(weights)
        if max_seq is None:
            max_seq = self._auto_max_seq()
        self.max_seq = max_seq
        self._alloc_state(prefill_chunk)
        self._graph = None
        self._graph_mode = None
        # side-stream weight prefetcher — measured NEGATIVE (the fork's
        # join makes the sequential prefetch chain the critical path:
        # fp8 1391 -> 764 tok/s), so off by default; LLM_PREFETCH=1 to
        # experiment
        import os as _os
        self.prefetch_on = _os.environ.get("LLM_PREFETCH", "0") == "1"
        self._pf_stream = torch.cuda.Stream(device=self.device)
        self._pf_sink = torch.zeros(256, dtype=torch.float32,
                                    device=self.device)
        self._pf_events = []

    # ------------------------------------------------------------------
    def _auto_max_seq(self) -> int:
        """KV pool length from free HBM (post-weights): fill the 288 GB
        MI355X budget instead of a fixed 4k default,
        capped at the model's max_position_embeddings."""
        cfg = self.config
        free, _total = torch.cuda.mem_get_info(self.device)
        esz = 1 if self.kv8 else 2
        per_pos = (cfg.num_hidden_layers * self.kvh_l * self.hd * 2 * esz
                   * self.max_batch)
        if self.kv8:
            per_pos += (cfg.num_hidden_layers * self.kvh_l * 2 * 4
                        * self.max_batch)  # scales
        per_pos += self.hd * 4 + 8  # rope tables + ring slot
        budget = int(free * 0.9) - (2 << 30)  # headroom for scratch/graph
        n = max(1024, budget // max(per_pos, 1))
        return int(min(cfg.max_position_embeddings, n))

    def _upload_weights(self, w: Dict[str, np.ndarray]):
        from ..io.loader import validate_weights

        validate_weights(self.config, w)  # actionable error vs KeyError
        cfg, dev = self.config, self.device
        tp, r = self.world, self.rank
        gemma = cfg.model_type == "gemma2"

        def bf16(a: np.ndarray) -> torch.Tensor:
            return torch.from_numpy(np.ascontiguousarray(a)).to(
                dev, torch.bfloat16, non_blocking=False)

        def gamma(a: np.ndarray) -> torch.Tensor:
            g = a.astype(np.float32) + (1.0 if gemma else 0.0)
            return torch.from_numpy(g).to(dev)

        def quant4(a: np.ndarray):
            """MXFP4: e2m1 packed 2/byte + e8m0 scale per 32 elems."""
            t = bf16(a)
            N, K = t.shape
            q = torch.empty(N, K // 2, dtype=torch.uint8, device=dev)
            e = torch.empty(N, K // 32, dtype=torch.uint8, device=dev)
            ho.quant_fp4(t, q, e)
            torch.cuda.synchronize()
            return q, e

        def quant8(a: np.ndarray):
            """Per-output-row e4m3fn quantization ON DEVICE (k_quant_
            fp8_rows): one bf16 upload, quantize, drop the bf16 copy —
            fp8 models hold a single weight copy in HBM and load in
            seconds (round-1 host quant took ~90 s for 9B)."""
            t = bf16(a)
            N, K = t.shape
            q = torch.empty(N, K, dtype=torch.uint8, device=dev)
            s = torch.empty(N, dtype=torch.float32, device=dev)
            ho.quant_fp8(t, q, s)
            torch.cuda.synchronize()  # t freed right after
            return q, s

        self.embed = bf16(w["model.embed_tokens.weight"])
        lm_w = w.get("lm_head.weight", w["model.embed_tokens.weight"])
        if self.fp8:
            lmw = lm_w if tp == 1 else tpu.shard_rows(lm_w, r, tp)
            if self.keep_fp8:
                self.lm_head_q, self.lm_head_s = quant8(lmw)
            self.lm_head = None  # single-copy: no bf16 duplicate
            if self.wq4:
                self.lm_head_q4, self.lm_head_e4 = quant4(lmw)
        elif tp > 1:
            self.lm_head = bf16(tpu.shard_rows(lm_w, r, tp))
        elif lm_w is w.get("model.embed_tokens.weight", None) or \
                cfg.tie_word_embeddings and "lm_head.weight" not in w:
            self.lm_head = self.embed  # tied, share device memory
        else:
            self.lm_head = bf16(lm_w)
        self.g_final = gamma(w["model.norm.weight"])

        hd = cfg.head_dim
        self.layers = []
        for i in range(cfg.num_hidden_layers):
            p = f"model.layers.{i}"
            a = f"{p}.self_attn"
            hrows = lambda x: tpu.shard_rows(x, r, tp) if tp > 1 else x
            hcols = lambda x: tpu.shard_cols(x, r, tp) if tp > 1 else x
            import numpy as _np
            qkv_np = _np.concatenate([
                hrows(w[f"{a}.q_proj.weight"]),
                hrows(w[f"{a}.k_proj.weight"]),
                hrows(w[f"{a}.v_proj.weight"])], axis=0)
            if self.moe:
                # stacked per-expert [gate; up] rows (E*2I, H) and down
                # rows (E*H, I) — ONE device tensor per projection so the
                # decode GEMV can index expert e as a row offset
                # (eidx * 2I resp. eidx * H) off a device int32
                m = f"{p}.block_sparse_moe"
                E = cfg.num_local_experts
                gu_np = _np.concatenate(
                    [x for e in range(E)
                     for x in (w[f"{m}.experts.{e}.w1.weight"],
                               w[f"{m}.experts.{e}.w3.weight"])], axis=0)
                down_np = _np.concatenate(
                    [w[f"{m}.experts.{e}.w2.weight"] for e in range(E)],
                    axis=0)
            else:
                gu_np = _np.concatenate([
                    hrows(w[f"{p}.mlp.gate_proj.weight"]),
                    hrows(w[f"{p}.mlp.up_proj.weight"])], axis=0)
                down_np = hcols(w[f"{p}.mlp.down_proj.weight"])
            o_np = hcols(w[f"{a}.o_proj.weight"])
            nq = self.nh_l * hd
            nkv = self.kvh_l * hd
            I = self.inter_l
            lw = {
                "g_in": gamma(w[f"{p}.input_layernorm.weight"]),
                "g_post": gamma(w[f"{p}.post_attention_layernorm.weight"]),
            }
            if self.fp8:
                # SINGLE fp8 copy: decode GEMV and prefill fp8-MFMA GEMM
                # both read these (no bf16 duplicate in HBM)
                for name, arr in [("wqkv", qkv_np), ("wgu", gu_np),
                                  ("wo", o_np), ("wdown", down_np)]:
                    if self.keep_fp8:
                        lw[name + "_q"], lw[name + "_s"] = quant8(arr)
                    if self.wq4:
                        lw[name + "_q4"], lw[name + "_e4"] = quant4(arr)
                if self.keep_fp8:
                    lw.update({
                        "wq_q": lw["wqkv_q"][:nq],
                        "wq_s": lw["wqkv_s"][:nq],
                        "wk_q": lw["wqkv_q"][nq:nq + nkv],
                        "wk_s": lw["wqkv_s"][nq:nq + nkv],
                        "wv_q": lw["wqkv_q"][nq + nkv:],
                        "wv_s": lw["wqkv_s"][nq + nkv:],
                        "wgate_q": lw["wgu_q"][:I],
                        "wgate_s": lw["wgu_s"][:I],
                        "wup_q": lw["wgu_q"][I:], "wup_s": lw["wgu_s"][I:],
                    })
                if self.wq4:
                    lw.update({
                        "wq_q4": lw["wqkv_q4"][:nq],
                        "wq_e4": lw["wqkv_e4"][:nq],
                        "wk_q4": lw["wqkv_q4"][nq:nq + nkv],
                        "wk_e4": lw["wqkv_e4"][nq:nq + nkv],
                        "wv_q4": lw["wqkv_q4"][nq + nkv:],
                        "wv_e4": lw["wqkv_e4"][nq + nkv:],
                        "wgate_q4": lw["wgu_q4"][:I],
                        "wgate_e4": lw["wgu_e4"][:I],
                        "wup_q4": lw["wgu_q4"][I:],
                        "wup_e4": lw["wgu_e4"][I:],
                    })
            else:
                wqkv = bf16(qkv_np)
                wgu = bf16(gu_np)
                lw.update({
                    # fused tensors (decode path) + contiguous row views
                    "wqkv": wqkv,
                    "wq": wqkv[:nq], "wk": wqkv[nq:nq + nkv],
                    "wv": wqkv[nq + nkv:],
                    "wgu": wgu, "wgate": wgu[:I], "wup": wgu[I:],
                    "wo": bf16(o_np),
                    "wdown": bf16(down_np),
                })
            if self.moe:
                # router weights + decode-shaped views: the GEMV sees the
                # FIRST expert's rows (N=2I / N=H) and offsets the base
                # pointer by eidx * wstride device-side
                lw["wg"] = bf16(w[f"{p}.block_sparse_moe.gate.weight"])
                I2, H2 = 2 * self.inter_l, self.H
                if self.fp8:
                    lw["wgu_q_all"] = lw.pop("wgu_q")
                    lw["wgu_s_all"] = lw.pop("wgu_s")
                    lw["wdown_q_all"] = lw.pop("wdown_q")
                    lw["wdown_s_all"] = lw.pop("wdown_s")
                    lw["wgu_q"] = lw["wgu_q_all"][:I2]
                    lw["wgu_s"] = lw["wgu_s_all"][:I2]
                    lw["wdown_q"] = lw["wdown_q_all"][:H2]
                    lw["wdown_s"] = lw["wdown_s_all"][:H2]
                else:
                    lw["wgu_all"] = lw.pop("wgu")
                    lw["wdown_all"] = lw.pop("wdown")
                    lw["wgu"] = lw["wgu_all"][:I2]
                    lw["wdown"] = lw["wdown_all"][:H2]
            if gemma:
                lw["g_preffn"] = gamma(w[f"{p}.pre_feedforward_layernorm.weight"])
                lw["g_postffn"] = gamma(w[f"{p}.post_feedforward_layernorm.weight"])
            if cfg.attention_bias:
                # Qwen-2 family: fused qkv bias (decode GEMV adds it via
                # the res slot; prefill adds it after the GEMM)
                bias_np = _np.concatenate([
                    hrows(w[f"{a}.q_proj.bias"]),
                    hrows(w[f"{a}.k_proj.bias"]),
                    hrows(w[f"{a}.v_proj.bias"])], axis=0)
                bqkv = bf16(bias_np.reshape(1, -1))[0]
                lw["bqkv"] = bqkv
                lw["bias_q"] = bqkv[:nq]
                lw["bias_k"] = bqkv[nq:nq + nkv]
                lw["bias_v"] = bqkv[nq + nkv:]
            self.layers.append(lw)

    def _alloc_state(self, prefill_chunk: int):
        cfg, dev = self.config, self.device
        S, hd = self.max_seq, self.hd
        L = cfg.num_hidden_layers
        bf = dict(dtype=torch.bfloat16, device=dev)
        kvt = torch.uint8 if self.kv8 else torch.bfloat16
        B = self.max_batch
        kv_shape = (self.kvh_l, S, hd) if B == 1 else (B, self.kvh_l, S, hd)
        self.k_cache = [torch.zeros(*kv_shape, dtype=kvt, device=dev)
                        for _ in range(L)]
        self.v_cache = [torch.zeros(*kv_shape, dtype=kvt, device=dev)
                        for _ in range(L)]
        if self.kv8:
            ks = (self.kvh_l, S) if B == 1 else (B, self.kvh_l, S)
            self.k_scale = [torch.zeros(*ks, dtype=torch.float32,
                                        device=dev) for _ in range(L)]
            self.v_scale = [torch.zeros(*ks, dtype=torch.float32,
                                        device=dev) for _ in range(L)]
        else:
            self.k_scale = [None] * L
            self.v_scale = [None] * L

        # RoPE tables fp32 (host-precomputed: guide App.B — no device trig)
        inv = cfg.rope_inv_freq()                      # (hd/2,) fp64
        t = np.arange(S, dtype=np.float64)
        freqs = np.outer(t, inv)
        self.cos_t = torch.from_numpy(np.cos(freqs).astype(np.float32)).to(dev)
        self.sin_t = torch.from_numpy(np.sin(freqs).astype(np.float32)).to(dev)

        PC = self.PC = min(prefill_chunk, max(256, self.max_seq))
        H, I = self.H, self.inter_l
        self.b_h = torch.zeros(PC, H, **bf)
        self.b_xn = torch.zeros(PC, H, **bf)
        self.b_t1 = torch.zeros(PC, H, **bf)
        self.b_q = torch.zeros(PC, self.nh_l * hd, **bf)
        self.b_k = torch.zeros(PC, self.kvh_l * hd, **bf)
        self.b_v = torch.zeros(PC, self.kvh_l * hd, **bf)
        self.b_att = torch.zeros(PC, self.nh_l * hd, **bf)
        self.b_gate = torch.zeros(PC, I, **bf)
        self.b_up = torch.zeros(PC, I, **bf)
        self.b_gemm_acc = torch.zeros(PC * max(I, H), dtype=torch.float32,
                                      device=dev)
        if self.fp8:
            # activation-quant scratch for the fp8 MFMA prefill GEMMs
            kmax = max(H, I, self.nh_l * hd)
            self.b_xq = torch.zeros(PC * kmax, dtype=torch.uint8, device=dev)
            self.b_sx = torch.zeros(PC, dtype=torch.float32, device=dev)
        self.b_logits_l = torch.zeros(self.vocab_l, dtype=torch.float32,
                                      device=dev)
        self.b_logits = (self.b_logits_l if not self.tp_branch else
                         torch.zeros(cfg.vocab_size, dtype=torch.float32,
                                     device=dev))
        self.b_qkv = torch.zeros((self.nh_l + 2 * self.kvh_l) * hd, **bf)
        if self.moe:
            E = cfg.num_local_experts
            self.moe_idx = torch.zeros(PC * self.topk, dtype=torch.int32,
                                       device=dev)
            self.moe_w = torch.zeros(PC * self.topk, dtype=torch.float32,
                                     device=dev)
            self.moe_dense = torch.zeros(PC * E, dtype=torch.float32,
                                         device=dev)
        self.b_hb = torch.zeros(H, **bf)   # gemma residual ping-pong
        self.b_t2 = torch.zeros(H, **bf)   # gemma ffn-out delta
        self.b_gu = torch.zeros(2 * I, **bf)
        # split-T decode attention scratch (partials + per-head tickets)
        import os as _os
        _sp = _os.environ.get("LLM_ATTN_SPLIT")
        # measured (fp8 llama-1b): T~100 split2 1493 > split4 1483;
        # T~4k split8 1108 > split16 1050; T~8k split16 963 > split32
        self.attn_split = (int(_sp) if _sp else
                           min(16, max(2, self.max_seq // 512)))
        self.attn_scratch = torch.zeros(
            self.nh_l * self.attn_split * (hd + 2), dtype=torch.float32,
            device=dev)
        self.attn_cnt = torch.zeros(self.nh_l, dtype=torch.int32, device=dev)
        self.s_gmax = torch.zeros(1, dtype=torch.int64, device=dev)
        self.s_pick = torch.zeros(1, dtype=torch.int64, device=dev)
        self.s_cnt = torch.zeros(1, dtype=torch.int32, device=dev)
        i32 = dict(dtype=torch.int32, device=dev)
        self.ids_buf = torch.zeros(PC, **i32)
        self.next_token = torch.zeros(1, **i32)
        self.out_ring = torch.zeros(max(65536, S + 16), **i32)
        self.nout = torch.zeros(1, **i32)
        self.len_buf = torch.zeros(1, **i32)
        self.rng_ctr = torch.zeros(1, dtype=torch.int64, device=dev)

        if self.max_batch > 1:
            B, nhh = self.max_batch, self.nh_l * hd
            i64 = dict(dtype=torch.int64, device=dev)
            self.bt_qkv = torch.zeros(B, (self.nh_l + 2 * self.kvh_l) * hd,
                                      **bf)
            self.bt_logits = torch.zeros(B, self.vocab_l, **bf)
            self.bt_logits32 = torch.zeros(B, cfg.vocab_size,
                                           dtype=torch.float32, device=dev)
            self.bt_next = torch.zeros(B, **i32)
            self.bt_lens = torch.zeros(B, **i32)  # ragged positions
            self.bt_ring = torch.zeros(B, S + 16, **i32)
            self.bt_nout = torch.zeros(B, **i32)
            self.bt_gmax = torch.zeros(B, **i64)
            self.bt_pick = torch.zeros(B, **i64)
            self.bt_cnt = torch.zeros(B, **i32)
            self.bt_scratch = torch.zeros(
                B * self.nh_l * self.attn_split * (hd + 2),
                dtype=torch.float32, device=dev)
            self.bt_attn_cnt = torch.zeros(B * self.nh_l, **i32)
            # fused multi-x path scratch (fp8, B<=8)
            self.bt_gu = torch.zeros(B, 2 * I, **bf)
            self.bt_t1 = torch.zeros(B, H, **bf)
            self.bt_t2 = torch.zeros(B, H, **bf)
            self.bt_hb = torch.zeros(B, H, **bf)

        self.act = 0 if cfg.hidden_act == "silu" else 1
        self.gemma = cfg.model_type == "gemma2"
        self.attn_softcap = float(cfg.attn_logit_softcapping or 0.0)
        self.final_softcap = float(cfg.final_logit_softcapping or 0.0)
        self.scale = float(cfg.attn_scale)

    # ------------------------------------------------------------------
    # cache interface (generate()-compatible)
    # ------------------------------------------------------------------
    def make_cache(self, n: int) -> DeviceCacheHandle:
        if n > self.max_seq:
            raise ValueError(f"requested cache {n} > max_seq {self.max_seq}")
        self.reset()
        return DeviceCacheHandle(self)

    def reset(self):
        self.len_buf.zero_()
        self.nout.zero_()
        self.rng_ctr.zero_()
        self.s_gmax.zero_()
        self.s_pick.zero_()
        self.s_cnt.zero_()
        self._host_len = 0  # host mirror of len_buf (overflow guard)

    # ------------------------------------------------------------------
    # layer stack over rows [0, M) of the scratch buffers
    # ------------------------------------------------------------------
    def _linear(self, lw, name, x, y, res=None, M: int = 1,
                softcap: float = 0.0):
        """Projection by weight NAME so dtype routing stays in one
        place: bf16 GEMV/MFMA-GEMM, or (fp8) GEMV / fp8-MFMA GEMM with
        on-the-fly activation row quantization."""
        if M == 1:
            if self.wq4:
                ho.gemv_fp4(lw[name + "_q4"], lw[name + "_e4"], x, y,
                            res=res, softcap=softcap)
            elif self.fp8:
                ho.gemv_fp8(lw[name + "_q"], lw[name + "_s"], x, y,
                            res=res, softcap=softcap)
            else:
                ho.gemv(lw[name], x, y, res=res, softcap=softcap)
        elif self.wq4:
            ho.gemm_fp4w(x[:M], lw[name + "_q4"], lw[name + "_e4"], y[:M],
                         res=res[:M] if res is not None else None,
                         accbuf=self.b_gemm_acc)
        elif self.fp8:
            K = x.shape[-1]
            ho.quant_fp8(x[:M], self.b_xq, self.b_sx)
            ho.gemm_fp8(self.b_xq, self.b_sx, lw[name + "_q"],
                        lw[name + "_s"], y[:M], M, K,
                        res=res[:M] if res is not None else None,
                        accbuf=self.b_gemm_acc)
        else:
            ho.gemm(x[:M], lw[name], y, res=res[:M] if res is not None else None,
                    accbuf=self.b_gemm_acc)

    def _rowpar_ar(self, lw, name, x, y, M: int):
        """Row-parallel projection followed by all-reduce over TP ranks.
        For prefill-sized M the rows are split in two: the all-reduce of
        chunk A runs on RCCL's comm stream while chunk B's GEMM occupies
        the compute stream (SURVEY §2.3 'overlap comm on 2nd stream' —
        the reference has no distributed code at all)."""
        if not (self._async_ar and M >= 512):
            self._linear(lw, name, x, y, M=M)
            tpu.all_reduce(y[:M])
            return
        import torch.distributed as dist
        K = x.shape[-1]
        if self.fp8 and not self.wq4:
            ho.quant_fp8(x[:M], self.b_xq, self.b_sx)
        works = []
        M2 = M // 2
        for a, b in ((0, M2), (M2, M)):
            if self.wq4:
                ho.gemm_fp4w(x[a:b], lw[name + "_q4"], lw[name + "_e4"],
                             y[a:b], accbuf=self.b_gemm_acc)
            elif self.fp8:
                ho.gemm_fp8(self.b_xq[a * K:], self.b_sx[a:],
                            lw[name + "_q"], lw[name + "_s"], y[a:b],
                            b - a, K, accbuf=self.b_gemm_acc)
            else:
                ho.gemm(x[a:b], lw[name], y[a:b], accbuf=self.b_gemm_acc)
            # async -> lands on the RCCL stream after the chunk's GEMM;
            # the compute stream proceeds to the next chunk immediately
            works.append(dist.all_reduce(y[a:b], async_op=True))
        for wk in works:
            wk.wait()  # compute stream waits on the comm stream

    def _kc(self, i):
        return self.k_cache[i] if self.max_batch == 1 else \
            self.k_cache[i][self._pb]

    def _vc(self, i):
        return self.v_cache[i] if self.max_batch == 1 else \
            self.v_cache[i][self._pb]

    def _ks(self, i):
        s = self.k_scale[i]
        return s if (s is None or self.max_batch == 1) else s[self._pb]

    def _vs(self, i):
        s = self.v_scale[i]
        return s if (s is None or self.max_batch == 1) else s[self._pb]

    def _layers_forward(self, M: int, layer_hook=None,
                        batch_attn: bool = False):
        """Layer stack over rows [0, M).  ``batch_attn``: rows are M
        INDEPENDENT sequences at the same position (lockstep batched
        decode) — the fused decode-attention kernel runs per row over
        per-sequence KV pools instead of the prefill attention."""
        cfg = self.config
        eps = cfg.rms_norm_eps
        h, xn, t1 = self.b_h, self.b_xn, self.b_t1
        if layer_hook is not None:
            layer_hook(-1, h[:M])  # embedding output
        for i, lw in enumerate(self.layers):
            window = cfg.sliding_window if cfg.is_sliding(i) else 0
            ho.rmsnorm(h[:M], lw["g_in"], xn[:M], eps=eps)
            if batch_attn:
                self._linear(lw, "wqkv", xn, self.bt_qkv, M=M)
                if "bqkv" in lw:
                    ho.bias_add(self.bt_qkv, lw["bqkv"], M)
                ho.attn_dec(self.bt_qkv, self.k_cache[i], self.v_cache[i],
                            self.b_att[0], self.bt_lens, self.cos_t,
                            self.sin_t, self.bt_scratch, self.bt_attn_cnt,
                            self.nh_l, self.kvh_l, self.hd, self.scale,
                            softcap=self.attn_softcap, window=window or 0,
                            split=self.attn_split, kS=self.k_scale[i],
                            vS=self.v_scale[i], batch=M)
            else:
                self._linear(lw, "wq", xn, self.b_q, M=M)
                self._linear(lw, "wk", xn, self.b_k, M=M)
                self._linear(lw, "wv", xn, self.b_v, M=M)
                if "bqkv" in lw:
                    ho.bias_add(self.b_q, lw["bias_q"], M)
                    ho.bias_add(self.b_k, lw["bias_k"], M)
                    ho.bias_add(self.b_v, lw["bias_v"], M)
                ho.rope_cache(self.b_q, self.b_k, self.b_v, self._kc(i),
                              self._vc(i), self.cos_t, self.sin_t,
                              self.len_buf, M, self.nh_l, self.kvh_l,
                              self.hd, kS=self._ks(i), vS=self._vs(i))
                if self.hd in (64, 128, 256):
                    ho.attn_prefill_mfma(
                        self.b_q, self._kc(i), self._vc(i), self.b_att,
                        self.len_buf, M, self.nh_l, self.kvh_l, self.hd,
                        self.scale, softcap=self.attn_softcap,
                        window=window or 0,
                        kS=self._ks(i), vS=self._vs(i))
                else:
                    # per-query VALU kernel: every production head_dim is
                    # 64/128/256 (MFMA above); kept as the documented
                    # debugging/odd-shape fallback
                    ho.attn(self.b_q, self._kc(i), self._vc(i),
                            self.b_att, self.len_buf, M, self.nh_l,
                            self.kvh_l, self.hd, self.scale,
                            softcap=self.attn_softcap,
                            window=window or 0,
                            kS=self._ks(i), vS=self._vs(i))
            if self.gemma:
                self._rowpar_ar(lw, "wo", self.b_att, t1, M)
                ho.rmsnorm(t1[:M], lw["g_post"], h[:M], res=h[:M], eps=eps)
                ho.rmsnorm(h[:M], lw["g_preffn"], xn[:M], eps=eps)
            else:
                if self.tp_branch:
                    self._rowpar_ar(lw, "wo", self.b_att, t1, M)
                    ho.addinto(h[:M], t1[:M])
                else:
                    self._linear(lw, "wo", self.b_att, h, res=h, M=M)
                ho.rmsnorm(h[:M], lw["g_post"], xn[:M], eps=eps)
            if self.moe:
                self._moe_mlp_rows(lw, M)
                if layer_hook is not None:
                    layer_hook(i, h[:M])
                continue
            self._linear(lw, "wgate", xn, self.b_gate, M=M)
            self._linear(lw, "wup", xn, self.b_up, M=M)
            ho.glu(self.b_gate[:M], self.b_up[:M], self.b_gate[:M], self.act)
            if self.gemma:
                self._rowpar_ar(lw, "wdown", self.b_gate, t1, M)
                ho.rmsnorm(t1[:M], lw["g_postffn"], h[:M], res=h[:M], eps=eps)
            else:
                if self.tp_branch:
                    self._rowpar_ar(lw, "wdown", self.b_gate, t1, M)
                    ho.addinto(h[:M], t1[:M])
                else:
                    self._linear(lw, "wdown", self.b_gate, h, res=h, M=M)
            if layer_hook is not None:
                layer_hook(i, h[:M])

    def _moe_mlp_rows(self, lw, M: int):
        """Prefill/batch MoE MLP over rows [0, M): router weights per
        row (dense M x E grid), then ONE GEMM pass per expert with the
        output folded into h weighted by that row's prob (0 if
        unrouted).  Work is E/topk x the routed minimum — prefill is
        compute-rich and this keeps the MFMA GEMM path; decode uses the
        expert-indexed GEMVs instead (`_decode_step`)."""
        cfg = self.config
        E, I, H = cfg.num_local_experts, self.inter_l, self.H
        eps = cfg.rms_norm_eps
        xn, gate, up, t1 = self.b_xn, self.b_gate, self.b_up, self.b_t1
        ho.moe_route(self.b_h, lw["g_post"], lw["wg"], M, self.topk,
                     self.moe_idx, self.moe_w, dense=self.moe_dense,
                     eps=eps)
        for e in range(E):
            g_lo, g_hi = e * 2 * I, e * 2 * I + I
            u_hi = (e + 1) * 2 * I
            d_lo, d_hi = e * H, (e + 1) * H
            if self.fp8:
                ho.quant_fp8(xn[:M], self.b_xq, self.b_sx)
                ho.gemm_fp8(self.b_xq, self.b_sx,
                            lw["wgu_q_all"][g_lo:g_hi],
                            lw["wgu_s_all"][g_lo:g_hi], gate[:M], M, H,
                            accbuf=self.b_gemm_acc)
                ho.gemm_fp8(self.b_xq, self.b_sx,
                            lw["wgu_q_all"][g_hi:u_hi],
                            lw["wgu_s_all"][g_hi:u_hi], up[:M], M, H,
                            accbuf=self.b_gemm_acc)
                ho.glu(gate[:M], up[:M], gate[:M], self.act)
                ho.quant_fp8(gate[:M], self.b_xq, self.b_sx)
                ho.gemm_fp8(self.b_xq, self.b_sx,
                            lw["wdown_q_all"][d_lo:d_hi],
                            lw["wdown_s_all"][d_lo:d_hi], t1[:M], M, I,
                            accbuf=self.b_gemm_acc)
            else:
                ho.gemm(xn[:M], lw["wgu_all"][g_lo:g_hi], gate[:M],
                        accbuf=self.b_gemm_acc)
                ho.gemm(xn[:M], lw["wgu_all"][g_hi:u_hi], up[:M],
                        accbuf=self.b_gemm_acc)
                ho.glu(gate[:M], up[:M], gate[:M], self.act)
                ho.gemm(gate[:M], lw["wdown_all"][d_lo:d_hi], t1[:M],
                        accbuf=self.b_gemm_acc)
            ho.moe_scale_add(self.b_h, t1, self.moe_dense[e:], E, M, H)

    def _lm_head_last(self, M: int):
        """Final norm + lm_head on the last row -> self.b_logits (f32, V);
        the norm rides in the GEMV staging pass (no separate launch)."""
        hrow = self.b_h[M - 1] if M > 1 else self.b_h[0]
        kw = dict(stage=ho.STAGE_NORM, g=self.g_final,
                  eps=self.config.rms_norm_eps, softcap=self.final_softcap)
        if self.wq4:
            ho.gemv_fp4(self.lm_head_q4, self.lm_head_e4, hrow,
                        self.b_logits_l, **kw)
        elif self.fp8:
            ho.gemv_fp8(self.lm_head_q, self.lm_head_s, hrow,
                        self.b_logits_l, **kw)
        else:
            ho.gemv(self.lm_head, hrow, self.b_logits_l, **kw)
        if self.tp_branch:
            tpu.all_gather_into(self.b_logits, self.b_logits_l)

    def forward_full(self, ids: np.ndarray,
                     return_hidden_states: bool = False):
        """All-positions logits (M, V) — reference-parity API (the HF
        tuple shape).  Prefill-style pass; vocab GEMM per
        chunk.  Resets the cache.  With ``return_hidden_states`` also
        returns the L+1 per-layer hidden states (embedding output +
        each decoder layer, reference ``all_hidden_states``
        llama3.2_model.py:682-716) as fp32 numpy arrays."""
        ids = np.asarray(ids, dtype=np.int32).ravel()
        self.reset()
        if not return_hidden_states:
            return self.forward_positions(ids, 0)
        n = len(ids)
        out = np.empty((n, self.config.vocab_size), dtype=np.float32)
        L = self.config.num_hidden_layers
        hidden = [np.empty((n, self.H), dtype=np.float32)
                  for _ in range(L + 1)]
        logits_buf = torch.empty(self.PC, self.vocab_l,
                                 dtype=torch.bfloat16, device=self.device)
        done = 0
        while done < n:
            M = min(n - done, self.PC)
            self.ids_buf[:M].copy_(
                torch.from_numpy(ids[done:done + M].astype(np.int32)))
            ho.i32_set(self.len_buf, done)
            ho.embed(self.embed, self.ids_buf, self.b_h, M,
                     self.config.embed_scale)
            d0 = done

            def hook(i, h):
                torch.cuda.synchronize()
                hidden[i + 1][d0:d0 + h.shape[0]] = \
                    h.float().cpu().numpy()
            self._layers_forward(M, layer_hook=hook)
            ho.rmsnorm(self.b_h[:M], self.g_final, self.b_xn[:M],
                       eps=self.config.rms_norm_eps)
            if self.wq4:
                ho.gemm_fp4w(self.b_xn[:M], self.lm_head_q4,
                             self.lm_head_e4, logits_buf[:M],
                             accbuf=self.b_gemm_acc)
            elif self.fp8:
                ho.quant_fp8(self.b_xn[:M], self.b_xq, self.b_sx)
                ho.gemm_fp8(self.b_xq, self.b_sx, self.lm_head_q,
                            self.lm_head_s, logits_buf[:M], M, self.H,
                            accbuf=self.b_gemm_acc)
            else:
                ho.gemm(self.b_xn[:M], self.lm_head, logits_buf[:M],
                        accbuf=self.b_gemm_acc)
            if self.final_softcap:
                ho.softcap(logits_buf[:M], self.final_softcap)
            if self.world > 1:
                # gather vocab shards on-device (each rank holds the
                # columns [r*vocab_l, (r+1)*vocab_l) of rows [0, M))
                import torch.distributed as dist
                t = logits_buf[:M].float().contiguous()
                chunks = [torch.empty_like(t) for _ in range(self.world)]
                dist.all_gather(chunks, t)
                loc = torch.cat(chunks, dim=1).cpu().numpy()
            else:
                torch.cuda.synchronize()
                loc = logits_buf[:M].float().cpu().numpy()
            out[done:done + M] = loc
            done += M
        ho.i32_set(self.len_buf, n)
        # mirror the device len so a following decode()'s overflow guard
        # counts from the right base (KV-pool OOB otherwise)
        self._host_len = n
        if return_hidden_states:
            return out, hidden
        return out

    def forward_positions(self, ids: np.ndarray, pos0: int) -> np.ndarray:
        """All-position logits (M, V) fp32, CONTINUING from pos0 (no
        cache reset: KV for ids lands at [pos0, pos0+M) and the pool can
        be rolled back afterwards with ``rewind``).  This is the
        speculative-decoding VERIFY pass: the target model scores k
        draft tokens in one prefill-shaped pass instead of k decode
        steps (ROADMAP §5; the reference has nothing comparable)."""
        ids = np.asarray(ids, dtype=np.int32).ravel()
        n = len(ids)
        if n == 0:
            raise ValueError("empty ids")
        if pos0 + n > self.max_seq:
            raise ValueError(f"sequence {pos0}+{n} exceeds max_seq "
                             f"{self.max_seq}")
        out = np.empty((n, self.config.vocab_size), dtype=np.float32)
        logits_buf = torch.empty(self.PC, self.vocab_l,
                                 dtype=torch.bfloat16, device=self.device)
        done = 0
        while done < n:
            M = min(n - done, self.PC)
            self.ids_buf[:M].copy_(
                torch.from_numpy(ids[done:done + M].astype(np.int32)))
            ho.i32_set(self.len_buf, pos0 + done)
            ho.embed(self.embed, self.ids_buf, self.b_h, M,
                     self.config.embed_scale)
            self._layers_forward(M)
            ho.rmsnorm(self.b_h[:M], self.g_final, self.b_xn[:M],
                       eps=self.config.rms_norm_eps)
            if self.wq4:
                ho.gemm_fp4w(self.b_xn[:M], self.lm_head_q4,
                             self.lm_head_e4, logits_buf[:M],
                             accbuf=self.b_gemm_acc)
            elif self.fp8:
                ho.quant_fp8(self.b_xn[:M], self.b_xq, self.b_sx)
                ho.gemm_fp8(self.b_xq, self.b_sx, self.lm_head_q,
                            self.lm_head_s, logits_buf[:M], M, self.H,
                            accbuf=self.b_gemm_acc)
            else:
                ho.gemm(self.b_xn[:M], self.lm_head, logits_buf[:M],
                        accbuf=self.b_gemm_acc)
            if self.final_softcap:
                ho.softcap(logits_buf[:M], self.final_softcap)
            if self.world > 1:
                # gather vocab shards on-device (each rank holds the
                # columns [r*vocab_l, (r+1)*vocab_l) of rows [0, M))
                import torch.distributed as dist
                t = logits_buf[:M].float().contiguous()
                chunks = [torch.empty_like(t) for _ in range(self.world)]
                dist.all_gather(chunks, t)
                loc = torch.cat(chunks, dim=1).cpu().numpy()
            else:
                torch.cuda.synchronize()
                loc = logits_buf[:M].float().cpu().numpy()
            out[done:done + M] = loc
            done += M
        ho.i32_set(self.len_buf, pos0 + n)
        self._host_len = pos0 + n
        return out

    def rewind(self, n: int):
        """Roll the sequence back to length n — O(1): the KV pool is
        preallocated and the attention kernels read the device length
        from len_buf, so positions >= n are dead until overwritten.
        Speculative decoding uses this to discard rejected draft KV."""
        if not 0 <= n <= self.max_seq:
            raise ValueError(f"rewind {n} outside [0, {self.max_seq}]")
        ho.i32_set(self.len_buf, n)
        self._host_len = n

    def forward_hf(self, ids: np.ndarray):
        """Reference-parity output tuple ``(loss, logits, kv_cache,
        hidden_states, attentions)`` (llama3.2_model.py:726-822).
        ``loss`` is None (inference-only, as in the reference) and
        ``attentions`` is None by design: the fused attention kernels
        use online softmax and never materialize the probability
        matrix."""
        logits, hidden = self.forward_full(ids, return_hidden_states=True)
        cache = DeviceCacheHandle(self)
        cache.seq_len = len(np.ravel(ids))
        return None, logits, cache, hidden, None

    def generate_tokens(self, prompt_ids, max_tokens: int,
                        greedy: bool = True, min_p: float = 0.1,
                        eos_id=None, chunk: int = 16, on_ids=None,
                        temperature: float = 1.0, stop_fn=None):
        """Fast generate: device-side hipGraph decode in chunks, host
        sees ids every `chunk` tokens (streaming + EOS stop).  Used by
        runtime.generate() for greedy/min-p on GPU models.
        ``eos_id`` may be an int or a collection of ints (HF configs
        often store a list, e.g. Llama-3.2-Instruct).  ``stop_fn``
        (optional) sees the accumulated ids after each chunk and returns
        True to end generation early (stop-string support; the caller
        truncates — chunking may have produced up to chunk-1 extra
        ids)."""
        import time as _time
        prompt_ids = np.asarray(prompt_ids)
        room = self.max_seq - len(prompt_ids.ravel()) - 1
        if room <= 0:
            raise ValueError(f"prompt fills the {self.max_seq}-token pool")
        max_tokens = min(max_tokens, room)
        eos_set = (set() if eos_id is None else
                   {int(eos_id)} if np.isscalar(eos_id) else
                   {int(e) for e in eos_id})
        t0 = _time.perf_counter()
        with trace_range("prefill"):
            self.prefill(prompt_ids)
        self.last_prefill_time_s = _time.perf_counter() - t0
        out = []
        produced = 0
        first = True
        with trace_range("decode"):
            while produced < max_tokens:
                n = min(chunk, max_tokens - produced)
                ids = self.decode(n, greedy=greedy, min_p=min_p,
                                  use_graph=True, first_from_logits=first,
                                  temperature=temperature)
                first = False
                produced += n
                stop = False
                take = list(ids)
                hit = [j for j, t in enumerate(take) if int(t) in eos_set]
                if hit:
                    take = take[:hit[0] + 1]
                    stop = True
                out.extend(take)
                if on_ids:
                    on_ids(take)
                if stop:
                    break
                if stop_fn is not None and stop_fn(out):
                    break
        return out

    # ------------------------------------------------------------------
    # generic forward (oracle-parity / generate()-compatible)
    # ------------------------------------------------------------------
    def forward(self, ids: np.ndarray, cache: DeviceCacheHandle,
                pos0: int) -> np.ndarray:
        """Returns last-position logits (1, V) fp32 numpy."""
        ids = np.asarray(ids, dtype=np.int32).ravel()
        if pos0 != cache.seq_len:
            raise ValueError("GPU engine requires sequential positions")
        n = len(ids)
        if n == 0:
            raise ValueError("empty ids")
        if pos0 + n > self.max_seq:
            raise ValueError(f"sequence {pos0}+{n} exceeds max_seq "
                             f"{self.max_seq}")
        done = 0
        while done < n:
            M = min(n - done, self.PC)
            self.ids_buf[:M].copy_(
                torch.from_numpy(ids[done:done + M].astype(np.int32)))
            ho.i32_set(self.len_buf, pos0 + done)
            ho.embed(self.embed, self.ids_buf, self.b_h, M,
                     self.config.embed_scale)
            self._layers_forward(M)
            done += M
        ho.i32_set(self.len_buf, pos0 + n)
        self._host_len = pos0 + n
        self._lm_head_last(M)
        cache.seq_len = pos0 + n
        torch.cuda.synchronize()
        return self.b_logits.float().cpu().numpy()[None, :]

    # ------------------------------------------------------------------
    # fast device-side decode loop (graph-replayable)
    # ------------------------------------------------------------------
    def _dgemv(self, lw, name, x, y, **kw):
        """Decode GEMV: MXFP4 / fp8 / bf16 weights by engine dtype."""
        if self.wq4:
            ho.gemv_fp4(lw[name + "_q4"], lw[name + "_e4"], x, y, **kw)
        elif self.fp8:
            ho.gemv_fp8(lw[name + "_q"], lw[name + "_s"], x, y, **kw)
        else:
            ho.gemv(lw[name], x, y, **kw)

    def _pf_tensors(self, lw):
        if self.wq4:
            return (lw["wqkv_q4"], lw["wo_q4"], lw["wgu_q4"],
                    lw["wdown_q4"])
        if self.fp8:
            return (lw["wqkv_q"], lw["wo_q"], lw["wgu_q"], lw["wdown_q"])
        return (lw["wqkv"], lw["wo"], lw["wgu"], lw["wdown"])

    def _decode_step(self, greedy: bool, min_p: float,
                     temperature: float = 1.0):
        """Fused decode path: 4 kernels/layer (llama) or 5 (gemma) —
        RMSNorm and GLU live inside the GEMV staging pass, RoPE + KV
        write inside the attention kernel; a side stream prefetches the
        next layer's weights through MALL, paced by per-layer events."""
        cfg = self.config
        eps = cfg.rms_norm_eps
        h = self.b_h[0]
        hnext = self.b_hb
        t1 = self.b_t1[0]
        t2 = self.b_t2
        s0 = torch.cuda.current_stream()
        pf = self.prefetch_on
        self._pf_events = []
        prev = None  # previous gemma layer (its postffn norm fused here)
        for i, lw in enumerate(self.layers):
            if pf and i + 1 < len(self.layers):
                ev = torch.cuda.Event()
                ev.record(s0)
                self._pf_events.append(ev)
                self._pf_stream.wait_event(ev)
                with torch.cuda.stream(self._pf_stream):
                    for t in self._pf_tensors(self.layers[i + 1]):
                        ho.prefetch(t, self._pf_sink)
            window = cfg.sliding_window if cfg.is_sliding(i) else 0
            if self.gemma and prev is not None:
                # h' = h + norm(t2_prev)*g_postffn_prev; stage norm(h')*g_in
                self._dgemv(lw, "wqkv", t2, self.b_qkv,
                            stage=ho.STAGE_NORM2, x2=h,
                            g=prev["g_postffn"], g2=lw["g_in"], res=hnext,
                            eps=eps)
                h, hnext = hnext, h
            elif i == 0 and "bqkv" not in lw:
                # layer 0: embed gather fused into the QKV staging pass
                # (x = table, x2 = sampled token id, res = persisted h)
                self._dgemv(lw, "wqkv", self.embed, self.b_qkv,
                            stage=ho.STAGE_NORM_EMBED, x2=self.next_token,
                            g=lw["g_in"], res=h, eps=eps,
                            escale=cfg.embed_scale)
            else:
                if i == 0:  # bias models keep the separate embed gather
                    ho.embed(self.embed, self.next_token, self.b_h, 1,
                             cfg.embed_scale)
                self._dgemv(lw, "wqkv", h, self.b_qkv, stage=ho.STAGE_NORM,
                            g=lw["g_in"], res=lw.get("bqkv"), eps=eps)
            prev = lw
            ho.attn_dec(self.b_qkv, self.k_cache[i], self.v_cache[i],
                        self.b_att[0], self.len_buf, self.cos_t, self.sin_t,
                        self.attn_scratch, self.attn_cnt,
                        self.nh_l, self.kvh_l, self.hd, self.scale,
                        softcap=self.attn_softcap, window=window or 0,
                        split=self.attn_split,
                        kS=self.k_scale[i], vS=self.v_scale[i])
            if self.gemma:
                self._dgemv(lw, "wo", self.b_att[0], t1)
                tpu.all_reduce(t1)
                # NORM2: h' = h + norm(t1)*g_post (persisted to hnext),
                # stage = norm(h')*g_preffn — both sandwich norms fused
                self._dgemv(lw, "wgu", t1, self.b_gu, stage=ho.STAGE_NORM2,
                            x2=h, g=lw["g_post"], g2=lw["g_preffn"],
                            res=hnext, eps=eps)
                h, hnext = hnext, h
                self._dgemv(lw, "wdown", self.b_gu[:self.inter_l], t2,
                            stage=ho.STAGE_GLU, x2=self.b_gu[self.inter_l:],
                            act=self.act)
                tpu.all_reduce(t2)
            else:
                if self.tp_branch:
                    self._dgemv(lw, "wo", self.b_att[0], t1)
                    tpu.all_reduce(t1)
                    ho.addinto(h, t1)
                else:
                    self._dgemv(lw, "wo", self.b_att[0], h, res=h)
                if self.moe:
                    # router -> topk expert-indexed GEMV pairs: W base
                    # offset by the device idx, output scaled by the
                    # device router prob, accumulated into h via res.
                    # xn is staged ONCE: expert 0's down-proj already
                    # updates h, so expert 1 must not re-norm it
                    ho.moe_route(h, lw["g_post"], lw["wg"], 1, self.topk,
                                 self.moe_idx, self.moe_w, eps=eps)
                    xn0 = self.b_xn[0]
                    ho.rmsnorm(h, lw["g_post"], xn0, eps=eps)
                    I, H = self.inter_l, self.H
                    for j in range(self.topk):
                        ei = self.moe_idx[j:j + 1]
                        osc = self.moe_w[j:j + 1]
                        if self.fp8:
                            ho.gemv_fp8(lw["wgu_q"], lw["wgu_s"], xn0,
                                        self.b_gu, eidx=ei,
                                        wstride=2 * I * H, sstride=2 * I)
                            ho.gemv_fp8(lw["wdown_q"], lw["wdown_s"],
                                        self.b_gu[:I], h, res=h,
                                        stage=ho.STAGE_GLU,
                                        x2=self.b_gu[I:], act=self.act,
                                        eidx=ei, wstride=H * I,
                                        sstride=H, oscale=osc)
                        else:
                            ho.gemv(lw["wgu"], xn0, self.b_gu, eidx=ei,
                                    wstride=2 * I * H)
                            ho.gemv(lw["wdown"], self.b_gu[:I], h, res=h,
                                    stage=ho.STAGE_GLU, x2=self.b_gu[I:],
                                    act=self.act, eidx=ei,
                                    wstride=H * I, oscale=osc)
                    continue
                self._dgemv(lw, "wgu", h, self.b_gu, stage=ho.STAGE_NORM,
                            g=lw["g_post"], eps=eps)
                if self.tp_branch:
                    self._dgemv(lw, "wdown", self.b_gu[:self.inter_l], t1,
                                stage=ho.STAGE_GLU,
                                x2=self.b_gu[self.inter_l:], act=self.act)
                    tpu.all_reduce(t1)
                    ho.addinto(h, t1)
                else:
                    self._dgemv(lw, "wdown", self.b_gu[:self.inter_l], h,
                                res=h, stage=ho.STAGE_GLU,
                                x2=self.b_gu[self.inter_l:], act=self.act)
        if pf:
            ev = torch.cuda.Event()
            ev.record(self._pf_stream)
            self._pf_events.append(ev)
            s0.wait_event(ev)  # join the fork before the step ends
        if self.gemma:
            # last layer's post-ffn sandwich norm + final norm, both
            # fused into the lm_head staging (NORM2: h' = h + norm(t2)*
            # g_postffn, then stage norm(h')*g_final)
            kw = dict(stage=ho.STAGE_NORM2, x2=h,
                      g=self.layers[-1]["g_postffn"], g2=self.g_final,
                      res=hnext, eps=eps, softcap=self.final_softcap)
            if self.wq4:
                ho.gemv_fp4(self.lm_head_q4, self.lm_head_e4, t2,
                            self.b_logits_l, **kw)
            elif self.fp8:
                ho.gemv_fp8(self.lm_head_q, self.lm_head_s, t2,
                            self.b_logits_l, **kw)
            else:
                ho.gemv(self.lm_head, t2, self.b_logits_l, **kw)
        else:
            # final norm fused into the lm_head staging pass
            kw = dict(stage=ho.STAGE_NORM, g=self.g_final, eps=eps,
                      softcap=self.final_softcap)
            if self.wq4:
                ho.gemv_fp4(self.lm_head_q4, self.lm_head_e4, h,
                            self.b_logits_l, **kw)
            elif self.fp8:
                ho.gemv_fp8(self.lm_head_q, self.lm_head_s, h,
                            self.b_logits_l, **kw)
            else:
                ho.gemv(self.lm_head, h, self.b_logits_l, **kw)
        if self.tp_branch:
            tpu.all_gather_into(self.b_logits, self.b_logits_l)
        ho.sample(self.b_logits, min_p, greedy, self.seed, self.rng_ctr,
                  self.s_gmax, self.s_pick, self.next_token, self.out_ring,
                  self.nout, self.len_buf, bump_len=True,
                  temperature=temperature, cnt=self.s_cnt)

    # ------------------------------------------------------------------
    # lockstep batched decode (throughput mode; beyond-parity capability)
    # ------------------------------------------------------------------
    def _decode_batch_step(self, B: int, greedy: bool, min_p: float,
                           temperature: float = 1.0):
        """One decode step for B lockstep sequences.  fp8 weights with
        B<=8 take the fused MULTI-X GEMV path (one weight stream feeds
        B accumulators — near single-sequence step time); otherwise the
        layer stack runs as M=B-row MFMA GEMMs.  Graph-replayable
        (all state device-side)."""
        cfg = self.config
        # measured (profiles/decode_kernels_r02.md): the SKINNY fp8 MFMA
        # GEMM wins at every B (even B=2: 1.34 vs the multi-x GEMV's
        # 1.44 ms — the GEMV is VALU-ISSUE-bound, 64 irreducible
        # v_pk_fma per 16 weight bytes at B=8).  The multi-x path stays
        # available via LLM_BATCH_MX_MAX for comparison.
        import os as _os  # noqa: delayed so env changes apply per call
        mx_max = int(_os.environ.get("LLM_BATCH_MX_MAX", "0"))
        if (self.fp8 and B <= mx_max and not cfg.attention_bias
                and not self.moe):
            return self._decode_batch_step_mx(B, greedy, min_p, temperature)
        if self.fp8 and B <= 16 and not self.moe:
            return self._decode_batch_step_skinny(B, greedy, min_p,
                                                  temperature)
        ho.embed(self.embed, self.bt_next, self.b_h, B, cfg.embed_scale)
        self._layers_forward(B, batch_attn=True)
        ho.rmsnorm(self.b_h[:B], self.g_final, self.b_xn[:B],
                   eps=cfg.rms_norm_eps)
        if self.fp8:
            ho.quant_fp8(self.b_xn[:B], self.b_xq, self.b_sx)
            ho.gemm_fp8(self.b_xq, self.b_sx, self.lm_head_q,
                        self.lm_head_s, self.bt_logits[:B], B, self.H,
                        accbuf=self.b_gemm_acc)
        else:
            ho.gemm(self.b_xn[:B], self.lm_head, self.bt_logits[:B],
                    accbuf=self.b_gemm_acc)
        if self.final_softcap:
            ho.softcap(self.bt_logits[:B], self.final_softcap)
        ho.sample(self.bt_logits[:B], min_p, greedy, self.seed,
                  self.rng_ctr, self.bt_gmax, self.bt_pick, self.bt_next,
                  self.bt_ring, self.bt_nout, self.bt_lens, bump_len=True,
                  temperature=temperature, cnt=self.bt_cnt, batch=B)

    def _decode_batch_step_mx(self, B: int, greedy: bool, min_p: float,
                              temperature: float = 1.0):
        """Fused batched decode (fp8, B<=8): mirrors _decode_step's
        4-5 kernels/layer with multi-x GEMVs over B rows."""
        cfg = self.config
        eps = cfg.rms_norm_eps
        H, I = self.H, self.inter_l
        h = self.b_h            # (PC, H): rows 0..B = per-seq h
        hb = self.bt_hb         # gemma ping-pong
        t1, t2, gu = self.bt_t1, self.bt_t2, self.bt_gu
        qkvw = (self.nh_l + 2 * self.kvh_l) * self.hd
        nhh = self.nh_l * self.hd
        prev = None
        for i, lw in enumerate(self.layers):
            window = cfg.sliding_window if cfg.is_sliding(i) else 0
            if self.gemma and prev is not None:
                ho.gemv_fp8_mx(lw["wqkv_q"], lw["wqkv_s"], t2, self.bt_qkv,
                               B, H, qkvw, stage=ho.STAGE_NORM2, x2=h,
                               x2stride=H, g=prev["g_postffn"],
                               g2=lw["g_in"], hout=hb, hstride=H, eps=eps)
                h, hb = hb, h
            elif i == 0:
                ho.gemv_fp8_mx(lw["wqkv_q"], lw["wqkv_s"], self.embed,
                               self.bt_qkv, B, 0, qkvw,
                               stage=ho.STAGE_NORM_EMBED,
                               x2=self.bt_next, g=lw["g_in"], hout=h,
                               hstride=H, eps=eps,
                               escale=cfg.embed_scale)
            else:
                ho.gemv_fp8_mx(lw["wqkv_q"], lw["wqkv_s"], h, self.bt_qkv,
                               B, H, qkvw, stage=ho.STAGE_NORM,
                               g=lw["g_in"], eps=eps)
            prev = lw
            ho.attn_dec(self.bt_qkv, self.k_cache[i], self.v_cache[i],
                        self.b_att[0], self.bt_lens, self.cos_t, self.sin_t,
                        self.bt_scratch, self.bt_attn_cnt,
                        self.nh_l, self.kvh_l, self.hd, self.scale,
                        softcap=self.attn_softcap, window=window or 0,
                        split=self.attn_split, kS=self.k_scale[i],
                        vS=self.v_scale[i], batch=B)
            if self.gemma:
                ho.gemv_fp8_mx(lw["wo_q"], lw["wo_s"], self.b_att, t1, B,
                               nhh, H)
                ho.gemv_fp8_mx(lw["wgu_q"], lw["wgu_s"], t1, gu, B, H,
                               2 * I, stage=ho.STAGE_NORM2, x2=h,
                               x2stride=H, g=lw["g_post"],
                               g2=lw["g_preffn"], hout=hb, hstride=H,
                               eps=eps)
                h, hb = hb, h
                ho.gemv_fp8_mx(lw["wdown_q"], lw["wdown_s"], gu, t2, B,
                               2 * I, H, stage=ho.STAGE_GLU,
                               x2=gu[:, I:], x2stride=2 * I, act=self.act)
            else:
                ho.gemv_fp8_mx(lw["wo_q"], lw["wo_s"], self.b_att, h, B,
                               nhh, H, res=h, rstride=H)
                ho.gemv_fp8_mx(lw["wgu_q"], lw["wgu_s"], h, gu, B, H,
                               2 * I, stage=ho.STAGE_NORM, g=lw["g_post"],
                               eps=eps)
                ho.gemv_fp8_mx(lw["wdown_q"], lw["wdown_s"], gu, h, B,
                               2 * I, H, stage=ho.STAGE_GLU, x2=gu[:, I:],
                               x2stride=2 * I, act=self.act, res=h,
                               rstride=H)
        if self.gemma:
            ho.gemv_fp8_mx(self.lm_head_q, self.lm_head_s, t2,
                           self.bt_logits, B, H, self.vocab_l,
                           stage=ho.STAGE_NORM2, x2=h, x2stride=H,
                           g=self.layers[-1]["g_postffn"], g2=self.g_final,
                           hout=hb, hstride=H, eps=eps,
                           softcap=self.final_softcap)
        else:
            ho.gemv_fp8_mx(self.lm_head_q, self.lm_head_s, h,
                           self.bt_logits, B, H, self.vocab_l,
                           stage=ho.STAGE_NORM, g=self.g_final, eps=eps,
                           softcap=self.final_softcap)
        ho.sample(self.bt_logits[:B], min_p, greedy, self.seed,
                  self.rng_ctr, self.bt_gmax, self.bt_pick, self.bt_next,
                  self.bt_ring, self.bt_nout, self.bt_lens, bump_len=True,
                  temperature=temperature, cnt=self.bt_cnt, batch=B)

    def _decode_batch_step_skinny(self, B: int, greedy: bool,
                                  min_p: float, temperature: float = 1.0):
        """Batched decode via (stage+quant, skinny fp8 MFMA GEMM) pairs:
        B = 3..16 lockstep rows, W streamed once per projection."""
        cfg = self.config
        eps = cfg.rms_norm_eps
        H, I = self.H, self.inter_l
        h = self.b_h
        hb = self.bt_hb
        t1, t2, gu = self.bt_t1, self.bt_t2, self.bt_gu
        qkvw = (self.nh_l + 2 * self.kvh_l) * self.hd
        nhh = self.nh_l * self.hd
        xq, sx = self.b_xq, self.b_sx
        prev = None
        for i, lw in enumerate(self.layers):
            window = cfg.sliding_window if cfg.is_sliding(i) else 0
            if self.gemma and prev is not None:
                ho.stage_quant_mx(t2, H, xq, sx, B, H, stage=ho.STAGE_NORM2,
                                  x2=h, x2stride=H, g=prev["g_postffn"],
                                  g2=lw["g_in"], hout=hb, hstride=H,
                                  eps=eps)
                h, hb = hb, h
            elif i == 0:
                ho.stage_quant_mx(self.embed, 0, xq, sx, B, H,
                                  stage=ho.STAGE_NORM_EMBED,
                                  x2=self.bt_next, g=lw["g_in"], hout=h,
                                  hstride=H, eps=eps,
                                  escale=cfg.embed_scale)
            else:
                ho.stage_quant_mx(h, H, xq, sx, B, H, stage=ho.STAGE_NORM,
                                  g=lw["g_in"], eps=eps)
            ho.gemm_fp8_skinny(xq, sx, lw["wqkv_q"], lw["wqkv_s"],
                               self.bt_qkv, B, qkvw, bias=lw.get("bqkv"),
                               accbuf=self.b_gemm_acc)
            prev = lw
            ho.attn_dec(self.bt_qkv, self.k_cache[i], self.v_cache[i],
                        self.b_att[0], self.bt_lens, self.cos_t, self.sin_t,
                        self.bt_scratch, self.bt_attn_cnt,
                        self.nh_l, self.kvh_l, self.hd, self.scale,
                        softcap=self.attn_softcap, window=window or 0,
                        split=self.attn_split, kS=self.k_scale[i],
                        vS=self.v_scale[i], batch=B)
            ho.stage_quant_mx(self.b_att, nhh, xq, sx, B, nhh)
            if self.gemma:
                ho.gemm_fp8_skinny(xq, sx, lw["wo_q"], lw["wo_s"], t1, B, H,
                                   accbuf=self.b_gemm_acc)
                ho.stage_quant_mx(t1, H, xq, sx, B, H,
                                  stage=ho.STAGE_NORM2, x2=h, x2stride=H,
                                  g=lw["g_post"], g2=lw["g_preffn"],
                                  hout=hb, hstride=H, eps=eps)
                h, hb = hb, h
                ho.gemm_fp8_skinny(xq, sx, lw["wgu_q"], lw["wgu_s"], gu, B,
                                   2 * I, accbuf=self.b_gemm_acc)
                ho.stage_quant_mx(gu, 2 * I, xq, sx, B, I,
                                  stage=ho.STAGE_GLU, x2=gu[:, I:],
                                  x2stride=2 * I, act=self.act)
                ho.gemm_fp8_skinny(xq, sx, lw["wdown_q"], lw["wdown_s"],
                                   t2, B, H, accbuf=self.b_gemm_acc)
            else:
                ho.gemm_fp8_skinny(xq, sx, lw["wo_q"], lw["wo_s"], h, B, H,
                                   res=h, rstride=H,
                                   accbuf=self.b_gemm_acc)
                ho.stage_quant_mx(h, H, xq, sx, B, H, stage=ho.STAGE_NORM,
                                  g=lw["g_post"], eps=eps)
                ho.gemm_fp8_skinny(xq, sx, lw["wgu_q"], lw["wgu_s"], gu, B,
                                   2 * I, accbuf=self.b_gemm_acc)
                ho.stage_quant_mx(gu, 2 * I, xq, sx, B, I,
                                  stage=ho.STAGE_GLU, x2=gu[:, I:],
                                  x2stride=2 * I, act=self.act)
                ho.gemm_fp8_skinny(xq, sx, lw["wdown_q"], lw["wdown_s"], h,
                                   B, H, res=h, rstride=H,
                                   accbuf=self.b_gemm_acc)
        if self.gemma:
            ho.stage_quant_mx(t2, H, xq, sx, B, H, stage=ho.STAGE_NORM2,
                              x2=h, x2stride=H,
                              g=self.layers[-1]["g_postffn"],
                              g2=self.g_final, hout=hb, hstride=H, eps=eps)
        else:
            ho.stage_quant_mx(h, H, xq, sx, B, H, stage=ho.STAGE_NORM,
                              g=self.g_final, eps=eps)
        ho.gemm_fp8_skinny(xq, sx, self.lm_head_q, self.lm_head_s,
                           self.bt_logits, B, self.vocab_l,
                           softcap=self.final_softcap,
                           accbuf=self.b_gemm_acc)
        ho.sample(self.bt_logits[:B], min_p, greedy, self.seed,
                  self.rng_ctr, self.bt_gmax, self.bt_pick, self.bt_next,
                  self.bt_ring, self.bt_nout, self.bt_lens, bump_len=True,
                  temperature=temperature, cnt=self.bt_cnt, batch=B)

    def prefill_batch(self, prompts) -> None:
        """Prefill B prompts (RAGGED lengths allowed) into per-sequence
        KV pools; each row then decodes from its own position
        (bt_lens[b] device-side)."""
        seqs = [np.asarray(p, dtype=np.int32).ravel() for p in prompts]
        B = len(seqs)
        assert 1 <= B <= self.max_batch and self.max_batch > 1
        assert self.world == 1, "batched decode is single-GPU for now"
        lens = [len(sq) for sq in seqs]
        if min(lens) == 0:
            raise ValueError("empty prompt in batch")
        if max(lens) + 1 >= self.max_seq:
            raise ValueError(f"prompt {max(lens)} fills the "
                             f"{self.max_seq} pool")
        self.reset()
        self.bt_nout.zero_()
        for b in range(B):
            self._pb = b
            P = lens[b]
            done = 0
            while done < P:
                M = min(P - done, self.PC)
                self.ids_buf[:M].copy_(
                    torch.from_numpy(seqs[b][done:done + M]))
                ho.i32_set(self.len_buf, done)
                ho.embed(self.embed, self.ids_buf, self.b_h, M,
                         self.config.embed_scale)
                self._layers_forward(M)
                done += M
            self._lm_head_last(M)
            self.bt_logits32[b].copy_(self.b_logits)
        self._pb = 0
        self.bt_lens[:B].copy_(torch.tensor(lens, dtype=torch.int32))
        self._host_lens = lens
        self._batch_n = B

    def decode_batch(self, n_tokens: int, greedy: bool = True,
                     min_p: float = 0.1, temperature: float = 1.0,
                     use_graph: bool = True,
                     first_from_logits: bool = True) -> np.ndarray:
        """Decode n_tokens for every prefilled sequence; returns int32
        ids of shape (B, n_tokens).  ``first_from_logits=False``
        continues a previous decode_batch (chunked serving)."""
        B = self._batch_n
        if max(self._host_lens) + n_tokens > self.max_seq:
            raise ValueError("decode_batch would overflow the KV pool")
        self._host_lens = [p + n_tokens for p in self._host_lens]
        if first_from_logits:
            # first tokens from the prefill logits (fp32, batch rows)
            ho.sample(self.bt_logits32[:B], min_p, greedy, self.seed,
                      self.rng_ctr, self.bt_gmax, self.bt_pick,
                      self.bt_next, self.bt_ring, self.bt_nout,
                      self.bt_lens, bump_len=False,
                      temperature=temperature, cnt=self.bt_cnt, batch=B)
        n_steps = n_tokens - (1 if first_from_logits else 0)
        key = ("batch", B, greedy, min_p, temperature)
        if use_graph and n_steps > 0 and self._graph_mode != key \
                and not getattr(self, "_graph_failed", False):
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                self._decode_batch_step(B, greedy, min_p, temperature)
            torch.cuda.current_stream().wait_stream(s)
            torch.cuda.synchronize()
            n_steps -= 1
            try:
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g):
                    self._decode_batch_step(B, greedy, min_p, temperature)
                self._graph = g
                self._graph_mode = key
            except Exception as e:
                self._graph_failed = True
                self._graph_error = e
                use_graph = False
        if use_graph and self._graph_mode == key:
            for _ in range(n_steps):
                self._graph.replay()
        else:
            for _ in range(n_steps):
                self._decode_batch_step(B, greedy, min_p, temperature)
        torch.cuda.synchronize()
        n = int(self.bt_nout[0].item())
        out = self.bt_ring[:B, :n].cpu().numpy()
        return out[:, max(0, n - n_tokens):]

    # ---- continuous batching primitives (rows join/leave the group) ----
    def prefill_row(self, b: int, ids, greedy: bool = True,
                    min_p: float = 0.1, temperature: float = 1.0):
        """Prefill ONE sequence into slot b (other rows untouched) and
        sample its first token — a request JOINING the lockstep group
        between decode chunks (server continuous batching)."""
        ids = np.asarray(ids, dtype=np.int32).ravel()
        P = len(ids)
        if P == 0:
            raise ValueError("empty prompt")
        assert 0 <= b < self.max_batch and self.max_batch > 1
        assert self.world == 1
        if P + 1 >= self.max_seq:
            raise ValueError(f"prompt {P} fills the {self.max_seq} pool")
        self._pb = b
        done = 0
        while done < P:
            M = min(P - done, self.PC)
            self.ids_buf[:M].copy_(torch.from_numpy(ids[done:done + M]))
            ho.i32_set(self.len_buf, done)
            ho.embed(self.embed, self.ids_buf, self.b_h, M,
                     self.config.embed_scale)
            self._layers_forward(M)
            done += M
        self._lm_head_last(M)
        self._pb = 0
        self.bt_lens[b:b + 1].fill_(P)
        self.bt_nout[b:b + 1].zero_()
        # first token for this row only (row-sliced sampler state)
        ho.sample(self.b_logits, min_p, greedy, self.seed, self.rng_ctr,
                  self.bt_gmax[b:b + 1], self.bt_pick[b:b + 1],
                  self.bt_next[b:b + 1], self.bt_ring[b:b + 1],
                  self.bt_nout[b:b + 1], self.bt_lens[b:b + 1],
                  bump_len=False, temperature=temperature,
                  cnt=self.bt_cnt[b:b + 1], batch=1)
        if not hasattr(self, "_host_lens") or self._host_lens is None:
            self._host_lens = [0] * self.max_batch
        while len(self._host_lens) < self.max_batch:
            self._host_lens.append(0)
        self._host_lens[b] = P
        self._batch_n = max(getattr(self, "_batch_n", 0), b + 1)

    def compact_row(self, dst: int, src: int):
        """Move sequence slot src -> dst (KV pools, scales, device row
        state) so retired rows leave the group dense."""
        if dst == src:
            return
        for i in range(self.config.num_hidden_layers):
            self.k_cache[i][dst].copy_(self.k_cache[i][src])
            self.v_cache[i][dst].copy_(self.v_cache[i][src])
            if self.kv8:
                self.k_scale[i][dst].copy_(self.k_scale[i][src])
                self.v_scale[i][dst].copy_(self.v_scale[i][src])
        for t in (self.bt_lens, self.bt_next, self.bt_nout, self.bt_gmax,
                  self.bt_pick, self.bt_cnt):
            t[dst:dst + 1].copy_(t[src:src + 1])
        self.bt_ring[dst].copy_(self.bt_ring[src])
        self._host_lens[dst] = self._host_lens[src]

    def decode_rows(self, B: int, n: int, greedy: bool = True,
                    min_p: float = 0.1, temperature: float = 1.0,
                    use_graph: bool = True):
        """n lockstep steps for rows [0, B) (each already holding a
        sampled next token from prefill_row / a previous chunk).
        Returns a list of B arrays: each row's n new token ids."""
        assert 1 <= B <= self.max_batch
        if max(self._host_lens[:B]) + n > self.max_seq:
            raise ValueError("decode_rows would overflow the KV pool")
        for b in range(B):
            self._host_lens[b] += n
        key = ("batchc", B, greedy, min_p, temperature)
        if not hasattr(self, "_bgraphs"):
            self._bgraphs = {}
        g = self._bgraphs.get(key)
        executed = 0
        if (use_graph and g is None and n > 1
                and not getattr(self, "_graph_failed", False)):
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                self._decode_batch_step(B, greedy, min_p, temperature)
            torch.cuda.current_stream().wait_stream(s)
            torch.cuda.synchronize()
            executed += 1  # the warm-up was a real step
            try:
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g):
                    self._decode_batch_step(B, greedy, min_p, temperature)
                self._bgraphs[key] = g
            except Exception as e:
                self._graph_failed = True
                self._graph_error = e
                g = None
        remaining = n - executed
        if g is not None and use_graph:
            for _ in range(remaining):
                g.replay()
        else:
            for _ in range(remaining):
                self._decode_batch_step(B, greedy, min_p, temperature)
        torch.cuda.synchronize()
        comm = tpu.xgmi_comm()
        if comm is not None:
            comm.check()
        nouts = self.bt_nout[:B].cpu().numpy()
        ring = self.bt_ring[:B].cpu().numpy()
        # each step appends exactly one id per row
        return [ring[b, max(0, int(nouts[b]) - n):int(nouts[b])]
                for b in range(B)]

    def generate_tokens_batch(self, prompts, max_tokens: int,
                              greedy: bool = True, min_p: float = 0.1,
                              temperature: float = 1.0) -> np.ndarray:
        """Convenience: prefill_batch + decode_batch -> (B, max_tokens)."""
        self.prefill_batch(prompts)
        return self.decode_batch(max_tokens, greedy=greedy, min_p=min_p,
                                 temperature=temperature)

    def capture_decode_graph(self, greedy: bool = True, min_p: float = 0.1,
                             temperature: float = 1.0):
        """Capture one decode step into a hipGraph (fixed shapes: KV pool
        is preallocated and the position is a device scalar — SURVEY §7
        'graph must be shape-stable').

        NOTE: the warm-up executes ONE REAL decode step (advances the
        sequence by one token).  Returns ``(steps_taken, ok)``:
        ``steps_taken`` is the number of real decode steps executed here
        (1 whenever the warm-up ran, 0 when the cached graph is reused)
        — callers MUST account for it even when ``ok`` is False.  On
        capture failure ``ok`` is False, ``self._graph_failed`` is set
        (no further capture attempts) and the exception is kept in
        ``self._graph_error``."""
        mode = (greedy, min_p, temperature)
        if self._graph_mode == mode:
            return 0, True
        if getattr(self, "_graph_failed", False):
            return 0, False
        # warm-up (a real step) on a side stream, then capture
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            self._decode_step(greedy, min_p, temperature)
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        try:
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                self._decode_step(greedy, min_p, temperature)
        except Exception as e:
            # the warm-up step above DID run (token committed to
            # out_ring, len_buf advanced) — report it so decode() does
            # not double-count
            self._graph_failed = True
            self._graph_error = e
            torch.cuda.synchronize()
            return 1, False
        self._graph = g
        self._graph_mode = mode
        return 1, True

    def prefill(self, ids: np.ndarray, cache: Optional[DeviceCacheHandle] = None):
        cache = cache or self.make_cache(self.max_seq)
        with trace_range(f"prefill[{len(np.asarray(ids).ravel())}]"):
            logits = self.forward(ids, cache, 0)
        return cache, logits

    def decode(self, n_tokens: int, greedy: bool = True, min_p: float = 0.1,
               use_graph: bool = True, first_from_logits: bool = True,
               temperature: float = 1.0):
        """Generate n_tokens ids device-side; returns int32 numpy ids.
        Assumes prefill() ran (len_buf == prompt length, logits ready)."""
        if getattr(self, "_host_len", 0) + n_tokens > self.max_seq:
            raise ValueError(
                f"decode would overflow the KV pool: len {self._host_len} "
                f"+ {n_tokens} > max_seq {self.max_seq}")
        self._host_len += n_tokens
        if first_from_logits:
            # sample token 0 from the prefill logits
            ho.sample(self.b_logits, min_p, greedy, self.seed, self.rng_ctr,
                      self.s_gmax, self.s_pick, self.next_token,
                      self.out_ring, self.nout, self.len_buf,
                      bump_len=False, temperature=temperature,
                      cnt=self.s_cnt)
        n_steps = n_tokens - (1 if first_from_logits else 0)
        if use_graph and n_steps > 0:
            # capture_decode_graph reports the warm-up step it executed
            # even when capture fails, so n_steps stays exact either way
            taken, ok = self.capture_decode_graph(greedy, min_p, temperature)
            n_steps -= taken
            if not ok:
                use_graph = False
        if use_graph and n_steps > 0:
            for _ in range(n_steps):
                self._graph.replay()
        else:
            for _ in range(n_steps):
                self._decode_step(greedy, min_p, temperature)
        torch.cuda.synchronize()
        comm = tpu.xgmi_comm()
        if comm is not None:
            comm.check()  # raise if a one-shot collective timed out
        n = int(self.nout.item())
        return self.out_ring[:n].cpu().numpy()[max(0, n - n_tokens):]
