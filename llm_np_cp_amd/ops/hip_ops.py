"""ctypes bindings for the hand-written CDNA4 kernel library.

The library (``_libllmops.so``, built in-tree by ``csrc/build.py`` /
``__graft_entry__.build()``) exposes plain-C launch functions taking raw
device pointers + the HIP stream; tensors stay ``torch`` tensors (memory
containers only) and every launch lands on the *current* torch stream, so
the whole decode step can be captured in a hipGraph via
``torch.cuda.CUDAGraph``.

This module fails loudly if the extension is missing while a GPU is
present — there is no silent eager fallback (the HIP path must be the one
that runs).
"""

from __future__ import annotations

import ctypes
import os

import torch

_LIB = None
_LIB_PATH = os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "_libllmops.so")

_SIGS = {
    "launch_gemv_bf16": [ctypes.c_void_p] * 7 + [ctypes.c_int] * 4 +
                        [ctypes.c_float, ctypes.c_int, ctypes.c_float,
                         ctypes.c_int, ctypes.c_int, ctypes.c_int,
                         ctypes.c_float, ctypes.c_void_p, ctypes.c_long,
                         ctypes.c_void_p, ctypes.c_void_p],
    "launch_gemv_fp8": [ctypes.c_void_p] * 8 + [ctypes.c_int] * 4 +
                       [ctypes.c_float, ctypes.c_int, ctypes.c_float,
                        ctypes.c_int, ctypes.c_int, ctypes.c_int,
                        ctypes.c_float, ctypes.c_void_p, ctypes.c_long,
                        ctypes.c_long, ctypes.c_void_p, ctypes.c_void_p],
    "launch_rmsnorm": [ctypes.c_void_p] * 4 + [ctypes.c_int] * 2 +
                      [ctypes.c_float, ctypes.c_int, ctypes.c_void_p],
    "launch_rope_cache": [ctypes.c_void_p] * 7 + [ctypes.c_int] +
                         [ctypes.c_void_p] * 3 + [ctypes.c_int] * 5 +
                         [ctypes.c_void_p],
    "launch_attn": [ctypes.c_void_p] * 7 + [ctypes.c_int] * 6 +
                   [ctypes.c_float, ctypes.c_float, ctypes.c_int,
                    ctypes.c_void_p],
    "launch_attn_prefill_mfma": [ctypes.c_void_p] * 7 + [ctypes.c_int] * 6 +
                                [ctypes.c_float, ctypes.c_float,
                                 ctypes.c_int, ctypes.c_void_p],
    "launch_attn_dec": [ctypes.c_void_p] * 9 + [ctypes.c_int] +
                       [ctypes.c_void_p] * 2 + [ctypes.c_int] * 6 +
                       [ctypes.c_float, ctypes.c_float, ctypes.c_int,
                        ctypes.c_void_p],
    "launch_glu": [ctypes.c_void_p] * 3 + [ctypes.c_long, ctypes.c_int,
                                           ctypes.c_void_p],
    "launch_embed": [ctypes.c_void_p] * 3 + [ctypes.c_int] * 2 +
                    [ctypes.c_float, ctypes.c_void_p],
    "launch_sample": [ctypes.c_void_p, ctypes.c_int, ctypes.c_int,
                      ctypes.c_int, ctypes.c_long, ctypes.c_float,
                      ctypes.c_int, ctypes.c_uint64, ctypes.c_float] +
                     [ctypes.c_void_p] * 8 +
                     [ctypes.c_int, ctypes.c_void_p],
    "launch_gemm_bf16": [ctypes.c_void_p] * 5 + [ctypes.c_int] * 3 +
                        [ctypes.c_void_p],
    "launch_prefetch": [ctypes.c_void_p, ctypes.c_long, ctypes.c_void_p,
                        ctypes.c_void_p],
    "launch_softcap": [ctypes.c_void_p, ctypes.c_long, ctypes.c_float,
                       ctypes.c_void_p],
    "launch_addinto": [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_long,
                       ctypes.c_void_p],
    "launch_bias_add": [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int,
                        ctypes.c_int, ctypes.c_void_p],
    "launch_i32_set": [ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p],
    "launch_quant_fp8": [ctypes.c_void_p] * 3 + [ctypes.c_int] * 2 +
                        [ctypes.c_void_p],
    "launch_quant_fp4": [ctypes.c_void_p] * 3 + [ctypes.c_int] * 2 +
                        [ctypes.c_void_p],
    "launch_gemv_fp4": [ctypes.c_void_p] * 8 + [ctypes.c_int] * 4 +
                       [ctypes.c_float, ctypes.c_int, ctypes.c_float,
                        ctypes.c_int, ctypes.c_int, ctypes.c_int,
                        ctypes.c_float, ctypes.c_void_p],
    "launch_stage_quant_mx": [ctypes.c_void_p, ctypes.c_long,
                              ctypes.c_void_p, ctypes.c_long] +
                             [ctypes.c_void_p] * 4 +
                             [ctypes.c_void_p, ctypes.c_long] +
                             [ctypes.c_int] * 4 +
                             [ctypes.c_float, ctypes.c_float,
                              ctypes.c_void_p],
    "launch_gemm_fp8_skinny": [ctypes.c_void_p] * 5 + [ctypes.c_long] +
                              [ctypes.c_void_p, ctypes.c_long] +
                              [ctypes.c_void_p, ctypes.c_void_p,
                               ctypes.c_int,
                               ctypes.c_float] + [ctypes.c_int] * 3 +
                              [ctypes.c_void_p],
    "launch_gemv_fp8_mx": [ctypes.c_void_p] * 3 + [ctypes.c_long] +
                          [ctypes.c_void_p, ctypes.c_long] +
                          [ctypes.c_void_p] * 3 + [ctypes.c_long] +
                          [ctypes.c_void_p, ctypes.c_long] +
                          [ctypes.c_void_p, ctypes.c_long] +
                          [ctypes.c_int] * 5 +
                          [ctypes.c_float, ctypes.c_int, ctypes.c_float,
                           ctypes.c_float, ctypes.c_void_p],
    "launch_gemm_fp8": [ctypes.c_void_p] * 7 + [ctypes.c_int] * 3 +
                       [ctypes.c_void_p],
    "launch_gemm_fp4w": [ctypes.c_void_p] * 6 + [ctypes.c_int] * 3 +
                        [ctypes.c_void_p],
    "launch_moe_route": [ctypes.c_void_p] * 3 + [ctypes.c_int] * 4 +
                        [ctypes.c_float] + [ctypes.c_void_p] * 4,
    "launch_moe_scale_add": [ctypes.c_void_p] * 3 +
                            [ctypes.c_long, ctypes.c_int, ctypes.c_int,
                             ctypes.c_void_p],
    # one-shot xGMI collectives (csrc/xgmi_comm.hip)
    "launch_xgmi_coll": [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                         ctypes.c_int, ctypes.c_int, ctypes.c_long,
                         ctypes.c_long, ctypes.c_int, ctypes.c_int,
                         ctypes.c_long, ctypes.c_void_p],
    "xc_alloc": [ctypes.c_long, ctypes.c_void_p],
    "xc_free": [ctypes.c_void_p],
    "xc_memset": [ctypes.c_void_p, ctypes.c_int, ctypes.c_long],
    "xc_h2d": [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_long],
    "xc_d2h": [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_long],
    "xc_ipc_handle": [ctypes.c_void_p, ctypes.c_void_p],
    "xc_ipc_open": [ctypes.c_void_p, ctypes.c_void_p],
    "xc_ipc_close": [ctypes.c_void_p],
}


def lib():
    global _LIB
    if _LIB is None:
        if not os.path.exists(_LIB_PATH):
            raise RuntimeError(
                f"HIP kernel library not built: {_LIB_PATH} missing. "
                f"Run `python csrc/build.py` (or __graft_entry__.build()).")
        _LIB = ctypes.CDLL(_LIB_PATH)
        for name, argtypes in _SIGS.items():
            fn = getattr(_LIB, name)
            fn.argtypes = argtypes
            fn.restype = ctypes.c_int  # hipError_t
    return _LIB


def _stream() -> int:
    return torch.cuda.current_stream().cuda_stream


def _check(err: int, name: str):
    if err != 0:
        raise RuntimeError(f"{name} failed: hipError_t={err}")


def _ptr(t) -> int:
    return 0 if t is None else t.data_ptr()


# ----------------------------------------------------------------------
# op wrappers (shapes validated here; kernels trust their args)
# ----------------------------------------------------------------------

STAGE_RAW, STAGE_NORM, STAGE_GLU, STAGE_NORM2 = 0, 1, 2, 3
# NORM_EMBED: W-input x is the embedding TABLE; the row at *x2 (device
# token id) is gathered, scaled by `escale` and normed — fuses the
# decode step's k_embed launch into the first QKV GEMV.  `res` receives
# the persisted h (residual stream).
STAGE_NORM_EMBED = 4


def gemv(W: torch.Tensor, x: torch.Tensor, y: torch.Tensor,
         res: torch.Tensor | None = None, softcap: float = 0.0,
         stage: int = 0, x2: torch.Tensor | None = None,
         g: torch.Tensor | None = None, act: int = 0, eps: float = 1e-5,
         nt: int = 1, rpw: int = 1, maxblocks: int = 0,
         g2: torch.Tensor | None = None, escale: float = 1.0,
         eidx: torch.Tensor | None = None, wstride: int = 0,
         oscale: torch.Tensor | None = None):
    """y[N] = W[N,K] @ stage(x)[K] (+res); stage fuses RMSNorm / GLU /
    the Gemma sandwich (NORM2: x2 = h_in, g/g2 = post/pre gammas, res =
    h_out ping-pong) / the embed gather (NORM_EMBED: x = embed table,
    x2 = token-id i32, res = persisted h) into the LDS staging pass;
    nt = non-temporal W."""
    N, K = W.shape
    out_f32 = 1 if y.dtype == torch.float32 else 0
    _check(lib().launch_gemv_bf16(
        _ptr(W), _ptr(x), _ptr(x2), _ptr(g), _ptr(g2), _ptr(y), _ptr(res),
        N, K, stage, act, ctypes.c_float(eps), out_f32,
        ctypes.c_float(softcap), nt, rpw, maxblocks,
        ctypes.c_float(escale), _ptr(eidx), ctypes.c_long(wstride),
        _ptr(oscale), _stream()), "gemv")


def gemv_fp8(Wq: torch.Tensor, scales: torch.Tensor, x: torch.Tensor,
             y: torch.Tensor, res: torch.Tensor | None = None,
             softcap: float = 0.0, stage: int = 0,
             x2: torch.Tensor | None = None, g: torch.Tensor | None = None,
             act: int = 0, eps: float = 1e-5, nt: int = 1, rpw: int = 1,
             maxblocks: int = 0, g2: torch.Tensor | None = None,
             escale: float = 1.0, eidx: torch.Tensor | None = None,
             wstride: int = 0, sstride: int = 0,
             oscale: torch.Tensor | None = None):
    """y[N] = scales * (Wq[N,K] @ stage(x)); Wq = e4m3fn bytes.
    eidx/wstride/sstride/oscale: MoE expert indexing — W and scales
    offset by the device int32 *eidx, output scaled by *oscale."""
    N, K = Wq.shape
    out_f32 = 1 if y.dtype == torch.float32 else 0
    _check(lib().launch_gemv_fp8(
        _ptr(Wq), _ptr(scales), _ptr(x), _ptr(x2), _ptr(g), _ptr(g2),
        _ptr(y), _ptr(res), N, K, stage, act, ctypes.c_float(eps), out_f32,
        ctypes.c_float(softcap), nt, rpw, maxblocks,
        ctypes.c_float(escale), _ptr(eidx), ctypes.c_long(wstride),
        ctypes.c_long(sstride), _ptr(oscale), _stream()), "gemv_fp8")


def rmsnorm(x: torch.Tensor, g: torch.Tensor, y: torch.Tensor,
            res: torch.Tensor | None = None, eps: float = 1e-5):
    """mode 0: y = norm(x)*g ; with res: y = res + norm(x)*g (Gemma post)."""
    M = 1 if x.dim() == 1 else x.shape[0]
    H = x.shape[-1]
    mode = 0 if res is None else 1
    _check(lib().launch_rmsnorm(
        _ptr(x), _ptr(g), _ptr(res), _ptr(y), M, H,
        ctypes.c_float(eps), mode, _stream()), "rmsnorm")


def rope_cache(q: torch.Tensor, k_in: torch.Tensor, v_in: torch.Tensor,
               k_cache: torch.Tensor, v_cache: torch.Tensor,
               cos_t: torch.Tensor, sin_t: torch.Tensor,
               pos_ptr: torch.Tensor, M: int, nh: int, kvh: int, hd: int,
               kS: torch.Tensor | None = None,
               vS: torch.Tensor | None = None):
    """kS/vS given => fp8 (e4m3) KV pool with per-(head,pos) scales."""
    S = k_cache.shape[1]
    _check(lib().launch_rope_cache(
        _ptr(q), _ptr(k_in), _ptr(v_in), _ptr(k_cache), _ptr(v_cache),
        _ptr(kS), _ptr(vS), 1 if kS is not None else 0,
        _ptr(cos_t), _ptr(sin_t), _ptr(pos_ptr), M, nh, kvh, hd, S,
        _stream()), "rope_cache")


def attn(q: torch.Tensor, k_cache: torch.Tensor, v_cache: torch.Tensor,
         out: torch.Tensor, len_ptr: torch.Tensor, M: int, nh: int,
         kvh: int, hd: int, scale: float, softcap: float = 0.0,
         window: int = 0, kS: torch.Tensor | None = None,
         vS: torch.Tensor | None = None):
    S = k_cache.shape[1]
    _check(lib().launch_attn(
        _ptr(q), _ptr(k_cache), _ptr(v_cache), _ptr(out), _ptr(len_ptr),
        _ptr(kS), _ptr(vS), 1 if kS is not None else 0,
        M, nh, kvh, hd, S, ctypes.c_float(scale), ctypes.c_float(softcap),
        window, _stream()), "attn")


def attn_prefill_mfma(q: torch.Tensor, k_cache: torch.Tensor,
                      v_cache: torch.Tensor, out: torch.Tensor,
                      len_ptr: torch.Tensor, M: int, nh: int, kvh: int,
                      hd: int, scale: float, softcap: float = 0.0,
                      window: int = 0, kS: torch.Tensor | None = None,
                      vS: torch.Tensor | None = None):
    """Flash prefill attention (MFMA, online softmax); q already roped,
    caches already written for [0, pos0+M)."""
    S = k_cache.shape[1]
    _check(lib().launch_attn_prefill_mfma(
        _ptr(q), _ptr(k_cache), _ptr(v_cache), _ptr(out), _ptr(len_ptr),
        _ptr(kS), _ptr(vS), 1 if kS is not None else 0,
        M, nh, kvh, hd, S, ctypes.c_float(scale), ctypes.c_float(softcap),
        window, _stream()), "attn_prefill_mfma")


def attn_dec(qkv: torch.Tensor, k_cache: torch.Tensor,
             v_cache: torch.Tensor, out: torch.Tensor,
             len_ptr: torch.Tensor, cos_t: torch.Tensor,
             sin_t: torch.Tensor, scratch: torch.Tensor,
             cnt: torch.Tensor, nh: int, kvh: int, hd: int,
             scale: float, softcap: float = 0.0, window: int = 0,
             split: int = 1, kS: torch.Tensor | None = None,
             vS: torch.Tensor | None = None, batch: int = 1):
    """Fused decode attention: RoPE(q,k) + KV write + online softmax,
    KV range split over `split` blocks/head (last-arriver merge).
    kS/vS given => fp8 KV pool (quantized write + dequant scan).
    batch>1: grid.z = lockstep sequence rows (per-b qkv/out rows, KV
    pools, scratch and tickets; shared device position)."""
    S = k_cache.shape[-2]
    _check(lib().launch_attn_dec(
        _ptr(qkv), _ptr(k_cache), _ptr(v_cache), _ptr(out), _ptr(len_ptr),
        _ptr(cos_t), _ptr(sin_t), _ptr(kS), _ptr(vS),
        1 if kS is not None else 0,
        _ptr(scratch), _ptr(cnt), split, batch,
        nh, kvh, hd, S, ctypes.c_float(scale),
        ctypes.c_float(softcap), window, _stream()), "attn_dec")


def glu(gate: torch.Tensor, up: torch.Tensor, out: torch.Tensor, act: int):
    """out = act(gate) * up; act 0 = SiLU, 1 = tanh-GELU."""
    total = gate.numel()
    assert total % 8 == 0
    _check(lib().launch_glu(_ptr(gate), _ptr(up), _ptr(out), total, act,
                            _stream()), "glu")


def embed(table: torch.Tensor, ids: torch.Tensor, out: torch.Tensor,
          M: int, scale: float = 1.0):
    H = table.shape[1]
    _check(lib().launch_embed(_ptr(table), _ptr(ids), _ptr(out), M, H,
                              ctypes.c_float(scale), _stream()), "embed")


def sample(logits: torch.Tensor, min_p: float, greedy: bool, seed: int,
           ctr: torch.Tensor, gmax: torch.Tensor, pick: torch.Tensor,
           next_token: torch.Tensor, out_ring: torch.Tensor,
           nout: torch.Tensor, len_ptr: torch.Tensor,
           bump_len: bool = True, temperature: float = 1.0,
           cnt: torch.Tensor | None = None, batch: int = 1):
    """min-p / greedy sampler: parallel max + Gumbel-argmax; the
    last-arriving pick block commits the winner (no 1-thread fin
    launch).  gmax/pick are u64 scratch (zeroed once; the commit path
    re-zeros); cnt is an i32 ticket counter (zeroed once, re-armed).
    Temperature scales device-side (min-p keep-set + Gumbel score), so
    the GPU fast path matches the CPU sample_token() semantics."""
    V = logits.shape[-1]
    lbf16 = 1 if logits.dtype == torch.bfloat16 else 0
    ring_stride = out_ring.shape[-1] if out_ring.dim() > 1 else 0
    if batch > 1:
        assert (gmax.numel() >= batch and pick.numel() >= batch
                and next_token.numel() >= batch and nout.numel() >= batch
                and cnt is not None and cnt.numel() >= batch)
    if cnt is None:
        global _SAMPLE_CNT
        try:
            _SAMPLE_CNT
        except NameError:
            _SAMPLE_CNT = {}
        key = logits.device
        if key not in _SAMPLE_CNT:
            _SAMPLE_CNT[key] = torch.zeros(1, dtype=torch.int32,
                                           device=logits.device)
        cnt = _SAMPLE_CNT[key]
    inv_temp = 1.0 / max(float(temperature), 1e-6)
    _check(lib().launch_sample(
        _ptr(logits), V, lbf16, batch, ctypes.c_long(ring_stride),
        ctypes.c_float(min_p), 1 if greedy else 0,
        ctypes.c_uint64(seed), ctypes.c_float(inv_temp),
        _ptr(ctr), _ptr(gmax), _ptr(pick), _ptr(cnt),
        _ptr(next_token), _ptr(out_ring), _ptr(nout), _ptr(len_ptr),
        1 if bump_len else 0, _stream()), "sample")


def gemm(X: torch.Tensor, W: torch.Tensor, Y: torch.Tensor,
         res: torch.Tensor | None = None,
         accbuf: torch.Tensor | None = None):
    """Y[M,N] = X[M,K] @ W[N,K]^T (+res), bf16, MFMA prefill path.
    With accbuf (fp32 scratch >= M*N) small-M launches split K over
    grid.z for chip fill."""
    M, K = X.shape
    N = W.shape[0]
    assert W.shape[1] == K and K % 64 == 0
    if accbuf is not None and accbuf.numel() < M * N:
        accbuf = None
    _check(lib().launch_gemm_bf16(_ptr(X), _ptr(W), _ptr(Y), _ptr(res),
                                  _ptr(accbuf), M, N, K, _stream()), "gemm")


def quant_fp8(x: torch.Tensor, q: torch.Tensor, s: torch.Tensor):
    """Per-row e4m3fn quantization on device: q = fp8(x / s_row),
    s_row = absmax/448.  x: (M,K) bf16 contiguous; q: >= M*K uint8;
    s: >= M fp32.  Used for weights at load and activations per GEMM."""
    M = 1 if x.dim() == 1 else x.shape[0]
    K = x.shape[-1]
    assert q.numel() >= M * K and s.numel() >= M
    _check(lib().launch_quant_fp8(_ptr(x), _ptr(q), _ptr(s), M, K,
                                  _stream()), "quant_fp8")


def quant_fp4(x: torch.Tensor, q: torch.Tensor, e: torch.Tensor):
    """MXFP4 (OCP MX) weight quantization on device: e2m1 nibbles packed
    2/byte + one e8m0 scale byte per 32 elements.  x: (M,K) bf16;
    q: >= M*K/2 uint8; e: >= M*K/32 uint8."""
    M = 1 if x.dim() == 1 else x.shape[0]
    K = x.shape[-1]
    assert q.numel() >= M * K // 2 and e.numel() >= M * K // 32
    _check(lib().launch_quant_fp4(_ptr(x), _ptr(q), _ptr(e), M, K,
                                  _stream()), "quant_fp4")


def gemv_fp4(Wq: torch.Tensor, We: torch.Tensor, x: torch.Tensor,
             y: torch.Tensor, res: torch.Tensor | None = None,
             softcap: float = 0.0, stage: int = 0,
             x2: torch.Tensor | None = None, g: torch.Tensor | None = None,
             act: int = 0, eps: float = 1e-5, nt: int = 1, rpw: int = 1,
             maxblocks: int = 0, g2: torch.Tensor | None = None,
             escale: float = 1.0):
    """y[N] = W4[N,K] @ stage(x): MXFP4 weights (Wq: N x K/2 packed
    bytes, We: N x K/32 e8m0 scale bytes) — half the fp8 stream."""
    N = Wq.shape[0]
    K = Wq.shape[1] * 2
    out_f32 = 1 if y.dtype == torch.float32 else 0
    _check(lib().launch_gemv_fp4(
        _ptr(Wq), _ptr(We), _ptr(x), _ptr(x2), _ptr(g), _ptr(g2),
        _ptr(y), _ptr(res), N, K, stage, act, ctypes.c_float(eps), out_f32,
        ctypes.c_float(softcap), nt, rpw, maxblocks,
        ctypes.c_float(escale), _stream()), "gemv_fp4")


def gemm_fp8(xq: torch.Tensor, sx: torch.Tensor, Wq: torch.Tensor,
             sw: torch.Tensor, y: torch.Tensor, M: int, K: int,
             res: torch.Tensor | None = None,
             accbuf: torch.Tensor | None = None):
    """Y[M,N] = (sx_m * sw_n) * Xq[M,K] @ Wq[N,K]^T (+res): fp8 MFMA
    prefill path (both operands e4m3 with per-row scales)."""
    N = Wq.shape[0]
    assert Wq.shape[-1] == K and K % 64 == 0
    if accbuf is not None and accbuf.numel() < M * N:
        accbuf = None
    _check(lib().launch_gemm_fp8(
        _ptr(xq), _ptr(sx), _ptr(Wq), _ptr(sw), _ptr(y), _ptr(res),
        _ptr(accbuf), M, N, K, _stream()), "gemm_fp8")


def gemv_fp8_mx(Wq: torch.Tensor, scales: torch.Tensor, x: torch.Tensor,
                y: torch.Tensor, B: int, xstride: int, ystride: int,
                res: torch.Tensor | None = None, rstride: int = 0,
                hout: torch.Tensor | None = None, hstride: int = 0,
                softcap: float = 0.0, stage: int = 0,
                x2: torch.Tensor | None = None, x2stride: int = 0,
                g: torch.Tensor | None = None,
                g2: torch.Tensor | None = None, act: int = 0,
                eps: float = 1e-5, escale: float = 1.0):
    """Multi-x fp8 GEMV: y[b, N] = scales * (Wq @ stage(x_b)) for B
    lockstep rows — one weight stream, B accumulators.  Strides are in
    ELEMENTS.  NORM_EMBED: x = embed table, x2 = int32 token ids."""
    N, K = Wq.shape
    out_f32 = 1 if y.dtype == torch.float32 else 0
    _check(lib().launch_gemv_fp8_mx(
        _ptr(Wq), _ptr(scales), _ptr(x), ctypes.c_long(xstride),
        _ptr(x2), ctypes.c_long(x2stride), _ptr(g), _ptr(g2),
        _ptr(y), ctypes.c_long(ystride), _ptr(res), ctypes.c_long(rstride),
        _ptr(hout), ctypes.c_long(hstride), N, K, B, stage, act,
        ctypes.c_float(eps), out_f32, ctypes.c_float(softcap),
        ctypes.c_float(escale), _stream()), "gemv_fp8_mx")


def stage_quant_mx(x: torch.Tensor, xstride: int, xq: torch.Tensor,
                   sx: torch.Tensor, B: int, K: int, stage: int = 0,
                   x2: torch.Tensor | None = None, x2stride: int = 0,
                   g: torch.Tensor | None = None,
                   g2: torch.Tensor | None = None,
                   hout: torch.Tensor | None = None, hstride: int = 0,
                   act: int = 0, eps: float = 1e-5, escale: float = 1.0):
    """Per-sequence-row staging op + e4m3 quantization -> xq[B,K], sx[B]
    (feeds the skinny fp8 MFMA GEMM).  Strides in elements."""
    _check(lib().launch_stage_quant_mx(
        _ptr(x), ctypes.c_long(xstride), _ptr(x2), ctypes.c_long(x2stride),
        _ptr(g), _ptr(g2), _ptr(xq), _ptr(sx), _ptr(hout),
        ctypes.c_long(hstride), K, B, stage, act, ctypes.c_float(eps),
        ctypes.c_float(escale), _stream()), "stage_quant_mx")


def gemm_fp8_skinny(xq: torch.Tensor, sx: torch.Tensor, Wq: torch.Tensor,
                    sw: torch.Tensor, y: torch.Tensor, B: int,
                    ystride: int, res: torch.Tensor | None = None,
                    rstride: int = 0, bias: torch.Tensor | None = None,
                    softcap: float = 0.0,
                    accbuf: torch.Tensor | None = None):
    """Y[B<=16, N] = (sx_b*sw_n) * Xq @ Wq^T: one 16-row fp8 MFMA tile
    per wave, W nt-streamed once (batched decode B=3..16).  accbuf
    (fp32, >= B*N) enables the K-split path for small-N occupancy."""
    N, K = Wq.shape
    out_f32 = 1 if y.dtype == torch.float32 else 0
    if accbuf is not None and accbuf.numel() < B * N:
        accbuf = None
    _check(lib().launch_gemm_fp8_skinny(
        _ptr(xq), _ptr(sx), _ptr(Wq), _ptr(sw), _ptr(y),
        ctypes.c_long(ystride), _ptr(res), ctypes.c_long(rstride),
        _ptr(bias), _ptr(accbuf), out_f32, ctypes.c_float(softcap),
        B, N, K, _stream()), "gemm_fp8_skinny")


def gemm_fp4w(X: torch.Tensor, Wq4: torch.Tensor, We4: torch.Tensor,
              Y: torch.Tensor, res: torch.Tensor | None = None,
              accbuf: torch.Tensor | None = None):
    """Y[M,N] = X[M,K] @ dequant(W4)[N,K]^T (+res): bf16-MFMA prefill
    over MXFP4 weights (single-copy fp4 engines; activations unquantized
    -> more accurate than the fp8 prefill path)."""
    M, K = X.shape
    N = Wq4.shape[0]
    assert Wq4.shape[1] * 2 == K and K % 64 == 0
    if accbuf is not None and accbuf.numel() < M * N:
        accbuf = None
    _check(lib().launch_gemm_fp4w(
        _ptr(X), _ptr(Wq4), _ptr(We4), _ptr(Y), _ptr(res), _ptr(accbuf),
        M, N, K, _stream()), "gemm_fp4w")


def prefetch(t: torch.Tensor, sink: torch.Tensor):
    """Stream t through the cache hierarchy (side-stream warmer)."""
    n = t.numel() * t.element_size() // 2  # treat as u16 elements
    _check(lib().launch_prefetch(_ptr(t), n, _ptr(sink), _stream()),
           "prefetch")


def softcap(y: torch.Tensor, cap: float):
    """in-place y = cap * tanh(y / cap) (bf16)."""
    total = y.numel()
    assert total % 8 == 0
    _check(lib().launch_softcap(_ptr(y), total, ctypes.c_float(cap),
                                _stream()), "softcap")


def bias_add(y: torch.Tensor, bias: torch.Tensor, M: int):
    """y[:M, :N] += bias[N] (bf16) — Qwen-2 qkv bias on GEMM outputs."""
    N = bias.numel()
    _check(lib().launch_bias_add(_ptr(y), _ptr(bias), M, N, _stream()),
           "bias_add")


def addinto(y: torch.Tensor, a: torch.Tensor):
    """y += a (bf16)."""
    total = y.numel()
    assert total % 8 == 0
    _check(lib().launch_addinto(_ptr(y), _ptr(a), total, _stream()), "addinto")


def i32_set(buf: torch.Tensor, v: int):
    _check(lib().launch_i32_set(_ptr(buf), v, _stream()), "i32_set")


# ----------------------------------------------------------------------
# one-shot xGMI collectives (csrc/xgmi_comm.hip) — raw-pointer plumbing
# for the peer-mapped comm buffers (outside torch's caching allocator so
# the IPC base pointer is stable)
# ----------------------------------------------------------------------

XC_MODE_AR_BF16, XC_MODE_AR_F32, XC_MODE_GATHER = 0, 1, 2
XC_OFF_EPOCH, XC_OFF_TICKET, XC_OFF_ERR = 64, 72, 76
XC_OFF_DATA = 4096
XC_IPC_HANDLE_BYTES = 64


def xc_alloc(nbytes: int) -> int:
    p = ctypes.c_void_p(0)
    _check(lib().xc_alloc(ctypes.c_long(nbytes), ctypes.byref(p)), "xc_alloc")
    return p.value


def xc_free(ptr: int):
    _check(lib().xc_free(ctypes.c_void_p(ptr)), "xc_free")


def xc_memset(ptr: int, val: int, nbytes: int):
    _check(lib().xc_memset(ctypes.c_void_p(ptr), val,
                           ctypes.c_long(nbytes)), "xc_memset")


def xc_h2d(dst: int, data: bytes):
    buf = (ctypes.c_char * len(data)).from_buffer_copy(data)
    _check(lib().xc_h2d(ctypes.c_void_p(dst), buf,
                        ctypes.c_long(len(data))), "xc_h2d")


def xc_d2h(src: int, nbytes: int) -> bytes:
    buf = (ctypes.c_char * nbytes)()
    _check(lib().xc_d2h(buf, ctypes.c_void_p(src),
                        ctypes.c_long(nbytes)), "xc_d2h")
    return bytes(buf)


def xc_ipc_handle(ptr: int) -> bytes:
    h = (ctypes.c_char * XC_IPC_HANDLE_BYTES)()
    _check(lib().xc_ipc_handle(ctypes.c_void_p(ptr), h), "xc_ipc_handle")
    return bytes(h)


def xc_ipc_open(handle: bytes) -> int:
    assert len(handle) == XC_IPC_HANDLE_BYTES
    buf = (ctypes.c_char * XC_IPC_HANDLE_BYTES).from_buffer_copy(handle)
    p = ctypes.c_void_p(0)
    _check(lib().xc_ipc_open(buf, ctypes.byref(p)), "xc_ipc_open")
    return p.value


def xc_ipc_close(ptr: int):
    _check(lib().xc_ipc_close(ctypes.c_void_p(ptr)), "xc_ipc_close")


def xgmi_coll(dst_ptr: int, src_ptr: int, mybase: int, rank: int,
              world: int, nbytes: int, slot_bytes: int, mode: int,
              nstripes: int, spin_limit: int = 5_000_000):
    """One one-shot collective on the current torch stream (capturable)."""
    _check(lib().launch_xgmi_coll(
        ctypes.c_void_p(dst_ptr), ctypes.c_void_p(src_ptr),
        ctypes.c_void_p(mybase), rank, world, ctypes.c_long(nbytes),
        ctypes.c_long(slot_bytes), mode, nstripes,
        ctypes.c_long(spin_limit), _stream()), "xgmi_coll")


def moe_route(h: torch.Tensor, g: torch.Tensor, wg: torch.Tensor, M: int,
              topk: int, idx: torch.Tensor, w: torch.Tensor,
              dense: torch.Tensor | None = None, eps: float = 1e-5):
    """Fused MoE router (Mixtral semantics): per row RMSNorm(h)*g ->
    E dots vs wg[E,H] (f32) -> softmax -> top-k renormalized.
    idx: (M*topk,) int32; w: (M*topk,) f32; dense: (M*E,) f32 per-expert
    weights (0 if unrouted) for the prefill expert-GEMM loop."""
    E, H = wg.shape
    assert idx.numel() >= M * topk and w.numel() >= M * topk
    if dense is not None:
        assert dense.numel() >= M * E
    _check(lib().launch_moe_route(
        _ptr(h), _ptr(g), _ptr(wg), M, H, E, topk, ctypes.c_float(eps),
        _ptr(idx), _ptr(w), _ptr(dense), _stream()), "moe_route")


def moe_scale_add(y: torch.Tensor, x: torch.Tensor, w: torch.Tensor,
                  wstride: int, M: int, H: int):
    """y[m, :H] += w[m * wstride] * x[m, :H] (bf16, f32 math)."""
    _check(lib().launch_moe_scale_add(
        _ptr(y), _ptr(x), _ptr(w), ctypes.c_long(wstride), M, H,
        _stream()), "moe_scale_add")
