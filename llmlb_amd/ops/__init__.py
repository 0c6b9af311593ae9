"""llmlb_amd.ops — MI355X (gfx950) HIP kernels with torch fp32 fallbacks.

The native library (``libllmlb_kernels.so``, built in-tree from
``csrc/kernels.hip`` by ``__graft_entry__.build()``) is loaded via ctypes
and called with raw device pointers on the current torch HIP stream.

Policy (required by the harness): on a GPU box the HIP path MUST run —
if a CUDA/HIP tensor reaches an op and the native library is missing,
we raise instead of silently falling back to eager PyTorch.  CPU tensors
always use the fp32 reference implementations (tests, CI without GPU).
"""
from __future__ import annotations

import ctypes
import os
from pathlib import Path

import torch

from . import reference

_LIB_NAME = "libllmlb_kernels.so"
_LIB_PATH = Path(__file__).resolve().parent / _LIB_NAME
_lib: ctypes.CDLL | None = None
_load_error: str | None = None


def _try_load() -> None:
    global _lib, _load_error
    if _lib is not None:
        return
    if not _LIB_PATH.exists():
        _load_error = f"{_LIB_PATH} not built (run __graft_entry__.build())"
        return
    try:
        _lib = ctypes.CDLL(str(_LIB_PATH))
    except OSError as e:  # e.g. no ROCm runtime on this host
        _lib = None
        _load_error = str(e)
        return
    _lib.ll_rmsnorm.restype = ctypes.c_int
    _lib.ll_rope.restype = ctypes.c_int
    _lib.ll_swiglu.restype = ctypes.c_int
    _lib.ll_kv_append.restype = ctypes.c_int
    _lib.ll_rope_kv_append.restype = ctypes.c_int
    _lib.ll_rmsnorm_q.restype = ctypes.c_int
    _lib.ll_swiglu_q.restype = ctypes.c_int
    _lib.ll_rows_q.restype = ctypes.c_int
    _lib.ll_decode_gemm_w8a8.restype = ctypes.c_int
    _lib.ll_decode_gemm_w4a8.restype = ctypes.c_int
    _lib.ll_decode_gemm_w8a8_ws_floats.restype = ctypes.c_longlong
    _lib.ll_paged_attn_decode.restype = ctypes.c_int
    _lib.ll_paged_attn_decode_fp8.restype = ctypes.c_int
    _lib.ll_kv_append_fp8.restype = ctypes.c_int
    _lib.ll_decode_gemm.restype = ctypes.c_int
    _lib.ll_decode_gemm_v2.restype = ctypes.c_int
    _lib.ll_decode_gemm_v3.restype = ctypes.c_int
    _lib.ll_decode_gemm_v3_ws_floats.restype = ctypes.c_longlong


_try_load()


def native_available() -> bool:
    return _lib is not None


def native_load_error() -> str | None:
    return _load_error


class NativeKernelsMissing(RuntimeError):
    pass


def _require_native() -> ctypes.CDLL:
    if _lib is None:
        raise NativeKernelsMissing(
            "HIP tensor reached llmlb_amd.ops but the native kernel library "
            f"is not loaded: {_load_error}. Refusing silent eager fallback "
            "on GPU — build with __graft_entry__.build()."
        )
    return _lib


def _stream() -> ctypes.c_void_p:
    return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)


def _ptr(t: torch.Tensor) -> ctypes.c_void_p:
    return ctypes.c_void_p(t.data_ptr())


def _check(rc: int, name: str) -> None:
    if rc != 0:
        raise RuntimeError(f"{name} failed with code {rc}")


# ---------------------------------------------------------------------------
# public ops
# ---------------------------------------------------------------------------

def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float,
            residual: torch.Tensor | None = None):
    """out = rmsnorm(x (+ residual)) * weight.

    Returns (out, residual'): when ``residual`` is given it is updated in
    place to x + residual and returned.  x: [T, H] bf16 (GPU path).
    """
    if not x.is_cuda or x.dtype != torch.bfloat16:
        # bf16 is the native serving dtype; fp32 GPU tensors (tests,
        # debugging) use the reference path
        return reference.rmsnorm(x, weight, eps, residual)
    lib = _require_native()
    assert x.is_contiguous()
    out = torch.empty_like(x)
    n_tokens = x.numel() // x.shape[-1]
    rc = lib.ll_rmsnorm(
        _ptr(out), _ptr(x), _ptr(weight),
        _ptr(residual) if residual is not None else None,
        ctypes.c_int(n_tokens), ctypes.c_int(x.shape[-1]),
        ctypes.c_float(eps), _stream())
    _check(rc, "ll_rmsnorm")
    return out, residual


def rope_apply(q: torch.Tensor, k: torch.Tensor, cos_sin: torch.Tensor,
               positions: torch.Tensor):
    """In-place rotate-half RoPE on q [T, n_q, D] and k [T, n_kv, D]."""
    if not q.is_cuda or q.dtype != torch.bfloat16:
        return reference.rope_apply(q, k, cos_sin, positions)
    lib = _require_native()
    # q/k may be strided views into the fused qkv projection (head dim
    # contiguous, head stride == D): no .contiguous() copies needed
    assert q.stride(-1) == 1 and q.stride(1) == q.shape[2]
    assert k.stride(-1) == 1 and k.stride(1) == k.shape[2]
    assert cos_sin.dtype == torch.float32
    pos = positions.to(torch.int32)
    rc = lib.ll_rope(
        _ptr(q), _ptr(k), _ptr(cos_sin), _ptr(pos),
        ctypes.c_int(q.shape[0]), ctypes.c_int(q.shape[1]),
        ctypes.c_int(k.shape[1]), ctypes.c_int(q.shape[2]),
        ctypes.c_int(q.stride(0)), ctypes.c_int(k.stride(0)), _stream())
    _check(rc, "ll_rope")
    return q, k


def rope_kv_append(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                   cos_sin: torch.Tensor, positions: torch.Tensor,
                   kcache: torch.Tensor, vcache: torch.Tensor,
                   slots: torch.Tensor) -> None:
    """Fused RoPE + paged-cache append: rotates q in place, rotates k
    in place AND into the K cache, copies v into the V cache — one
    launch instead of the rope + kv_append pair on the decode path.
    bf16 caches only (the fp8 path keeps the separate quantizing
    append)."""
    if (not q.is_cuda or q.dtype != torch.bfloat16
            or kcache.dtype != torch.bfloat16):
        # the reference rope returns NEW tensors; emulate the fused
        # kernel's in-place contract by copying back into the views
        q2, k2 = reference.rope_apply(q, k, cos_sin, positions)
        q.copy_(q2)
        k.copy_(k2)
        reference.kv_append(kcache, vcache, k, v, slots,
                            kcache.shape[2])
        return
    lib = _require_native()
    assert q.stride(-1) == 1 and q.stride(1) == q.shape[2]
    assert k.stride(-1) == 1 and k.stride(1) == k.shape[2]
    assert v.stride(-1) == 1 and v.stride(1) == v.shape[2]
    assert cos_sin.dtype == torch.float32
    pos = positions.to(torch.int32)
    slots_i = slots.to(torch.int32)
    rc = lib.ll_rope_kv_append(
        _ptr(q), _ptr(k), _ptr(v), _ptr(cos_sin), _ptr(pos),
        _ptr(kcache), _ptr(vcache), _ptr(slots_i),
        ctypes.c_int(q.shape[0]), ctypes.c_int(q.shape[1]),
        ctypes.c_int(k.shape[1]), ctypes.c_int(q.shape[2]),
        ctypes.c_int(kcache.shape[2]), ctypes.c_int(q.stride(0)),
        ctypes.c_int(k.stride(0)), ctypes.c_int(v.stride(0)), _stream())
    _check(rc, "ll_rope_kv_append")


def swiglu(gateup: torch.Tensor) -> torch.Tensor:
    """gateup [T, 2*inter] -> silu(gate) * up  [T, inter]."""
    if not gateup.is_cuda or gateup.dtype != torch.bfloat16:
        return reference.swiglu(gateup)
    lib = _require_native()
    assert gateup.is_contiguous()
    inter = gateup.shape[-1] // 2
    n_tokens = gateup.numel() // gateup.shape[-1]
    out = torch.empty(*gateup.shape[:-1], inter, dtype=gateup.dtype,
                      device=gateup.device)
    rc = lib.ll_swiglu(_ptr(out), _ptr(gateup),
                       ctypes.c_longlong(n_tokens), ctypes.c_int(inter),
                       _stream())
    _check(rc, "ll_swiglu")
    return out


def kv_append(kcache: torch.Tensor, vcache: torch.Tensor, k: torch.Tensor,
              v: torch.Tensor, slots: torch.Tensor,
              k_scale: torch.Tensor | None = None,
              v_scale: torch.Tensor | None = None) -> None:
    """Scatter k/v [T, n_kv, D] into paged caches at flat slot ids.
    uint8 caches are the FP8 (OCP e4m3) path: values are quantized with
    one scale per (page, head, token) written to k_scale/v_scale."""
    block_size = kcache.shape[2]
    if kcache.dtype == torch.uint8:
        if not kcache.is_cuda:
            reference.kv_append_fp8(kcache, vcache, k_scale, v_scale,
                                    k, v, slots, block_size)
            return
        lib = _require_native()
        assert k.stride(-1) == 1 and k.stride(1) == k.shape[2]
        slots_i = slots.to(torch.int32)
        rc = lib.ll_kv_append_fp8(
            _ptr(kcache), _ptr(vcache), _ptr(k_scale), _ptr(v_scale),
            _ptr(k), _ptr(v), _ptr(slots_i), ctypes.c_int(k.shape[0]),
            ctypes.c_int(k.shape[1]), ctypes.c_int(k.shape[2]),
            ctypes.c_int(block_size), ctypes.c_int(k.stride(0)),
            ctypes.c_int(v.stride(0)), _stream())
        _check(rc, "ll_kv_append_fp8")
        return
    if not kcache.is_cuda or kcache.dtype != torch.bfloat16:
        reference.kv_append(kcache, vcache, k, v, slots, block_size)
        return
    lib = _require_native()
    assert k.stride(-1) == 1 and k.stride(1) == k.shape[2]
    assert v.stride(-1) == 1 and v.stride(1) == v.shape[2]
    slots_i = slots.to(torch.int32)
    rc = lib.ll_kv_append(
        _ptr(kcache), _ptr(vcache), _ptr(k), _ptr(v), _ptr(slots_i),
        ctypes.c_int(k.shape[0]), ctypes.c_int(k.shape[1]),
        ctypes.c_int(k.shape[2]), ctypes.c_int(block_size),
        ctypes.c_int(k.stride(0)), ctypes.c_int(v.stride(0)), _stream())
    _check(rc, "ll_kv_append")


_USE_CUSTOM_DECODE_GEMM = os.environ.get(
    "LLMLB_CUSTOM_DECODE_GEMM", "auto").lower()


# split-K fp32 workspaces, reused per shape: allocated zeroed once;
# every kernel invocation's convert pass re-zeroes after reading, so
# replays (and hipGraph captures) carry no per-call memset.  Calls are
# stream-ordered on the engine's stream, so reuse is race-free.
_WS_CACHE: dict = {}


def _split_k_ws(N: int, M: int, device) -> torch.Tensor:
    key = (N, M, str(device))
    t = _WS_CACHE.get(key)
    if t is None:
        t = torch.zeros(N, M, dtype=torch.float32, device=device)
        _WS_CACHE[key] = t
    return t


def _v3_ws(N: int, M: int, K: int, device) -> torch.Tensor | None:
    """fp32 accumulator workspace for the v3 split-K atomic reduce
    (sized by the kernel library; the f32->bf16 convert pass re-zeroes
    it as it drains, so reuse needs no memset).  Both slab-reduce
    variants (last-arriver and every-slice spin) measured WORSE than
    atomics+convert on the GPU — see profiles/r02_decode_gemm_v3_ab.md."""
    n = int(_lib.ll_decode_gemm_v3_ws_floats(N, M, K))
    if n == 0:
        return None
    key = ("v3", N, M, K, str(device))
    t = _WS_CACHE.get(key)
    if t is None:
        t = torch.zeros(n, dtype=torch.float32, device=device)
        _WS_CACHE[key] = t
    return t


# staged decode-GEMM generation: "3" (default) = software-pipelined v3
# (double-buffered W register sets + LDS x buffers, nt weight stream);
# "2" = the r1 single-buffered kernel, kept for A/B.
_DECODE_GEMM_VER = os.environ.get("LLMLB_DECODE_GEMM_VER", "3")


def decode_gemm_v2(x: torch.Tensor, weight: torch.Tensor,
                   ws: torch.Tensor | None = None,
                   w_scale: torch.Tensor | None = None,
                   ver: str | None = None) -> torch.Tensor:
    """Staged-x decode GEMM (v3 pipelined by default; see
    _DECODE_GEMM_VER).  Raises on unsupported shapes.
    ``w_scale`` non-None = fp8-weight arm: ``weight`` holds e4m3 bytes
    and per-row scales fold into the store (half the weight traffic on
    the bandwidth-bound decode shapes)."""
    lib = _require_native()
    N, K = x.shape
    M = weight.shape[0]
    out = torch.empty(N, M, dtype=x.dtype, device=x.device)
    use_v3 = (ver or _DECODE_GEMM_VER) == "3"
    if ws is None:
        ws = (_v3_ws(N, M, K, x.device) if use_v3
              else (_split_k_ws(N, M, x.device)
                    if (M // 128) < 224 else None))
    entry = lib.ll_decode_gemm_v3 if use_v3 else lib.ll_decode_gemm_v2
    rc = entry(
        _ptr(out), _ptr(ws) if ws is not None else None, _ptr(x),
        _ptr(weight),
        _ptr(w_scale) if w_scale is not None else None,
        ctypes.c_int(N), ctypes.c_int(M), ctypes.c_int(K),
        ctypes.c_int(0), _stream())
    if rc == -2:
        raise ValueError("unsupported shape for decode_gemm_v2")
    _check(rc, "ll_decode_gemm_v2")
    return out


# EXPERIMENTAL: W8A16 measured SLOWER than bf16+hipBLASLt (the
# e4m3->bf16 converts sit on the MFMA operand path —
# profiles/r01_fp8_weights_ab.md).  Kept for A/B; default OFF.
# engine-maintained phase flag: the fp8-weight (W8A16) decode arm and
# similar decode-only dispatches must not trigger on small PREFILL
# fragments (a 16-token prompt chunk has T <= 128 too) — measured 4.3x
# prefill slowdown on llama-3-70b before the gate (profiles/r02_w8a8.md)
_DECODE_PHASE = False


def set_decode_phase(v: bool) -> None:
    global _DECODE_PHASE
    _DECODE_PHASE = bool(v)


_FP8_WEIGHTS = os.environ.get("LLMLB_FP8_WEIGHTS",
                              "0").lower() in ("1", "true", "on")

# W8A8: BOTH operands e4m3, consumed directly by the gfx950 fp8 MFMA
# (no convert chain); activations are quantized per-token INSIDE the
# producing kernels (rmsnorm_q / swiglu_q).  Implies fp8 weights.
# Opt-in: reduced model precision is a capacity/throughput trade the
# operator must choose (the bf16 default serves the BASELINE bench).
# MXFP4 weights under the A8 activation path (W4A8): block-scaled fp4
# via the f8f6f4 MFMA's scale operands.  Requires LLMLB_FP8_A8=1 too.
_FP4_WEIGHTS = os.environ.get("LLMLB_FP4_WEIGHTS",
                              "0").lower() in ("1", "true", "on")


def w4_active() -> bool:
    return _FP4_WEIGHTS and _lib is not None


# fp8 PREFILL GEMMs (torch._scaled_mm -> hipBLASLt fp8, rowwise
# scales): +86% whole-prefill on llama-3-8b and +76% on qwen2-7b at
# 8x2048 (profiles/r02_fp8_prefill_ab.md — read the measured arc
# before changing ANY of the policy pieces below).  The fp8-vs-bf16
# choice is tuned per (M, K, row-bucket) at ENGINE INIT
# (tune_prefill_shapes); the serving path is a pure table lookup.
# Auto-enabled with the A8 arm (same e4m3-activation numerics
# class); LLMLB_FP8_PREFILL=0 opts out.
_FP8_PREFILL = os.environ.get("LLMLB_FP8_PREFILL",
                              "1").lower() not in ("0", "false", "off")
# keyed by (M, K, pow2-bucket of T): hipBLASLt's rowwise-fp8 kernels
# are shape- and row-count-sensitive, so the winner is measured per
# row-count bucket (~15 ms per bucket, once, at engine init)
_PREFILL_CHOICE: dict[tuple[int, int, int], bool] = {}


def fp8_prefill_active() -> bool:
    return _FP8_A8 and _FP8_PREFILL and _lib is not None


# persistent per-K quantization workspaces for the prefill path: a
# fresh 100-700 MB xq allocation per GEMM call, interleaved with the
# chunk-2 math-path attention's ~13 GB/layer buffers, splits the
# caching allocator's large blocks and forces hipMalloc/free-sync
# churn (measured 2.3x whole-prefill regression at 8x6144).  One
# buffer per K, sliced to the call's row count, removes the churn.
_PREFILL_WS: dict[int, tuple[torch.Tensor, torch.Tensor]] = {}


def _rows_q_ws(x: torch.Tensor):
    T, K = x.shape
    ws = _PREFILL_WS.get(K)
    if ws is None or ws[0].shape[0] < T or ws[0].device != x.device:
        rows = max(T, MAX_FP8_PREFILL_ROWS)
        ws = (torch.empty(rows, K, dtype=torch.uint8, device=x.device),
              torch.empty(rows, dtype=torch.float32, device=x.device))
        _PREFILL_WS[K] = ws
    q, sc = ws[0][:T], ws[1][:T]
    lib = _require_native()
    rc = lib.ll_rows_q(_ptr(q), _ptr(sc), _ptr(x), ctypes.c_int(T),
                       ctypes.c_int(K), _stream())
    _check(rc, "ll_rows_q")
    return q, sc


def _scaled_mm_raw(x: torch.Tensor, wq, wsc) -> torch.Tensor:
    xq, xs = _rows_q_ws(x.contiguous())
    return torch._scaled_mm(
        xq.view(torch.float8_e4m3fn),
        wq.view(torch.float8_e4m3fn).t(),
        scale_a=xs.unsqueeze(1), scale_b=wsc.unsqueeze(0),
        out_dtype=torch.bfloat16)


def _tune_prefill(x: torch.Tensor, weight, wq, wsc) -> bool:
    """Time the fp8 path against the bf16 GEMM on this exact call's
    shapes; runs ONCE per (M, K), ~15 ms inside the first long
    prefill of that shape.

    The 256 MB MALL (Infinity Cache) is EVICTED between timed
    iterations: a hot loop re-reads the same weights from cache, which
    erases exactly the HBM-bandwidth advantage the fp8 weight stream
    exists for (measured: the hot-loop tuner picked bf16 on llama-3-8b
    where the cold-stream fp8 path is +16.7% whole-prefill)."""
    import torch.nn.functional as F

    try:
        # > 256 MB MALL + L2; transient, freed on return
        evict = torch.empty(320 * 1024 * 1024, dtype=torch.uint8,
                            device=x.device)
    except torch.cuda.OutOfMemoryError:
        return False  # memory-tight run: keep bf16 (no extra traffic)

    def timed(fn, iters=3):
        es = [torch.cuda.Event(enable_timing=True)
              for _ in range(2 * iters)]
        for i in range(iters):
            evict.fill_(i)          # thrash MALL + L2
            es[2 * i].record()
            fn()
            es[2 * i + 1].record()
        torch.cuda.synchronize(x.device)
        return sum(es[2 * i].elapsed_time(es[2 * i + 1])
                   for i in range(iters))

    # warmup both (hipBLASLt algorithm selection happens here)
    _scaled_mm_raw(x, wq, wsc)
    F.linear(x, weight)
    torch.cuda.synchronize(x.device)
    t_fp8 = timed(lambda: _scaled_mm_raw(x, wq, wsc))
    t_bf16 = timed(lambda: F.linear(x, weight))
    return t_fp8 < t_bf16


TUNE_BUCKETS = (1024, 4096, 16384)

# fp8 prefill row cap: 32k-row chunks regressed in-engine while the
# fp8 GEMMs win in isolation at that size; the regression was later
# traced to the chunk-2 math-attention interleaving (now gated by
# _PREFILL_FP8_OK), so this cap is conservative and unproven — first
# thing to re-measure next round (profiles/r02_fp8_prefill_ab.md).
# <= 16384-row calls measured +74-84% at 8x2048.
MAX_FP8_PREFILL_ROWS = 16384

# fp8 prefill applies only to P==0 chunks (first/only chunk of a
# prefill group).  Continuation chunks (P>0: cached prefix or chunk 2+
# of a long prompt) run the mask-path SDPA whose fp32 math fallback
# dominates the step AND loses 30%+ wall time when the fp8 GEMMs are
# interleaved with its ~13 GB/layer buffers (same-box A/B at 8x6144);
# single-chunk prefills measured +74-84% (profiles/r02_fp8_prefill_ab
# .md).  The engine toggles this around each chunk.
_PREFILL_FP8_OK = True


def set_prefill_fp8_ok(v: bool) -> None:
    global _PREFILL_FP8_OK
    _PREFILL_FP8_OK = bool(v)


def tune_prefill_shapes(pairs, device) -> int:
    """Tune the fp8-vs-bf16 prefill choice for every (weight, w8) pair
    at the representative row counts in ``TUNE_BUCKETS``, filling
    ``_PREFILL_CHOICE``.  Called ONCE at engine init: tuning inline
    during a live prefill was measured to cost ~2 s — each of its
    ``torch.cuda.synchronize`` calls drains the host's 32-layer
    launch-ahead queue mid-prefill (profiles/r02_fp8_prefill_ab.md).
    Returns the number of (shape, bucket) decisions recorded."""
    global _FP8_PREFILL
    done = set()
    n = 0
    for weight, (wq, wsc) in pairs:
        M, K = wq.shape[0], weight.shape[1]
        if (M, K) in done or K % 16 != 0 or M % 16 != 0:
            continue
        done.add((M, K))
        for T in TUNE_BUCKETS:
            key = (M, K, T.bit_length())
            if key in _PREFILL_CHOICE:
                n += 1
                continue
            x = torch.randn(T, K, dtype=torch.bfloat16, device=device)
            try:
                _PREFILL_CHOICE[key] = _tune_prefill(x, weight, wq, wsc)
                n += 1
            except Exception:
                _FP8_PREFILL = False
                return n
            finally:
                del x
    return n


def scaled_mm_prefill(x: torch.Tensor, w8,
                      weight: torch.Tensor | None = None
                      ) -> torch.Tensor | None:
    """Rowwise-scaled fp8 prefill GEMM: quantize x per token (rows_q),
    reuse the per-row fp8 weight copies, dispatch hipBLASLt fp8 via
    torch._scaled_mm.  Pure LOOKUP against the init-time tuning table
    (nearest row-count bucket); returns None when inapplicable, when
    the tuner picked bf16 for this shape, or when the shape was never
    tuned — never measures (and never syncs) on the serving path."""
    global _FP8_PREFILL
    if w8 is None:
        return None
    wq, wsc = w8
    T, K = x.shape
    M = wq.shape[0]
    if K % 16 != 0 or M % 16 != 0 or T < 512 \
            or T > MAX_FP8_PREFILL_ROWS or not _PREFILL_FP8_OK:
        return None
    tb = max(T, 1).bit_length()
    use = _PREFILL_CHOICE.get((M, K, tb))
    if use is None:
        cands = [k for k in _PREFILL_CHOICE if k[0] == M and k[1] == K]
        if not cands:
            return None
        use = _PREFILL_CHOICE[min(cands, key=lambda k: abs(k[2] - tb))]
    if not use:
        return None
    try:
        return _scaled_mm_raw(x, wq, wsc)
    except Exception:  # unsupported torch build: disable for the run
        _FP8_PREFILL = False
        return None


_FP8_A8 = os.environ.get("LLMLB_FP8_A8",
                         "0").lower() in ("1", "true", "on")


def w8a8_active() -> bool:
    return _FP8_A8 and _lib is not None


def rmsnorm_q(x: torch.Tensor, weight: torch.Tensor, eps: float,
              residual: torch.Tensor | None = None):
    """Fused rmsnorm + per-token e4m3 quantization: returns
    (q u8 [T, H], scale f32 [T], residual').  GPU bf16 only; the CPU
    reference path normalizes then simulates e4m3 via torch float8."""
    if not x.is_cuda or x.dtype != torch.bfloat16:
        out, residual = reference.rmsnorm(x, weight, eps, residual)
        q, sc = reference.quantize_rows_fp8(out)
        return q, sc, residual
    lib = _require_native()
    T, H = x.shape[0], x.shape[-1]
    q = torch.empty(T, H, dtype=torch.uint8, device=x.device)
    sc = torch.empty(T, dtype=torch.float32, device=x.device)
    rc = lib.ll_rmsnorm_q(
        _ptr(q), _ptr(sc), _ptr(x), _ptr(weight),
        _ptr(residual) if residual is not None else None,
        ctypes.c_int(T), ctypes.c_int(H), ctypes.c_float(eps), _stream())
    _check(rc, "ll_rmsnorm_q")
    return q, sc, residual


def swiglu_q(gateup: torch.Tensor):
    """Fused silu(gate)*up + per-token e4m3 quantization: returns
    (q u8 [T, inter], scale f32 [T])."""
    if not gateup.is_cuda or gateup.dtype != torch.bfloat16:
        act = reference.swiglu(gateup)
        return reference.quantize_rows_fp8(act)
    lib = _require_native()
    T = gateup.shape[0]
    inter = gateup.shape[-1] // 2
    q = torch.empty(T, inter, dtype=torch.uint8, device=gateup.device)
    sc = torch.empty(T, dtype=torch.float32, device=gateup.device)
    rc = lib.ll_swiglu_q(_ptr(q), _ptr(sc), _ptr(gateup),
                         ctypes.c_int(T), ctypes.c_int(inter), _stream())
    _check(rc, "ll_swiglu_q")
    return q, sc


def rows_q(x: torch.Tensor):
    """Plain per-row e4m3 quantization (amax/448): returns
    (q u8 [T, H], scale f32 [T]).  Quantizes the attention output so
    o_proj can take the quantized decode GEMMs."""
    if not x.is_cuda:
        q, sc = reference.quantize_rows_fp8(x)
        return q, sc
    lib = _require_native()
    T, H = x.shape
    q = torch.empty(T, H, dtype=torch.uint8, device=x.device)
    sc = torch.empty(T, dtype=torch.float32, device=x.device)
    rc = lib.ll_rows_q(_ptr(q), _ptr(sc), _ptr(x), ctypes.c_int(T),
                       ctypes.c_int(H), _stream())
    _check(rc, "ll_rows_q")
    return q, sc


def decode_gemm_w8a8(xq: torch.Tensor, x_scale: torch.Tensor,
                     wq: torch.Tensor,
                     w_scale: torch.Tensor) -> torch.Tensor:
    """out[r][c] = x_scale[r] * w_scale[c] * (q_x . q_w) via the fp8
    MFMA.  Raises ValueError on unsupported shapes (caller falls back
    to the bf16 path)."""
    lib = _require_native()
    N, K = xq.shape
    M = wq.shape[0]
    out = torch.empty(N, M, dtype=torch.bfloat16, device=xq.device)
    # kernel tiles cap at 128 rows; larger decode buckets (bs 256)
    # run as two row-halves into the same output (weights stream
    # twice, but fp8/fp4 still totals less than one bf16 pass)
    for lo in range(0, N, 128):
        hi = min(lo + 128, N)
        n = hi - lo
        nws = int(lib.ll_decode_gemm_w8a8_ws_floats(n, M, K))
        ws = None
        if nws:
            key = ("a8", n, M, K, str(xq.device))
            ws = _WS_CACHE.get(key)
            if ws is None:
                ws = torch.zeros(nws, dtype=torch.float32,
                                 device=xq.device)
                _WS_CACHE[key] = ws
        rc = lib.ll_decode_gemm_w8a8(
            _ptr(out[lo:hi]), _ptr(ws) if ws is not None else None,
            _ptr(xq[lo:hi]), ctypes.c_void_p(x_scale.data_ptr()
                                             + lo * 4),
            _ptr(wq), _ptr(w_scale), ctypes.c_int(n),
            ctypes.c_int(M), ctypes.c_int(K), _stream())
        if rc == -2:
            raise ValueError("unsupported shape for decode_gemm_w8a8")
        _check(rc, "ll_decode_gemm_w8a8")
    return out


def quantize_fp4_weight(w: torch.Tensor):
    """MXFP4 weight quantization (kernel register layout); see
    reference.quantize_fp4_weight."""
    return reference.quantize_fp4_weight(w)


def decode_gemm_w4a8(xq: torch.Tensor, x_scale: torch.Tensor,
                     w4: torch.Tensor, w4_scales: torch.Tensor,
                     M: int) -> torch.Tensor:
    """out[r][c] = x_scale[r] * (q_x . deq_mxfp4(w4)) via the
    block-scaled fp8 x fp4 MFMA (per-32-value E8M0 scales applied by
    the matrix core).  Raises ValueError on unsupported shapes."""
    lib = _require_native()
    N, K = xq.shape
    out = torch.empty(N, M, dtype=torch.bfloat16, device=xq.device)
    for lo in range(0, N, 128):     # see decode_gemm_w8a8
        hi = min(lo + 128, N)
        n = hi - lo
        nws = int(lib.ll_decode_gemm_w8a8_ws_floats(n, M, K))
        ws = None
        if nws:
            key = ("a8", n, M, K, str(xq.device))
            ws = _WS_CACHE.get(key)
            if ws is None:
                ws = torch.zeros(nws, dtype=torch.float32,
                                 device=xq.device)
                _WS_CACHE[key] = ws
        rc = lib.ll_decode_gemm_w4a8(
            _ptr(out[lo:hi]), _ptr(ws) if ws is not None else None,
            _ptr(xq[lo:hi]), ctypes.c_void_p(x_scale.data_ptr()
                                             + lo * 4),
            _ptr(w4), _ptr(w4_scales), ctypes.c_int(n),
            ctypes.c_int(M), ctypes.c_int(K), _stream())
        if rc == -2:
            raise ValueError("unsupported shape for decode_gemm_w4a8")
        _check(rc, "ll_decode_gemm_w4a8")
    return out


def quantize_fp8_weight(w: torch.Tensor):
    """Per-output-row e4m3 quantization of a [M, K] weight: returns
    (bytes [M, K] u8, scale [M] f32) with scale = row amax / 448."""
    amax = w.float().abs().amax(dim=1, keepdim=True)
    scale = torch.where(amax > 0, amax / 448.0, torch.ones_like(amax))
    q = (w.float() / scale).to(torch.float8_e4m3fn)
    return q.view(torch.uint8).contiguous(), scale.squeeze(1).contiguous()


def linear_decode(x: torch.Tensor, weight: torch.Tensor,
                  w8: tuple | None = None) -> torch.Tensor:
    """out = x @ W^T.

    Dispatch policy (measured IN-ENGINE, profiles/r01_attn_lds_and_
    gemm_policy.md): isolated microbenches had v2 winning the
    down-projection, but inside the captured decode step hipBLASLt is
    ~3% faster end-to-end at bs64 (cache interactions differ), so auto
    keeps v2 only for tiny batches (N <= 16) where its advantage held.
    LLMLB_CUSTOM_DECODE_GEMM: "auto" (default) = this policy,
    "1"/"on" = v2 everywhere it is legal, "0"/"off" = hipBLASLt only."""
    mode = _USE_CUSTOM_DECODE_GEMM
    if (mode in ("0", "false", "off") or not x.is_cuda or _lib is None
            or x.dtype != torch.bfloat16 or x.dim() != 2):
        return torch.nn.functional.linear(x, weight)
    N, K = x.shape
    M = weight.shape[0]
    shape_ok = N <= 128 and K % 256 == 0 and M % 128 == 0 and K >= 4096
    if w8 is not None and shape_ok and _DECODE_PHASE:
        # fp8 weights halve the dominant traffic: always take them
        try:
            return decode_gemm_v2(x, w8[0], w_scale=w8[1])
        except ValueError:
            pass
    # measured policy (profiles/r02_decode_gemm_v3_ab.md): the
    # pipelined v3 beats hipBLASLt 15-20% in ISOLATED probes on
    # o_proj/down, and the rocprof trace shows its kernels ~4% faster
    # in-engine too — but every split-K reduction scheme tried (fp32
    # atomics + convert pass, last-arriver slab, every-slice spin)
    # costs more than that margin inside the captured decode step
    # (best custom arm 5.32 vs blas 5.25 ms/step at bs64).  auto
    # therefore keeps hipBLASLt at N>16 and the v1-style fragment
    # kernel at tiny N where it wins outright; explicit modes force
    # the staged kernels for experiments.
    if shape_ok:
        try:
            if N <= 16:
                return decode_gemm_v2(x, weight, ver="2")
            if mode != "auto":
                return decode_gemm_v2(x, weight)
        except ValueError:
            pass
    return torch.nn.functional.linear(x, weight)


# fusing swiglu into the staging measured ~4% SLOWER end-to-end (the
# per-M-block silu recompute sits on the single-buffered staging
# critical path; profiles/r01_fused_swiglu_ab.md) — default OFF
_FUSED_SWIGLU = os.environ.get("LLMLB_FUSED_SWIGLU",
                               "0").lower() in ("1", "true", "on")


def swiglu_linear(gateup: torch.Tensor, weight: torch.Tensor):
    """Fused silu(gate)*up @ W^T: the SwiGLU epilogue is applied while
    staging x chunks into LDS inside the v2 decode GEMM, removing the
    standalone swiglu kernel and the activation round trip.  Returns
    None when the fused path does not apply (caller falls back)."""
    if (not _FUSED_SWIGLU
            or _USE_CUSTOM_DECODE_GEMM in ("0", "false", "off")
            or not gateup.is_cuda or _lib is None
            or gateup.dtype != torch.bfloat16 or gateup.dim() != 2
            or not gateup.is_contiguous()):
        return None
    N, K2 = gateup.shape
    K = K2 // 2
    M = weight.shape[0]
    if weight.shape[1] != K:
        return None
    # same measured policy as linear_decode's down-projection arm
    if not (N <= 128 and K % 256 == 0 and M % 128 == 0 and K >= 8192
            and (M // 128) < 224):
        return None
    lib = _require_native()
    out = torch.empty(N, M, dtype=gateup.dtype, device=gateup.device)
    use_v3 = _DECODE_GEMM_VER == "3"
    ws = (_v3_ws(N, M, K, gateup.device) if use_v3
          else (_split_k_ws(N, M, gateup.device)
                if (M // 128) < 224 else None))
    entry = (lib.ll_decode_gemm_v3 if use_v3
             else lib.ll_decode_gemm_v2)
    rc = entry(
        _ptr(out), _ptr(ws) if ws is not None else None, _ptr(gateup),
        _ptr(weight), None,
        ctypes.c_int(N), ctypes.c_int(M), ctypes.c_int(K),
        ctypes.c_int(1), _stream())
    if rc == -2:
        return None
    _check(rc, "ll_decode_gemm_v2(swiglu)")
    return out


_PAGED_ATTN_V2 = os.environ.get("LLMLB_PAGED_ATTN_V2",
                                "1").lower() not in ("0", "false", "off")


def paged_attn_decode(q: torch.Tensor, kcache: torch.Tensor,
                      vcache: torch.Tensor, block_tables: torch.Tensor,
                      seq_lens: torch.Tensor, scale: float,
                      n_parts: int | None = None,
                      workspace: tuple | None = None,
                      k_scale: torch.Tensor | None = None,
                      v_scale: torch.Tensor | None = None
                      ) -> torch.Tensor:
    """Decode attention over the paged KV cache.

    q: [S, n_q, D] bf16; caches [num_blocks, n_kv, block_size, D].
    ``n_parts`` splits long sequences across workgroups (flash-decode);
    chosen automatically to fill the 256-CU chip when not given.
    """
    if not q.is_cuda or q.dtype != torch.bfloat16:
        if kcache.dtype == torch.uint8:
            kd, vd = reference.dequant_fp8_cache(kcache, k_scale), \
                reference.dequant_fp8_cache(vcache, v_scale)
            return reference.paged_attn_decode(q, kd, vd, block_tables,
                                               seq_lens, scale)
        return reference.paged_attn_decode(q, kcache, vcache, block_tables,
                                           seq_lens, scale)
    lib = _require_native()
    S, n_q, D = q.shape
    n_kv = kcache.shape[1]
    block_size = kcache.shape[2]
    if n_parts is None:
        # fill ≫256 workgroups on the 256-CU chip (8 XCDs)
        base = S * n_kv
        n_parts = 1
        while base * n_parts < 512 and n_parts < 16:
            n_parts *= 2
    assert q.stride(-1) == 1 and q.stride(1) == D
    out = torch.empty(S, n_q, D, dtype=q.dtype, device=q.device)
    group = n_q // n_kv
    # v2 (GQA-amortized) always reduces through the slice workspace
    use_v2 = _PAGED_ATTN_V2 and D == 128 and group <= 16
    slices = n_parts * 4 if use_v2 else n_parts
    if n_parts > 1:
        if workspace is not None:
            ws_acc, ws_ml = workspace
        else:
            ws_acc = torch.empty(S, n_q, slices, D, dtype=torch.float32,
                                 device=q.device)
            ws_ml = torch.empty(S, n_q, slices, 2, dtype=torch.float32,
                                device=q.device)
        acc_p, ml_p = _ptr(ws_acc), _ptr(ws_ml)
    else:
        acc_p = ml_p = None
    bt = block_tables if block_tables.dtype == torch.int32 \
        else block_tables.to(torch.int32)
    sl = seq_lens if seq_lens.dtype == torch.int32 \
        else seq_lens.to(torch.int32)
    if kcache.dtype == torch.uint8:
        assert use_v2, "FP8 KV requires the v2 attention shape " \
            "(D=128, GQA group <= 16)"
        rc = lib.ll_paged_attn_decode_fp8(
            _ptr(out), acc_p, ml_p, _ptr(q), _ptr(kcache), _ptr(vcache),
            _ptr(k_scale), _ptr(v_scale), _ptr(bt), _ptr(sl),
            ctypes.c_int(S), ctypes.c_int(bt.shape[1]),
            ctypes.c_int(n_q), ctypes.c_int(n_kv), ctypes.c_int(D),
            ctypes.c_int(block_size), ctypes.c_int(n_parts),
            ctypes.c_float(scale), ctypes.c_int(q.stride(0)), _stream())
        _check(rc, "ll_paged_attn_decode_fp8")
        return out
    rc = lib.ll_paged_attn_decode(
        _ptr(out), acc_p, ml_p, _ptr(q), _ptr(kcache), _ptr(vcache),
        _ptr(bt), _ptr(sl), ctypes.c_int(S), ctypes.c_int(bt.shape[1]),
        ctypes.c_int(n_q), ctypes.c_int(n_kv), ctypes.c_int(D),
        ctypes.c_int(block_size), ctypes.c_int(n_parts),
        ctypes.c_float(scale), ctypes.c_int(1 if use_v2 else 0),
        ctypes.c_int(q.stride(0)), _stream())
    _check(rc, "ll_paged_attn_decode")
    return out
