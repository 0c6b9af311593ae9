"""Decode-GEMM microbench: custom weight-streaming kernel vs hipBLASLt
(torch.nn.functional.linear) on the llama-3-8b decode shapes.

Within-probe interleaved A/B (guide §5.4 rule 24): N rounds alternating
both paths in one process, report median ms and effective weight TB/s.
"""
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch

import os
os.environ["LLMLB_CUSTOM_DECODE_GEMM"] = "1"
import llmlb_amd.ops as ops
ops._USE_CUSTOM_DECODE_GEMM = "1"

SHAPES = [  # (N, M, K, label)
    (64, 6144, 4096, "qkv"),
    (64, 4096, 4096, "o_proj"),
    (64, 28672, 4096, "gate_up"),
    (64, 4096, 14336, "down"),
    (128, 28672, 4096, "gate_up_n128"),
    (16, 28672, 4096, "gate_up_n16"),
]
ROUNDS = 30


def bench_one(fn, x, w, rounds=ROUNDS):
    for _ in range(5):
        fn(x, w)
    torch.cuda.synchronize()
    times = []
    for _ in range(rounds):
        t0 = time.perf_counter()
        fn(x, w)
        torch.cuda.synchronize()
        times.append(time.perf_counter() - t0)
    times.sort()
    return times[len(times) // 2]


def main():
    assert torch.cuda.is_available() and ops.native_available(), \
        ops.native_load_error()
    dev = "cuda:0"
    out = {}
    for N, M, K, label in SHAPES:
        x = torch.randn(N, K, dtype=torch.bfloat16, device=dev) * 0.3
        w = torch.randn(M, K, dtype=torch.bfloat16, device=dev) * 0.05
        # interleaved A/B (v1, v2, hipBLASLt)
        def v1(a, b):
            import ctypes as _ct
            N2, K2 = a.shape
            M2 = b.shape[0]
            o = torch.empty(N2, M2, dtype=a.dtype, device=a.device)
            ops._lib.ll_decode_gemm(
                ops._ptr(o), ops._ptr(a), ops._ptr(b), _ct.c_int(N2),
                _ct.c_int(M2), _ct.c_int(K2), ops._stream())
            return o

        def v2(a, b):
            return ops.decode_gemm_v2(a, b, ver="2")

        def v3(a, b):
            return ops.decode_gemm_v2(a, b, ver="3")

        w8, w8s = ops.quantize_fp8_weight(w)

        def v3_fp8(a, b):
            return ops.decode_gemm_v2(a, w8, w_scale=w8s, ver="3")

        from llmlb_amd.ops import reference as _R
        xq, xs = _R.quantize_rows_fp8(x)
        xq, xs = xq.to(x.device), xs.to(x.device)

        def w8a8(a, b):
            return ops.decode_gemm_w8a8(xq, xs, w8, w8s)

        w4, w4s = ops.quantize_fp4_weight(w)
        w4, w4s = w4.to(x.device), w4s.to(x.device)

        def w4a8(a, b):
            return ops.decode_gemm_w4a8(xq, xs, w4, w4s, M)

        ref = torch.nn.functional.linear(x, w).float()
        t_blas = bench_one(torch.nn.functional.linear, x, w)
        res = {"shape": f"N{N} M{M} K{K}",
               "hipblaslt_us": round(t_blas * 1e6, 1)}
        wbytes = M * K * 2
        res["hipblaslt_tbps"] = round(wbytes / t_blas / 1e12, 2)
        for name, fn, wb in (("v2", v2, wbytes), ("v3", v3, wbytes),
                             ("v3_fp8", v3_fp8, wbytes // 2),
                             ("w8a8", w8a8, wbytes // 2),
                             ("w4a8", w4a8, wbytes // 4)):
            try:
                t = bench_one(fn, x, w)
                d = (fn(x, w).float() - ref).abs()
                rel = (d / ref.abs().clamp(min=1e-3)).max().item()
            except ValueError:
                t, rel = float("inf"), -1
            res[f"{name}_us"] = round(t * 1e6, 1)
            res[f"{name}_tbps"] = round(wb / t / 1e12, 2)
            res[f"{name}_vs_blas"] = round(t_blas / t, 2)
            res[f"{name}_rel_err"] = round(rel, 4)
        out[label] = res
        print(json.dumps({label: res}), flush=True)
    # fused-swiglu arm on the down shape (gateup [N, 2K] -> [N, M])
    for N, M, K, label in [(64, 4096, 14336, "down_fused_swiglu")]:
        gu = torch.randn(N, 2 * K, dtype=torch.bfloat16, device=dev) * 0.3
        w = torch.randn(M, K, dtype=torch.bfloat16, device=dev) * 0.05

        def fused(a, b):
            o = ops.swiglu_linear(a, b)
            assert o is not None
            return o

        def unfused(a, b):
            g, u = a[:, :K], a[:, K:]
            act = torch.nn.functional.silu(g) * u
            return torch.nn.functional.linear(act, b)

        os.environ["LLMLB_FUSED_SWIGLU"] = "1"
        ops._FUSED_SWIGLU = True
        ref = unfused(gu, w).float()
        try:
            t_f = bench_one(fused, gu, w)
            rel = ((fused(gu, w).float() - ref).abs()
                   / ref.abs().clamp(min=1e-3)).max().item()
        except (ValueError, AssertionError):
            t_f, rel = float("inf"), -1
        t_u = bench_one(unfused, gu, w)
        out[label] = {"fused_us": round(t_f * 1e6, 1),
                      "unfused_us": round(t_u * 1e6, 1),
                      "fused_vs_unfused": round(t_u / t_f, 2),
                      "rel_err": round(rel, 4)}
        print(json.dumps({label: out[label]}), flush=True)
    print(json.dumps(out))


if __name__ == "__main__":
    main()
