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
            return ops.decode_gemm_v2(a, b)

        t_v1 = bench_one(v1, x, w)
        t_blas = bench_one(torch.nn.functional.linear, x, w)
        try:
            t_v2 = bench_one(v2, x, w)
            err = (v2(x, w).float()
                   - torch.nn.functional.linear(x, w).float()) \
                .abs().max().item()
        except ValueError:
            t_v2, err = float("inf"), -1
        wbytes = M * K * 2
        out[label] = {
            "shape": f"N{N} M{M} K{K}",
            "v1_us": round(t_v1 * 1e6, 1),
            "v2_us": round(t_v2 * 1e6, 1),
            "hipblaslt_us": round(t_blas * 1e6, 1),
            "v2_tbps": round(wbytes / t_v2 / 1e12, 2),
            "hipblaslt_tbps": round(wbytes / t_blas / 1e12, 2),
            "v2_speedup_vs_blas": round(t_blas / t_v2, 2),
            "v2_max_err": round(err, 4),
        }
        print(json.dumps({label: out[label]}), flush=True)
    print(json.dumps(out))


if __name__ == "__main__":
    main()
