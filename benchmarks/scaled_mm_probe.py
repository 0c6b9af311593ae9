"""Probe torch._scaled_mm (hipBLASLt fp8) at PREFILL shapes: is an
fp8 prefill GEMM worth wiring into the quantized serving modes?

Compares bf16 F.linear vs _scaled_mm with per-tensor scales on the
llama-3-8b prefill shapes at T=1024/4096 chunk sizes.
"""
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch


def bench(fn, reps=20):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps


def main():
    dev = "cuda:0"
    out = {}
    for T in (1024, 4096):
        for M, K, label in ((6144, 4096, "qkv"), (28672, 4096, "gate_up"),
                            (4096, 14336, "down")):
            x = torch.randn(T, K, dtype=torch.bfloat16, device=dev) * 0.3
            w = torch.randn(M, K, dtype=torch.bfloat16, device=dev) * 0.05
            xs = (x.abs().amax() / 448.0).float()
            ws = (w.abs().amax() / 448.0).float()
            xq = (x.float() / xs).to(torch.float8_e4m3fn)
            wq = (w.float() / ws).to(torch.float8_e4m3fn)
            wq_t = wq.t()  # [K, M] col-major view of row-major wq

            ref = torch.nn.functional.linear(x, w)

            def run_bf16():
                return torch.nn.functional.linear(x, w)

            def run_fp8():
                return torch._scaled_mm(xq, wq_t, scale_a=xs, scale_b=ws,
                                        out_dtype=torch.bfloat16)

            # rowwise scales: per-token x rows, per-output-col w rows
            xs_r = (x.float().abs().amax(dim=1, keepdim=True) / 448.0)
            ws_r = (w.float().abs().amax(dim=1, keepdim=True) / 448.0)
            xq_r = (x.float() / xs_r).to(torch.float8_e4m3fn)
            wq_r = (w.float() / ws_r).to(torch.float8_e4m3fn)
            wq_rt = wq_r.t()

            def run_fp8_row():
                return torch._scaled_mm(xq_r, wq_rt, scale_a=xs_r,
                                        scale_b=ws_r.t(),
                                        out_dtype=torch.bfloat16)

            try:
                y = run_fp8()
                rel = ((y.float() - ref.float()).abs().max()
                       / ref.float().abs().max()).item()
                t8 = bench(run_fp8)
            except Exception as e:  # noqa: BLE001
                out[f"T{T}_{label}"] = {"error": str(e)[:200]}
                print(json.dumps({f"T{T}_{label}": out[f"T{T}_{label}"]}),
                      flush=True)
                continue
            try:
                y2 = run_fp8_row()
                rel_r = ((y2.float() - ref.float()).abs().max()
                         / ref.float().abs().max()).item()
                t8r = bench(run_fp8_row)
            except Exception as e:  # noqa: BLE001
                rel_r, t8r = str(e)[:120], float("inf")
            t16 = bench(run_bf16)
            flops = 2.0 * T * M * K
            out[f"T{T}_{label}"] = {
                "bf16_us": round(t16 * 1e6, 1),
                "fp8_us": round(t8 * 1e6, 1),
                "speedup": round(t16 / t8, 2),
                "fp8_row_us": (round(t8r * 1e6, 1)
                               if t8r != float("inf") else None),
                "row_err_or_exc": (round(rel_r, 4)
                                   if isinstance(rel_r, float) else rel_r),
                "bf16_tflops": round(flops / t16 / 1e12, 1),
                "fp8_tflops": round(flops / t8 / 1e12, 1),
                "rel_err": round(rel, 4),
            }
            print(json.dumps({f"T{T}_{label}": out[f"T{T}_{label}"]}),
                  flush=True)
    print(json.dumps(out))


if __name__ == "__main__":
    main()
