"""Debug probe for the prefill fp8/bf16 auto-tuner: print per-shape
tuner timings (cache-evicted) AND a cold-stream style comparison for
the llama-3-8b prefill GEMM shapes.

  gpurun -- 'LLMLB_FP8_A8=1 python benchmarks/prefill_tune_probe.py'
"""
import os
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))
os.environ.setdefault("LLMLB_FP8_A8", "1")

import torch
import torch.nn.functional as F

import llmlb_amd.ops as ops

DEV = "cuda:0"
SHAPES = [  # (name, M, K) llama-3-8b prefill GEMMs
    ("qkv", 6144, 4096),
    ("o_proj", 4096, 4096),
    ("gate_up", 28672, 4096),
    ("down", 4096, 14336),
]
T = int(os.environ.get("T", "4096"))


def main():
    assert ops.native_available(), ops.native_load_error()
    evict = torch.empty(320 * 1024 * 1024, dtype=torch.uint8, device=DEV)

    for name, M, K in SHAPES:
        torch.manual_seed(0)
        w = torch.randn(M, K, dtype=torch.bfloat16, device=DEV) * 0.02
        x = torch.randn(T, K, dtype=torch.bfloat16, device=DEV)
        wq, wsc = ops.quantize_fp8_weight(w)

        # warmup both
        y8 = ops._scaled_mm_raw(x, wq, wsc)
        yb = F.linear(x, w)
        torch.cuda.synchronize()
        err = (y8.float() - yb.float()).abs().max().item()

        def timed(fn, iters=5, cold=True):
            es = [torch.cuda.Event(enable_timing=True)
                  for _ in range(2 * iters)]
            for i in range(iters):
                if cold:
                    evict.fill_(i)
                es[2 * i].record()
                fn()
                es[2 * i + 1].record()
            torch.cuda.synchronize()
            ts = [es[2 * i].elapsed_time(es[2 * i + 1])
                  for i in range(iters)]
            return min(ts), sum(ts) / iters

        f8_cold = timed(lambda: ops._scaled_mm_raw(x, wq, wsc))
        bf_cold = timed(lambda: F.linear(x, w))
        f8_hot = timed(lambda: ops._scaled_mm_raw(x, wq, wsc), cold=False)
        bf_hot = timed(lambda: F.linear(x, w), cold=False)
        dec = ops._tune_prefill(x, w, wq, wsc)
        # component split: rows_q alone
        rq = timed(lambda: ops.rows_q(x), cold=False)
        print(f"{name:8s} M={M:6d} K={K:6d} T={T} | "
              f"fp8 cold {f8_cold[1]:7.3f} ms  bf16 cold {bf_cold[1]:7.3f}"
              f" | fp8 hot {f8_hot[1]:7.3f}  bf16 hot {bf_hot[1]:7.3f}"
              f" | rows_q {rq[1]:6.3f} | tuner picks "
              f"{'fp8' if dec else 'bf16'} | max|d|={err:.4f}",
              flush=True)


if __name__ == "__main__":
    main()
