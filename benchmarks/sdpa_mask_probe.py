"""Chunk-2 prefill attention probe: torch SDPA with the rectangular
causal mask (queries [P, L) vs keys [0, L)) falls to the fp32 math
path with a bool mask (profiles/r02_fp8_prefill_ab.md).  Try the same
shapes with (a) bool mask, (b) bf16 additive mask, (c) explicit
backend restriction, and time each — the round-3 prefill lever.

  gpurun -- 'python benchmarks/sdpa_mask_probe.py'
"""
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))
import torch
import torch.nn.functional as F
from torch.nn.attention import SDPBackend, sdpa_kernel

DEV = "cuda:0"
B, HQ, HKV, D = 8, 32, 8, 128
P, Ls = 4096, 2048
L = P + Ls


def run(mask, label, backends=None):
    q = torch.randn(B, HQ, Ls, D, dtype=torch.bfloat16, device=DEV)
    k = torch.randn(B, HKV, L, D, dtype=torch.bfloat16, device=DEV)
    v = torch.randn_like(k)

    def call():
        return F.scaled_dot_product_attention(
            q, k, v, attn_mask=mask, enable_gqa=True)
    try:
        if backends:
            with sdpa_kernel(backends):
                call()
                torch.cuda.synchronize()
                t0 = time.monotonic()
                for _ in range(5):
                    call()
                torch.cuda.synchronize()
        else:
            call()
            torch.cuda.synchronize()
            t0 = time.monotonic()
            for _ in range(5):
                call()
            torch.cuda.synchronize()
        print(f"{label:42s} {(time.monotonic()-t0)/5*1e3:9.2f} ms/call",
              flush=True)
    except Exception as e:
        print(f"{label:42s} FAILED: {str(e)[:90]}", flush=True)


def main():
    bool_mask = (torch.arange(L, device=DEV)[None, :]
                 <= torch.arange(P, L, device=DEV)[:, None])
    add_mask = torch.zeros(Ls, L, dtype=torch.bfloat16, device=DEV) \
        .masked_fill_(~bool_mask, float("-inf"))
    run(bool_mask, "bool mask, default dispatch")
    run(add_mask, "bf16 additive mask, default dispatch")
    run(bool_mask, "bool mask, EFFICIENT only",
        [SDPBackend.EFFICIENT_ATTENTION])
    run(add_mask, "bf16 additive mask, EFFICIENT only",
        [SDPBackend.EFFICIENT_ATTENTION])
    run(bool_mask, "bool mask, FLASH only",
        [SDPBackend.FLASH_ATTENTION])
    run(add_mask, "bf16 additive mask, FLASH only",
        [SDPBackend.FLASH_ATTENTION])
    run(bool_mask, "bool mask, MATH only", [SDPBackend.MATH])


if __name__ == "__main__":
    main()
