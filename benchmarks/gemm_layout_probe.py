"""Weight-layout microbench for the decode GEMMs: hipBLASLt via
F.linear(x, W[M,K]) vs matmul with a pre-transposed W^T [K,M] (both
bf16).  Different operand layouts steer hipBLASLt to different tile
kernels; if K-major wins we change the weight storage at init."""
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch

SHAPES = [
    (64, 6144, 4096, "qkv"),
    (64, 4096, 4096, "o_proj"),
    (64, 28672, 4096, "gate_up"),
    (64, 4096, 14336, "down"),
]
ROUNDS = 30


def bench(fn, rounds=ROUNDS):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    ts = []
    for _ in range(rounds):
        t0 = time.perf_counter()
        fn()
        torch.cuda.synchronize()
        ts.append(time.perf_counter() - t0)
    ts.sort()
    return ts[len(ts) // 2]


def main():
    dev = "cuda:0"
    out = {}
    for N, M, K, label in SHAPES:
        x = torch.randn(N, K, dtype=torch.bfloat16, device=dev)
        w = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
        wt = w.t().contiguous()          # [K, M]
        t_lin = bench(lambda: torch.nn.functional.linear(x, w))
        t_mm = bench(lambda: torch.matmul(x, wt))
        bytes_w = M * K * 2
        out[label] = {
            "linear_us": round(t_lin * 1e6, 1),
            "matmul_t_us": round(t_mm * 1e6, 1),
            "linear_tbs": round(bytes_w / t_lin / 1e12, 2),
            "matmul_t_tbs": round(bytes_w / t_mm / 1e12, 2),
            "ratio": round(t_lin / t_mm, 3),
        }
    print(json.dumps(out))


if __name__ == "__main__":
    main()
