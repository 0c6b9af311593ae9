"""Minimal single-shape W8A8 kernel loop for rocprofv3 PMC collection.

Runs ONLY the a8 decode GEMM (default: gate_up, the furthest-from-floor
shape) so per-kernel counter aggregation is unambiguous.

  rocprofv3 --pmc SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY \
      SQ_WAVE_CYCLES -d out -- python benchmarks/a8_pmc_probe.py
  rocprofv3 --pmc TCC_EA0_RDREQ TCC_REQ -d out2 -- python ...
"""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch  # noqa: E402

import llmlb_amd.ops as ops  # noqa: E402
from llmlb_amd.ops import reference as ref_mod  # noqa: E402


def main():
    shape = sys.argv[1] if len(sys.argv) > 1 else "gate_up"
    N, M, K = {"gate_up": (64, 28672, 4096),
               "down": (64, 4096, 14336),
               "qkv": (64, 6144, 4096)}[shape]
    reps = int(sys.argv[2]) if len(sys.argv) > 2 else 200
    dev = "cuda:0"
    x = torch.randn(N, K, dtype=torch.bfloat16, device=dev) * 0.3
    w = torch.randn(M, K, dtype=torch.bfloat16, device=dev) * 0.05
    wq, ws = ops.quantize_fp8_weight(w)
    xq, xs = ref_mod.quantize_rows_fp8(x)
    xq, xs = xq.to(dev), xs.to(dev)
    for _ in range(reps):
        ops.decode_gemm_w8a8(xq, xs, wq, ws)
    torch.cuda.synchronize()
    print("done", shape, reps)


if __name__ == "__main__":
    main()
