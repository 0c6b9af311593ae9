"""Time-between-tokens probe: N streaming decodes run while long
prompts keep arriving; reports TBT p50/p99 over the streams.  A/B:
LLMLB_PREFILL_PIGGYBACK=0 disables the chunk-boundary decode step.

  python benchmarks/tbt_probe.py [--streams 8] [--prompt-len 8192]
"""
import argparse
import json
import sys
import threading
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch

from llmlb_amd.backend.launcher import build_engine
from llmlb_amd.backend.sampler import SamplingParams


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default=None)
    ap.add_argument("--streams", type=int, default=8)
    ap.add_argument("--prompt-len", type=int, default=8192)
    ap.add_argument("--chunk", type=int, default=2048)
    ap.add_argument("--duration", type=float, default=20.0)
    args = ap.parse_args()
    have_gpu = torch.cuda.is_available()
    model = args.model or ("llama-3-8b" if have_gpu else "llama-tiny")
    dev = "cuda:0" if have_gpu else "cpu"
    eng = build_engine(model, dev, kv_blocks=16384 if have_gpu else 512,
                       max_batch=256)
    eng.cfg.prefill_chunk = args.chunk
    eng.start()

    gaps: list[float] = []
    stop = threading.Event()

    def stream_worker():
        params = SamplingParams(max_new_tokens=100000, temperature=0.0)
        rid, q = eng.submit(list(range(1, 33)), params)
        last = time.monotonic()
        while not stop.is_set():
            try:
                tok, fin, _ = q.get(timeout=5.0)
            except Exception:
                continue
            now = time.monotonic()
            gaps.append(now - last)
            last = now
            if fin:
                rid, q = eng.submit(list(range(1, 33)), params)
                last = time.monotonic()

    threads = [threading.Thread(target=stream_worker, daemon=True)
               for _ in range(args.streams)]
    for t in threads:
        t.start()
    time.sleep(2.0)
    gaps.clear()
    t_end = time.time() + args.duration
    prompts = 0
    while time.time() < t_end:
        # long prompt arrives; its chunked prefill is the TBT hazard
        eng.submit(list(range(3, 3 + args.prompt_len)),
                   SamplingParams(max_new_tokens=2, temperature=0.0))
        prompts += 1
        time.sleep(1.0)
    stop.set()
    time.sleep(0.5)
    eng.stop()
    gaps.sort()
    n = len(gaps)
    out = {"streams": args.streams, "prompt_len": args.prompt_len,
           "chunk": eng.cfg.prefill_chunk, "long_prompts": prompts,
           "tokens": n,
           "tbt_p50_ms": round(gaps[n // 2] * 1e3, 1) if n else None,
           "tbt_p99_ms": round(gaps[int(n * 0.99)] * 1e3, 1)
           if n else None,
           "tbt_max_ms": round(gaps[-1] * 1e3, 1) if n else None}
    print(json.dumps(out))


if __name__ == "__main__":
    main()
