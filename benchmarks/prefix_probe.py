"""Prefix-cache A/B probe: time-to-first-token for a wave of requests
that share a long system-prompt prefix, with the cache on vs off.

  python benchmarks/prefix_probe.py [--model llama-3-8b] \
      [--prefix-len 1024] [--suffix-len 32] [--batch 32]

Each arm builds a fresh engine, primes the cache with one request (arm
"on" only benefits from page reuse after the prefix pages exist), then
submits a wave of requests sharing the prefix and measures wall time
until every request has produced its first token.
"""
import argparse
import json
import random
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch

from llmlb_amd.backend.launcher import build_engine
from llmlb_amd.backend.sampler import SamplingParams


def run_arm(model, device, enable, prefix_len, suffix_len, batch,
            kv_blocks):
    import llmlb_amd.backend.engine as eng_mod
    eng = build_engine(model, device, kv_blocks=kv_blocks, max_batch=512)
    if not enable:
        eng.prefix = None
    rng = random.Random(7)
    vocab = eng.cfg.model.vocab
    prefix = [rng.randrange(vocab) for _ in range(prefix_len)]

    def wave(n):
        sinks = []
        for i in range(n):
            suffix = [rng.randrange(vocab) for _ in range(suffix_len)]
            _, q = eng.submit(prefix + suffix,
                              SamplingParams(max_new_tokens=4,
                                             temperature=0.0))
            sinks.append(q)
        while any(s.qsize() == 0 for s in sinks):
            eng.step()
        while eng.running or not eng.waiting.empty():
            eng.step()

    wave(1)                       # prime (and warm up kernels/graphs)
    if device.startswith("cuda"):
        torch.cuda.synchronize()
    t0 = time.monotonic()
    wave(batch)
    if device.startswith("cuda"):
        torch.cuda.synchronize()
    dt = time.monotonic() - t0
    hits = eng.prefix.hit_tokens if eng.prefix else 0
    return dt, hits


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default=None)
    ap.add_argument("--prefix-len", type=int, default=1024)
    ap.add_argument("--suffix-len", type=int, default=32)
    ap.add_argument("--batch", type=int, default=32)
    ap.add_argument("--kv-blocks", type=int, default=16384)
    args = ap.parse_args()
    have_gpu = torch.cuda.is_available()
    model = args.model or ("llama-3-8b" if have_gpu else "llama-tiny")
    device = "cuda:0" if have_gpu else "cpu"
    out = {"model": model, "prefix_len": args.prefix_len,
           "suffix_len": args.suffix_len, "batch": args.batch}
    for name, enable in (("cache_on", True), ("cache_off", False)):
        dt, hits = run_arm(model, device, enable, args.prefix_len,
                           args.suffix_len, args.batch, args.kv_blocks)
        out[name] = {"wave_s": round(dt, 4), "hit_tokens": hits}
    on, off = out["cache_on"]["wave_s"], out["cache_off"]["wave_s"]
    out["speedup"] = round(off / on, 3)
    print(json.dumps(out))


if __name__ == "__main__":
    main()
