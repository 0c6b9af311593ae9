"""Speculative-decoding A/B: greedy long-generation throughput with
prompt-lookup drafting on vs off.  Greedy decoding of random-init
models typically falls into cycles, so the n-gram drafter gets real
acceptances — reported alongside tok/s.

  python benchmarks/spec_probe.py [--model llama-3-8b] [--batch 32]
      [--gen 256]
"""
import argparse
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch

from llmlb_amd.backend.engine import Engine, EngineConfig
from llmlb_amd.backend.launcher import PRESETS
from llmlb_amd.backend.sampler import SamplingParams


def run_arm(model, device, spec, batch, gen, kv_blocks):
    eng = Engine(EngineConfig(model=PRESETS[model](), device=device,
                              dtype=("bfloat16" if device.startswith("cuda")
                                     else "float32"),
                              kv_cache_blocks=kv_blocks, max_batch=512,
                              spec_ngram=spec))
    p = SamplingParams(temperature=0.0, max_new_tokens=gen)
    import random
    rng = random.Random(3)
    vocab = eng.cfg.model.vocab
    sinks = [eng.submit([rng.randrange(vocab) for _ in range(32)], p)[1]
             for _ in range(batch)]
    # warm up kernels/graphs with the prefill + a few steps
    for _ in range(4):
        eng.step()
    if device.startswith("cuda"):
        torch.cuda.synchronize()
    t0 = time.monotonic()
    while eng.running or not eng.waiting.empty():
        eng.step()
    if device.startswith("cuda"):
        torch.cuda.synchronize()
    dt = time.monotonic() - t0
    toks = sum(q.qsize() for q in sinks)
    acc = (eng.spec_accepted / eng.spec_proposed
           if eng.spec_proposed else 0.0)
    return {"tokens": toks, "s": round(dt, 3),
            "tok_per_s": round(toks / dt, 1),
            "acceptance": round(acc, 3),
            "steps": eng.steps_done}


def run_oracle(model, device, spec, batch, gen, kv_blocks):
    """Oracle-draft arm: drafts ARE the model's own future greedy
    tokens (pre-computed), i.e. 100% acceptance — measures the
    verification machinery's ceiling (one (K+1)-token forward vs K+1
    single steps).  Random-init weights cannot produce organic
    repetition (profiles/r01_spec_decode.md), so this is the honest
    way to quantify what the machinery delivers when the WORKLOAD
    repeats (extraction, code edits, RAG quoting)."""
    import random
    from llmlb_amd.backend.engine import Engine, EngineConfig
    rng = random.Random(3)
    cfg = EngineConfig(model=PRESETS[model](), device=device,
                       dtype=("bfloat16" if device.startswith("cuda")
                              else "float32"),
                       kv_cache_blocks=kv_blocks, max_batch=512,
                       spec_ngram=spec, seed=7)
    eng = Engine(cfg)
    vocab = eng.cfg.model.vocab
    prompts = [[rng.randrange(vocab) for _ in range(32)]
               for _ in range(batch)]
    # pass 1 (untimed): capture each prompt's greedy continuation
    p = SamplingParams(temperature=0.0, max_new_tokens=gen)
    sinks = [eng.submit(list(pr), p) for pr in prompts]
    while eng.running or not eng.waiting.empty():
        eng.step()
    outs = []
    for rid, q in sinks:
        toks = []
        while not q.empty():
            toks.append(q.get()[0])
        outs.append(toks)
    # pass 2: fresh engine, oracle drafts from the captured outputs
    eng2 = Engine(cfg)
    orig_draft = eng2._draft_for
    future_by_rid = {}

    def oracle_draft(seq):
        fut = future_by_rid.get(seq.rid)
        if fut is None:
            return orig_draft(seq)
        done = len(seq.out_ids)
        return fut[done:done + eng2.spec_ngram]

    eng2._draft_for = oracle_draft
    sinks2 = []
    for pr, fut in zip(prompts, outs):
        rid, q = eng2.submit(list(pr), SamplingParams(
            temperature=0.0, max_new_tokens=gen))
        future_by_rid[rid] = fut
        sinks2.append((rid, q))
    for _ in range(4):
        eng2.step()
    if device.startswith("cuda"):
        torch.cuda.synchronize()
    t0 = time.monotonic()
    while eng2.running or not eng2.waiting.empty():
        eng2.step()
    if device.startswith("cuda"):
        torch.cuda.synchronize()
    dt = time.monotonic() - t0
    toks = sum(q.qsize() for _, q in sinks2)
    acc = (eng2.spec_accepted / eng2.spec_proposed
           if eng2.spec_proposed else 0.0)
    # verify exactness vs pass 1
    exact = True
    for (rid, q), fut in zip(sinks2, outs):
        got = []
        while not q.empty():
            got.append(q.get()[0])
        if got != fut:
            exact = False
    return {"tokens": toks, "s": round(dt, 3),
            "tok_per_s": round(toks / dt, 1),
            "acceptance": round(acc, 3), "exact": exact,
            "steps": eng2.steps_done}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default=None)
    ap.add_argument("--batch", type=int, default=32)
    ap.add_argument("--gen", type=int, default=256)
    ap.add_argument("--spec", type=int, default=4)
    ap.add_argument("--kv-blocks", type=int, default=8192)
    ap.add_argument("--oracle", action="store_true")
    args = ap.parse_args()
    have_gpu = torch.cuda.is_available()
    model = args.model or ("llama-3-8b" if have_gpu else "llama-tiny")
    device = "cuda:0" if have_gpu else "cpu"
    out = {"model": model, "batch": args.batch, "gen": args.gen}
    for name, k in (("spec_on", args.spec), ("spec_off", 0),
                    ("spec_on2", args.spec), ("spec_off2", 0)):
        out[name] = run_arm(model, device, k, args.batch, args.gen,
                            args.kv_blocks)
    out["speedup"] = round(out["spec_on"]["tok_per_s"]
                           / out["spec_off"]["tok_per_s"], 3)
    if args.oracle:
        out["oracle"] = run_oracle(model, device, args.spec, args.batch,
                                   args.gen, args.kv_blocks)
        out["oracle_speedup"] = round(out["oracle"]["tok_per_s"]
                                      / out["spec_off"]["tok_per_s"], 3)
    print(json.dumps(out))


if __name__ == "__main__":
    main()
