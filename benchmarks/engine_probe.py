"""Direct engine probe (no HTTP): prefill and decode throughput per
batch size, for kernel/profiling work on the GPU box.

  python benchmarks/engine_probe.py [--model llama-3-8b] [--batches 1,8,64]
"""
import argparse
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch

from llmlb_amd.backend.launcher import build_engine
from llmlb_amd.backend.sampler import SamplingParams


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default=None)
    ap.add_argument("--batches", default="1,8,32,64")
    ap.add_argument("--prompt-len", type=int, default=128)
    ap.add_argument("--gen-len", type=int, default=64)
    ap.add_argument("--kv-blocks", type=int, default=8192)
    args = ap.parse_args()

    have_gpu = torch.cuda.is_available()
    model = args.model or ("llama-3-8b" if have_gpu else "llama-tiny")
    device = "cuda:0" if have_gpu else "cpu"
    t0 = time.monotonic()
    # admit every probe prompt in one step so the decode phase runs at
    # the full batch (staggered admission pollutes decode ms/step)
    eng = build_engine(model, device, kv_blocks=args.kv_blocks,
                       max_batch=512,
                       max_prefill_tokens=max(
                           8192, args.prompt_len *
                           max(int(b) for b in
                               args.batches.split(","))))
    init_s = time.monotonic() - t0
    results = {"model": model, "device": device, "init_s": round(init_s, 2)}

    def sync():
        if have_gpu:
            torch.cuda.synchronize()

    for bs in [int(b) for b in args.batches.split(",")]:
        # submit bs requests, run prefill-only steps, then decode
        qs = []
        for i in range(bs):
            _, q = eng.submit(
                [(i * 7 + j) % 1000 + 2 for j in range(args.prompt_len)],
                SamplingParams(max_new_tokens=args.gen_len))
            qs.append(q)
        sync()
        t_pre = time.monotonic()
        eng.step()          # admits + prefills all + first decode emit
        sync()
        prefill_s = time.monotonic() - t_pre
        t_dec = time.monotonic()
        steps = 0
        while eng.num_running and steps < args.gen_len + 4:
            eng.step()
            steps += 1
        sync()
        decode_s = time.monotonic() - t_dec
        total_decoded = sum(q.qsize() for q in qs)
        timing = {}
        if eng.timing is not None and eng.timing.get("steps"):
            n = eng.timing["steps"]
            timing = {k: round(v / n * 1000, 3)
                      for k, v in eng.timing.items() if k != "steps"}
            eng.timing.clear()
        results[f"bs{bs}"] = {
            "graphs": sorted(eng._graphs.keys()),
            "use_graphs": eng.use_graphs,
            "timing_ms_per_step": timing,
            "prefill_tokens_per_s": round(bs * args.prompt_len / prefill_s,
                                          1),
            "prefill_s": round(prefill_s, 3),
            "decode_steps": steps,
            "decode_tokens_per_s": round(
                bs * max(steps - 1, 1) / decode_s, 1),
            "decode_ms_per_step": round(decode_s / max(steps, 1) * 1000,
                                        2),
            "emitted": total_decoded,
        }
        print(json.dumps({f"bs{bs}": results[f"bs{bs}"]}), flush=True)
    print(json.dumps(results))


if __name__ == "__main__":
    main()
