"""2-rank gloo check of the TP lockstep coordinator: full-params
broadcast, generation, abort propagation and group shutdown.
Launched by tests/test_tp_cpu.py via torchrun; exits 0 on success."""
import queue
import sys
import threading
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))

import torch.distributed as dist

from llmlb_amd.backend.engine import Engine, EngineConfig
from llmlb_amd.backend.sampler import SamplingParams
from llmlb_amd.backend.tp_runner import TPEngineCoordinator
from llmlb_amd.models.llama import LlamaConfig


def main():
    dist.init_process_group("gloo")
    rank = dist.get_rank()
    cfg = EngineConfig(model=LlamaConfig.tiny(), device="cpu",
                       dtype="float32", kv_cache_blocks=64, seed=21)
    eng = Engine(cfg)
    assert eng.model.tp.world_size == 2, eng.model.tp.world_size
    coord = TPEngineCoordinator(eng, group_cpu=dist.group.WORLD)
    if rank != 0:
        coord.loop()
        return
    t = threading.Thread(target=coord.loop, daemon=True)
    t.start()
    sink = queue.Queue()
    coord.submit([1, 2, 3],
                 SamplingParams(temperature=0.0, max_new_tokens=6,
                                top_k=1),
                 sink=sink)
    toks = []
    deadline = time.time() + 60
    while len(toks) < 6 and time.time() < deadline:
        try:
            toks.append(sink.get(timeout=5)[0])
        except queue.Empty:
            pass
    assert len(toks) == 6, toks
    # repeatability: the same greedy request returns identical tokens
    # (weights are per-rank random shards, so there is no solo
    # reference to compare against — determinism is the contract)
    sink_b = queue.Queue()
    coord.submit([1, 2, 3],
                 SamplingParams(temperature=0.0, max_new_tokens=6,
                                top_k=1),
                 sink=sink_b)
    toks_b = []
    deadline = time.time() + 60
    while len(toks_b) < 6 and time.time() < deadline:
        try:
            toks_b.append(sink_b.get(timeout=5)[0])
        except queue.Empty:
            pass
    assert toks_b == toks, (toks_b, toks)
    # abort propagates to every rank without desync
    rid2, _ = coord.submit([4, 5, 6],
                           SamplingParams(temperature=0.0,
                                          max_new_tokens=500),
                           sink=queue.Queue())
    time.sleep(0.5)
    coord.abort(rid2)
    deadline = time.time() + 30
    while time.time() < deadline and eng.num_running:
        time.sleep(0.1)
    assert eng.num_running == 0
    coord.stop()
    t.join(timeout=30)
    assert not t.is_alive()
    print("tp-coord-check OK")


if __name__ == "__main__":
    main()
