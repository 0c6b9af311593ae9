"""CPU tests of the backend engine with the fp32 reference ops."""
import queue

import pytest
import torch

from llmlb_amd.backend.engine import Engine, EngineConfig
from llmlb_amd.backend.sampler import SamplingParams
from llmlb_amd.models.llama import LlamaConfig


@pytest.fixture(scope="module")
def tiny_engine():
    cfg = LlamaConfig.tiny()
    return Engine(EngineConfig(model=cfg, device="cpu", dtype="float32",
                               kv_cache_blocks=64))


def test_generate_greedy_deterministic(tiny_engine):
    r1 = tiny_engine.generate_sync([1, 2, 3, 4, 5], max_new_tokens=8)
    r2 = tiny_engine.generate_sync([1, 2, 3, 4, 5], max_new_tokens=8)
    assert len(r1.token_ids) == 8
    assert r1.token_ids == r2.token_ids
    assert r1.finish_reason == "length"


def test_decode_matches_full_forward(tiny_engine):
    """Engine incremental decode must agree with a from-scratch forward
    over the grown sequence (same model, teacher-forced)."""
    eng = tiny_engine
    prompt = [7, 11, 13]
    res = eng.generate_sync(prompt, max_new_tokens=4)
    # re-run with the full sequence as a prompt; next-token logits after
    # position k must reproduce token k+1 of the engine's output
    full = prompt + res.token_ids
    for cut in range(len(prompt), len(full)):
        r = eng.generate_sync(full[:cut], max_new_tokens=1)
        assert r.token_ids[0] == full[cut], (
            f"mismatch at position {cut}: {r.token_ids[0]} vs {full[cut]}")


def test_concurrent_requests_isolated(tiny_engine):
    eng = tiny_engine
    ra, qa = eng.submit([1, 2, 3], SamplingParams(max_new_tokens=6))
    rb, qb = eng.submit([9, 8, 7, 6], SamplingParams(max_new_tokens=6))
    for _ in range(40):
        if not eng.step():
            break
    a = _drain(qa)
    b = _drain(qb)
    assert len(a) == 6 and len(b) == 6
    # each must match its solo run
    solo_a = tiny_engine.generate_sync([1, 2, 3], max_new_tokens=6)
    assert a == solo_a.token_ids


def test_kv_pages_released(tiny_engine):
    eng = tiny_engine
    free0 = eng.kv.free_blocks()
    eng.generate_sync([1] * 20, max_new_tokens=20)
    # full prompt pages park in the prefix-cache LRU; everything else
    # returns to the free list, and parked pages are reclaimable
    parked = len(eng.prefix.lru) if eng.prefix else 0
    assert eng.kv.free_blocks() + parked == free0
    if eng.prefix:
        assert eng.prefix.ensure_free(free0)
        assert eng.kv.free_blocks() == free0


def test_stop_tokens(tiny_engine):
    res = tiny_engine.generate_sync(
        [1, 2, 3], max_new_tokens=50,
        params=SamplingParams(max_new_tokens=50,
                              stop_token_ids=tuple(range(1024))))
    assert res.finish_reason == "stop"
    assert len(res.token_ids) == 1


def _drain(q: queue.Queue) -> list:
    toks = []
    while True:
        try:
            tok, fin, _ex = q.get_nowait()
        except queue.Empty:
            break
        toks.append(tok)
        if fin:
            break
    return toks


def test_sampling_temperature_seeded(tiny_engine):
    p = SamplingParams(temperature=0.8, top_p=0.9, max_new_tokens=6, seed=42)
    r1 = tiny_engine.generate_sync([5, 6], max_new_tokens=6, params=p)
    p2 = SamplingParams(temperature=0.8, top_p=0.9, max_new_tokens=6, seed=42)
    r2 = tiny_engine.generate_sync([5, 6], max_new_tokens=6, params=p2)
    assert r1.token_ids == r2.token_ids


def test_sampling_filters_and_penalties():
    """top_k / min_p / penalties: shape-level behavior on a controlled
    logits row (unit test of sampler.sample, no engine needed)."""
    import torch
    from llmlb_amd.backend.sampler import SamplingParams, sample
    logits = torch.tensor([[5.0, 4.0, 3.0, 2.0, 1.0]])
    # top_k=1 must always pick the argmax even at high temperature
    g = torch.Generator().manual_seed(0)
    p = SamplingParams(temperature=5.0, top_k=1)
    assert sample(logits, [p], [g]).item() == 0
    # presence penalty pushes a repeated token below the runner-up
    p2 = SamplingParams(temperature=0.0, presence_penalty=2.0)
    tok = sample(logits, [p2], [None],
                 histories=[([1, 2], [0])]).item()
    assert tok == 1
    # frequency penalty scales with count
    p3 = SamplingParams(temperature=0.0, frequency_penalty=0.6)
    tok = sample(logits, [p3], [None],
                 histories=[([], [0, 0])]).item()
    assert tok == 1
    # repetition penalty suppresses prompt tokens too
    p4 = SamplingParams(temperature=0.0, repetition_penalty=2.0)
    tok = sample(logits, [p4], [None], histories=[([0], [])]).item()
    assert tok == 1
    # min_p keeps only near-max tokens
    p5 = SamplingParams(temperature=1.0, min_p=0.9, seed=1)
    g5 = torch.Generator().manual_seed(1)
    for _ in range(5):
        assert sample(logits, [p5], [g5]).item() == 0


def test_logprobs_chat(tiny_engine):
    """params.logprobs=k yields per-token chosen logprob + top-k
    alternatives; chosen logprob must be <= 0 and appear in top-k for
    greedy decoding."""
    from llmlb_amd.backend.sampler import SamplingParams
    eng = tiny_engine
    p = SamplingParams(temperature=0.0, max_new_tokens=4, logprobs=3)
    rid, q = eng.submit([5, 6, 7], p)
    for _ in range(30):
        if not eng.step():
            break
    items = []
    while not q.empty():
        items.append(q.get_nowait())
    assert len(items) == 4
    for tok, fin, extra in items:
        assert extra is not None
        assert extra["logprob"] <= 0.0
        assert len(extra["top"]) == 3
        # greedy: the chosen token is the top-1 alternative
        assert extra["top"][0][0] == tok


def test_capacity_never_kills_engine():
    """Worst-case page reservation at admission: a flood of requests
    against a tiny KV pool queues (head-of-line) instead of crashing
    the engine thread mid-decode, and everything eventually finishes."""
    from llmlb_amd.models.llama import LlamaConfig
    from llmlb_amd.backend.engine import Engine, EngineConfig
    eng = Engine(EngineConfig(model=LlamaConfig.tiny(), device="cpu",
                              dtype="float32", kv_cache_blocks=24,
                              seed=3))
    p = SamplingParams(temperature=0.0, max_new_tokens=40)
    sinks = [eng.submit([i, i + 1, i + 2], p)[1] for i in range(12)]
    for _ in range(3000):
        busy = eng.step()
        if not busy and eng.waiting.empty() and not eng.running:
            break
    for i, q in enumerate(sinks):
        toks = []
        while not q.empty():
            toks.append(q.get_nowait())
        assert len(toks) == 40, (i, len(toks))
    assert eng.kv.free_blocks() > 0


def test_abort_frees_pages(tiny_engine):
    """abort(rid) retires a running sequence at the next step and
    releases everything (pages, row, queue slot)."""
    eng = tiny_engine
    free0 = eng.kv.free_blocks()
    p = SamplingParams(temperature=0.0, max_new_tokens=200)
    rid, q = eng.submit([1, 2, 3, 4], p)
    for _ in range(5):
        eng.step()
    assert eng.num_running == 1
    eng.abort(rid)
    eng.step()
    assert eng.num_running == 0
    parked = len(eng.prefix.lru) if eng.prefix else 0
    assert eng.kv.free_blocks() + parked == free0
    # aborting a queued (not yet admitted) request also works
    rid2, _ = eng.submit([5] * 8, p)
    eng.abort(rid2)
    eng.step()
    assert eng.num_running == 0 and eng.waiting.empty()


def test_engine_reproducibility_across_instances():
    """Same seed -> identical weights and identical greedy output in a
    fresh engine (deterministic synthetic serving guarantee)."""
    from llmlb_amd.models.llama import LlamaConfig
    from llmlb_amd.backend.engine import Engine, EngineConfig
    outs = []
    for _ in range(2):
        eng = Engine(EngineConfig(model=LlamaConfig.tiny(), device="cpu",
                                  dtype="float32", kv_cache_blocks=32,
                                  seed=77))
        outs.append(eng.generate_sync(
            [9, 1, 1], 6, SamplingParams(temperature=0.0)).token_ids)
    assert outs[0] == outs[1]
    # and sampled output with a fixed request seed is reproducible too
    eng = Engine(EngineConfig(model=LlamaConfig.tiny(), device="cpu",
                              dtype="float32", kv_cache_blocks=32,
                              seed=77))
    p = SamplingParams(temperature=0.8, seed=5)
    a = eng.generate_sync([4, 4, 4], 6, p).token_ids
    p2 = SamplingParams(temperature=0.8, seed=5)
    b = eng.generate_sync([4, 4, 4], 6, p2).token_ids
    assert a == b


def test_logit_bias(tiny_engine):
    """A +100 logit_bias forces the token; -100 suppresses it."""
    eng = tiny_engine
    p0 = SamplingParams(temperature=0.0, max_new_tokens=3)
    base = eng.generate_sync([2, 3, 4], 3, p0).token_ids
    p1 = SamplingParams(temperature=0.0, max_new_tokens=3,
                        logit_bias=((7, 100.0),))
    forced = eng.generate_sync([2, 3, 4], 3, p1).token_ids
    assert forced == [7, 7, 7]
    p2 = SamplingParams(temperature=0.0, max_new_tokens=1,
                        logit_bias=((base[0], -100.0),))
    avoided = eng.generate_sync([2, 3, 4], 1, p2).token_ids
    assert avoided[0] != base[0]


def test_decode_piggyback_between_prefill_chunks():
    """A long chunked prefill must not starve running streams: each
    chunk boundary runs one decode step for them."""
    from llmlb_amd.backend.engine import Engine, EngineConfig
    from llmlb_amd.backend.launcher import PRESETS
    from llmlb_amd.backend.sampler import SamplingParams
    cfg = EngineConfig(model=PRESETS["llama-tiny"](), device="cpu",
                       kv_cache_blocks=256, prefill_chunk=32,
                       enable_prefix_cache=False, seed=3)
    eng = Engine(cfg)
    # one running stream
    rid1, q1 = eng.submit([1, 2, 3],
                          SamplingParams(max_new_tokens=64,
                                         temperature=0.0))
    eng.step()
    before = q1.qsize()   # prefill token + the step's decode
    # admit a 4-chunk prompt (128 tokens / chunk 32)
    rid2, q2 = eng.submit(list(range(5, 133)),
                          SamplingParams(max_new_tokens=4,
                                         temperature=0.0))
    eng.step()
    # the running stream got a token at each of the 3 interior chunk
    # boundaries plus the step's own decode
    assert q1.qsize() - before >= 4, (before, q1.qsize())
