"""Automatic prefix caching (kv_cache.PrefixCache): page-aligned prompt
KV reuse must be output-invariant and must actually hit on shared
prefixes.  CPU-only; the GPU path is covered by test_engine_gpu."""
import pytest
import torch

from llmlb_amd.backend.engine import Engine, EngineConfig
from llmlb_amd.backend.kv_cache import BLOCK_SIZE, PrefixCache
from llmlb_amd.backend.sampler import SamplingParams
from llmlb_amd.models.llama import LlamaConfig


def _engine(blocks=64, prefix=True):
    cfg = EngineConfig(model=LlamaConfig.tiny(),
                       device="cpu", dtype="float32",
                       kv_cache_blocks=blocks,
                       enable_prefix_cache=prefix, seed=7)
    return Engine(cfg)


def _greedy(engine, prompt, n=8):
    p = SamplingParams(temperature=0.0)
    return engine.generate_sync(prompt, max_new_tokens=n, params=p).token_ids


def test_hit_and_output_invariance():
    prompt = list(range(40))  # 2 full pages cacheable
    ref = _greedy(_engine(prefix=False), prompt)
    eng = _engine(prefix=True)
    first = _greedy(eng, prompt)
    assert eng.prefix.hit_tokens == 0
    second = _greedy(eng, prompt)
    assert first == ref
    assert second == ref
    assert eng.prefix.hit_tokens == (len(prompt) - 1) // BLOCK_SIZE \
        * BLOCK_SIZE


def test_partial_prefix_hit():
    eng = _engine()
    a = list(range(64))
    b = list(range(48)) + [99] * 16   # shares 3 pages with a
    _greedy(eng, a)
    out_b1 = _greedy(eng, b)
    assert eng.prefix.hit_tokens == 48
    # same request again must reproduce (all full pages of b now cached)
    out_b2 = _greedy(eng, b)
    assert out_b2 == out_b1
    # and an uncached engine agrees
    assert _greedy(_engine(prefix=False), b) == out_b1


def test_eviction_under_pressure():
    eng = _engine(blocks=24)          # tiny pool forces LRU eviction
    outs = {}
    for base in range(5):
        prompt = [base * 100 + i for i in range(48)]
        outs[base] = _greedy(eng, prompt, n=4)
    # pool is oversubscribed: parked pages were evicted, engine still
    # serves correct results for a re-run of the first prompt
    again = _greedy(eng, [0 * 100 + i for i in range(48)], n=4)
    assert again == outs[0]
    assert eng.kv.free_blocks() >= 0


def test_chain_hash_is_prefix_sensitive():
    a = PrefixCache.chain_hashes(list(range(32)))
    b = PrefixCache.chain_hashes(list(range(16)) + list(range(100, 116)))
    assert a[0] == b[0]
    assert a[1] != b[1]
    assert len(PrefixCache.chain_hashes(list(range(15)))) == 0


def test_chunked_prefill_matches_unchunked():
    prompt = list(range(100))
    ref = _greedy(_engine(prefix=False), prompt, n=6)
    cfg = EngineConfig(model=LlamaConfig.tiny(), device="cpu",
                       dtype="float32", kv_cache_blocks=64,
                       enable_prefix_cache=True, prefill_chunk=32, seed=7)
    eng = Engine(cfg)
    assert _greedy(eng, prompt, n=6) == ref
    # chunking composes with a cached prefix (second request hits)
    assert _greedy(eng, prompt, n=6) == ref
    assert eng.prefix.hit_tokens > 0
