"""GPU end-to-end engine tests (tiny + small Llama-3-shaped models)."""
import pytest
import torch

import llmlb_amd.ops as ops
from llmlb_amd.backend.engine import Engine, EngineConfig
from llmlb_amd.backend.sampler import SamplingParams
from llmlb_amd.models.llama import LlamaConfig

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def gpu_engine():
    assert ops.native_available(), ops.native_load_error()
    cfg = LlamaConfig.tiny()
    return Engine(EngineConfig(model=cfg, device="cuda:0", dtype="bfloat16",
                               kv_cache_blocks=256))


def test_generate_gpu(gpu_engine):
    res = gpu_engine.generate_sync([1, 2, 3, 4, 5], max_new_tokens=8)
    assert len(res.token_ids) == 8
    torch.cuda.synchronize()


def test_gpu_decode_consistency(gpu_engine):
    """Incremental decode agrees with teacher-forced re-prefill on GPU."""
    prompt = [3, 1, 4, 1, 5]
    res = gpu_engine.generate_sync(prompt, max_new_tokens=4)
    full = prompt + res.token_ids
    mismatch = 0
    for cut in range(len(prompt), len(full)):
        r = gpu_engine.generate_sync(full[:cut], max_new_tokens=1)
        mismatch += (r.token_ids[0] != full[cut])
    # bf16 prefill-vs-decode may rarely flip an argmax on a random-init
    # model with near-tied logits; require agreement on most positions
    assert mismatch <= 1, f"{mismatch} mismatches"


def test_gpu_matches_cpu_reference_structure(gpu_engine):
    """Same tiny config on CPU fp32 vs GPU bf16: greedy tokens should
    broadly agree for a short horizon (random-init model)."""
    cfg = LlamaConfig.tiny()
    cpu = Engine(EngineConfig(model=cfg, device="cpu", dtype="float32",
                              kv_cache_blocks=64))
    g = gpu_engine.generate_sync([2, 4, 6], max_new_tokens=3)
    c = cpu.generate_sync([2, 4, 6], max_new_tokens=3)
    agree = sum(int(a == b) for a, b in zip(g.token_ids, c.token_ids))
    assert agree >= 2, (g.token_ids, c.token_ids)


def test_gpu_many_concurrent(gpu_engine):
    eng = gpu_engine
    qs = []
    for i in range(32):
        _, q = eng.submit([1 + i, 2, 3],
                          SamplingParams(max_new_tokens=8))
        qs.append(q)
    for _ in range(100):
        if not eng.step():
            break
    for q in qs:
        toks = []
        while True:
            tok, fin = q.get_nowait()
            toks.append(tok)
            if fin:
                break
        assert len(toks) == 8
