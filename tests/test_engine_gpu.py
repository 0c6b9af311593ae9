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


def test_gpu_matches_cpu_reference_logits(gpu_engine):
    """Prefill logits of the GPU bf16 stack (HIP kernels + hipBLASLt)
    must correlate with the CPU fp32 reference stack on the same
    random-init weights.  (Token-level argmax agreement is flaky by
    design on a random-init model with near-tied logits.)"""
    cfg = LlamaConfig.tiny()
    cpu = Engine(EngineConfig(model=cfg, device="cpu", dtype="float32",
                              kv_cache_blocks=64))
    # CUDA and CPU RNGs produce different streams for the same seed, so
    # sync the CPU model to the GPU model's weights before comparing
    gpu_state = {k: v.detach().float().cpu()
                 for k, v in gpu_engine.model.state_dict().items()}
    for name, p in cpu.model.state_dict().items():
        p.copy_(gpu_state[name])

    def prefill_logits(eng, prompt):
        ids = torch.tensor(prompt, dtype=torch.long, device=eng.device)
        pos = torch.arange(len(prompt), dtype=torch.int32,
                           device=eng.device)
        h = eng.model.embed_tokens(ids)
        residual = None
        T = len(prompt)
        for li in range(eng.cfg.model.layers):
            q, k, v, residual = eng.model.layer_attn_prelude(
                li, h, residual, pos)
            qh = q.transpose(0, 1).unsqueeze(0)
            kh = k.transpose(0, 1).unsqueeze(0)
            vh = v.transpose(0, 1).unsqueeze(0)
            o = torch.nn.functional.scaled_dot_product_attention(
                qh, kh, vh, is_causal=T > 1, enable_gqa=True)
            attn = o.squeeze(0).transpose(0, 1).contiguous()
            h, residual = eng.model.layer_attn_epilogue(li, attn, residual)
        return eng.model.final(h[-1:], residual[-1:]).float().cpu()[0]

    prompt = [2, 4, 6, 8, 10]
    lg = prefill_logits(gpu_engine, prompt)
    lc = prefill_logits(cpu, prompt)
    cos = torch.nn.functional.cosine_similarity(lg, lc, dim=0)
    assert cos > 0.98, f"cosine {cos}"


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
            tok, fin, _ex = q.get_nowait()
            toks.append(tok)
            if fin:
                break
        assert len(toks) == 8


@pytest.mark.gpu
def test_forward_dense_embeddings_gpu():
    """Dense (no-KV) forward used by /v1/embeddings matches the
    engine's incremental path closely at the final position."""
    import torch
    from llmlb_amd.models.llama import LlamaConfig, LlamaForCausalLM
    cfg = LlamaConfig.small()
    m = LlamaForCausalLM(cfg, dtype=torch.bfloat16, device="cuda:0",
                         seed=9)
    ids = torch.randint(0, cfg.vocab, (2, 24), device="cuda:0")
    h = m.forward_dense(ids)
    assert h.shape == (2, 24, cfg.hidden)
    assert torch.isfinite(h.float()).all()
    # deterministic across calls
    h2 = m.forward_dense(ids)
    assert torch.equal(h, h2)


def test_multichunk_prefill_and_cached_prefix_gpu():
    """Exercise the P>0 prefill branch on GPU: (a) a prompt longer
    than prefill_chunk runs in chunks (chunk 2+ takes the
    repeated-KV fused masked SDPA — engine.py _prefill_chunk), and
    (b) a re-submitted prompt reuses cached prefix pages (cached_len
    > 0, same branch).  Determinism across identical runs is also
    asserted (greedy, fixed seed)."""
    assert ops.native_available(), ops.native_load_error()
    cfg = LlamaConfig.tiny()
    prompt = [(3 * i + 5) % cfg.vocab for i in range(200)]
    greedy = SamplingParams(temperature=0.0, max_new_tokens=6)

    def run():
        eng = Engine(EngineConfig(model=cfg, device="cuda:0",
                                  dtype="bfloat16", kv_cache_blocks=256,
                                  prefill_chunk=64,
                                  enable_prefix_cache=True, seed=21))
        a = eng.generate_sync(prompt, max_new_tokens=6, params=greedy)
        hits0 = eng.prefix.hit_tokens if eng.prefix else 0
        b = eng.generate_sync(prompt, max_new_tokens=6, params=greedy)
        hits1 = eng.prefix.hit_tokens if eng.prefix else 0
        return a.token_ids, b.token_ids, hits1 - hits0

    a1, b1, hits = run()
    assert len(a1) == 6 and all(0 <= t < cfg.vocab for t in a1)
    # second submission hit the prefix cache -> cached_len > 0 branch
    assert hits > 0, "prefix cache should serve the repeated prompt"
    # cached-prefix continuation must agree with the full prefill
    assert b1 == a1, (a1, b1)
    # same seed, same engine config -> deterministic
    a2, _, _ = run()
    assert a2 == a1, (a1, a2)


def test_spec_decode_verify_gpu():
    """Spec-decode verify on GPU exercises the repeated-KV fused
    masked SDPA (engine.py _spec verify).  Exact spec==greedy
    equality is a CPU contract (tests/test_spec_decode.py) where the
    verify and decode paths share one attention implementation; on
    GPU the verify SDPA and the paged-attention decode kernel differ
    in bf16 rounding, so here we assert the spec engine itself is
    deterministic, produces valid tokens, and actually drafts."""
    assert ops.native_available(), ops.native_load_error()
    cfg = LlamaConfig.tiny()
    greedy = SamplingParams(temperature=0.0, max_new_tokens=16)

    def eng(spec, seed):
        return Engine(EngineConfig(model=cfg, device="cuda:0",
                                   dtype="bfloat16", kv_cache_blocks=128,
                                   spec_ngram=spec, seed=seed))

    base = [3, 1, 4, 1, 5]
    proposed = 0
    for seed in (7, 3, 5):
        cont = eng(0, seed).generate_sync(
            base, max_new_tokens=16, params=greedy).token_ids
        prompt = base + cont + base
        e1 = eng(6, seed)
        out1 = e1.generate_sync(prompt, max_new_tokens=16,
                                params=greedy).token_ids
        out2 = eng(6, seed).generate_sync(prompt, max_new_tokens=16,
                                          params=greedy).token_ids
        assert len(out1) == 16
        assert all(0 <= t < cfg.vocab for t in out1)
        assert out1 == out2, (seed, out1, out2)
        proposed += e1.spec_proposed
        if proposed:
            break
    # whether the drafter fires depends on the (random-init) model's
    # continuation shape; across three seeds at least one must draft
    assert proposed > 0, "drafter never proposed on GPU"
