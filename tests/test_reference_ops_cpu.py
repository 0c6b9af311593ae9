"""The fp32 reference ops themselves (llmlb_amd/ops/reference.py) vs
straightforward formula implementations — these are the ground truth
the GPU kernels are tested against, so they get their own checks."""
import math

import torch

from llmlb_amd.ops import reference as R


def test_rmsnorm_formula():
    x = torch.randn(4, 64)
    w = torch.randn(64)
    out, _ = R.rmsnorm(x, w, 1e-5, None)
    manual = x / torch.sqrt((x * x).mean(-1, keepdim=True) + 1e-5) * w
    assert torch.allclose(out, manual, atol=1e-5)


def test_rmsnorm_fused_residual_semantics():
    x = torch.randn(3, 32)
    res = torch.randn(3, 32)
    w = torch.ones(32)
    out, new_res = R.rmsnorm(x, w, 1e-6, res)
    assert torch.allclose(new_res, x + res, atol=1e-6)
    manual = new_res / torch.sqrt(
        (new_res * new_res).mean(-1, keepdim=True) + 1e-6)
    assert torch.allclose(out, manual, atol=1e-5)


def test_swiglu_formula():
    gu = torch.randn(5, 16)
    out = R.swiglu(gu)
    g, u = gu[:, :8], gu[:, 8:]
    manual = g * torch.sigmoid(g) * u
    assert torch.allclose(out, manual, atol=1e-5)


def test_rope_is_norm_preserving_rotation():
    q = torch.randn(6, 2, 32)
    k = torch.randn(6, 1, 32)
    cs = R.build_cos_sin_table(64, 32, 1e4)
    pos = torch.arange(6)
    q2, k2 = R.rope_apply(q.clone(), k.clone(), cs, pos)
    # a rotation preserves pairwise norms
    assert torch.allclose(q2.norm(dim=-1), q.norm(dim=-1), atol=1e-4)
    assert torch.allclose(k2.norm(dim=-1), k.norm(dim=-1), atol=1e-4)
    # position 0 is the identity
    q0, _ = R.rope_apply(q[:1].clone(), k[:1].clone(), cs,
                         torch.zeros(1, dtype=torch.long))
    assert torch.allclose(q0, q[:1], atol=1e-5)


def test_paged_attn_reference_matches_sdpa():
    torch.manual_seed(0)
    S, nq, nkv, D, BS, nb = 2, 4, 2, 32, 16, 6
    kc = torch.randn(nb, nkv, BS, D)
    vc = torch.randn(nb, nkv, BS, D)
    q = torch.randn(S, nq, D)
    bt = torch.tensor([[1, 2], [3, 4]], dtype=torch.int32)
    sl = torch.tensor([20, 30], dtype=torch.int32)
    out = R.paged_attn_decode(q, kc, vc, bt, sl, D ** -0.5)
    for s in range(S):
        L = int(sl[s])
        ks = torch.cat([kc[bt[s, i]].transpose(0, 1)
                        for i in range((L + BS - 1) // BS)], 0)[:L]
        vs = torch.cat([vc[bt[s, i]].transpose(0, 1)
                        for i in range((L + BS - 1) // BS)], 0)[:L]
        ref = torch.nn.functional.scaled_dot_product_attention(
            q[s].unsqueeze(0).transpose(0, 1),
            ks.transpose(0, 1).repeat_interleave(nq // nkv, 0),
            vs.transpose(0, 1).repeat_interleave(nq // nkv, 0),
        ).transpose(0, 1).squeeze(0)
        assert torch.allclose(out[s], ref, atol=1e-4), s


def test_fp4_quantizer_round_trip():
    """MXFP4 quantize -> dequantize: block-bounded error and exact
    representability of in-grid values."""
    import torch
    from llmlb_amd.ops import reference as R
    torch.manual_seed(5)
    w = torch.randn(128, 1024) * 0.05
    w4, sc = R.quantize_fp4_weight(w)
    assert w4.shape == (128, 512) and sc.shape == (128, 32)
    wd = R.dequantize_fp4_weight(w4, sc, 1024)
    # per-element error bounded by half an e2m1 step at the block scale
    S = 1024 // 128
    e = sc.view(128, S // 4, 4, 4).reshape(128, S, 4).float() - 127.0
    scale = torch.pow(2.0, e)                    # [M, S, 4]
    kl = R._fp4_k_local().reshape(-1)
    verr = (wd - w).view(128, S, 128)[:, :, kl].view(128, S, 4, 32).abs()
    # largest e2m1 gap is 2.0 (4 -> 6); half-gap = 1.0 x scale
    assert (verr <= scale.unsqueeze(-1) * 1.0 + 1e-6).all()
    # exact grid values survive the round trip
    g = torch.tensor([0.0, 0.5, 1.0, -1.5, 2.0, -3.0, 4.0, 6.0])
    wg = g.repeat(2, 64)                         # [2, 512]
    q4, qs = R.quantize_fp4_weight(wg)
    assert torch.allclose(R.dequantize_fp4_weight(q4, qs, 512), wg)


def test_fp8_prefill_gating_logic(monkeypatch):
    """The fp8-prefill dispatch is a pure lookup with hard gates; all
    early-return paths are CUDA-free and must reject without touching
    the fp8 kernels: row caps, the P==0 chunk flag, untuned shapes,
    and nearest-bucket fallback (profiles/r02_fp8_prefill_ab.md)."""
    import llmlb_amd.ops as ops

    calls = []
    monkeypatch.setattr(ops, "_scaled_mm_raw",
                        lambda x, wq, wsc: calls.append(x.shape) or "HIT")
    # the engine sets the chunk flag per prefill chunk and may leave
    # it False after a P>0 chunk (harmless outside prefill); pin it
    monkeypatch.setattr(ops, "_PREFILL_FP8_OK", True)
    wq = torch.zeros(64, 32, dtype=torch.uint8)
    wsc = torch.ones(64)
    w8 = (wq, wsc)

    def x(rows, k=32):
        return torch.zeros(rows, k, dtype=torch.bfloat16)

    # no w8 copies (HBM guard skipped them) -> bf16
    assert ops.scaled_mm_prefill(x(1024), None) is None
    # too few rows to have been representative at tune time
    assert ops.scaled_mm_prefill(x(256), w8) is None
    # above the host-launch-bound row cap
    assert ops.scaled_mm_prefill(x(ops.MAX_FP8_PREFILL_ROWS + 1),
                                 w8) is None
    # K not 16-aligned
    assert ops.scaled_mm_prefill(x(1024, 24), (torch.zeros(
        64, 24, dtype=torch.uint8), wsc)) is None
    # untuned shape -> bf16 (never measures on the serving path)
    monkeypatch.setattr(ops, "_PREFILL_CHOICE", {})
    assert ops.scaled_mm_prefill(x(1024), w8) is None
    # tuned: exact bucket hit
    monkeypatch.setattr(ops, "_PREFILL_CHOICE",
                        {(64, 32, (1024).bit_length()): True})
    assert ops.scaled_mm_prefill(x(1024), w8) == "HIT"
    # nearest-bucket fallback (2048 rows, only the 1024 bucket tuned)
    assert ops.scaled_mm_prefill(x(2048), w8) == "HIT"
    # tuner picked bf16 for the shape
    monkeypatch.setattr(ops, "_PREFILL_CHOICE",
                        {(64, 32, (1024).bit_length()): False})
    assert ops.scaled_mm_prefill(x(1024), w8) is None
    # continuation chunk (P>0): engine clears the flag
    monkeypatch.setattr(ops, "_PREFILL_CHOICE",
                        {(64, 32, (1024).bit_length()): True})
    monkeypatch.setattr(ops, "_PREFILL_FP8_OK", False)
    assert ops.scaled_mm_prefill(x(1024), w8) is None
    assert calls == [(1024, 32), (2048, 32)]


def test_tune_prefill_shapes_contract(monkeypatch):
    """Engine-init tuning fills one decision per (M, K, bucket) for
    every distinct weight shape, dedupes repeated shapes, and skips
    non-16-aligned shapes; a tuner failure disables the arm instead
    of raising into engine init."""
    import llmlb_amd.ops as ops

    monkeypatch.setattr(ops, "_PREFILL_CHOICE", {})
    monkeypatch.setattr(ops, "_tune_prefill",
                        lambda x, w, wq, wsc: x.shape[0] <= 4096)
    def pair(m, k):
        return (torch.zeros(m, k, dtype=torch.bfloat16),
                (torch.zeros(m, k, dtype=torch.uint8), torch.ones(m)))

    n = ops.tune_prefill_shapes(
        [pair(64, 32), pair(64, 32), pair(48, 64), pair(50, 24)], "cpu")
    # 2 distinct aligned shapes x all buckets; (50, 24) skipped (K%16)
    assert n == 2 * len(ops.TUNE_BUCKETS)
    for (m, k) in [(64, 32), (48, 64)]:
        for t in ops.TUNE_BUCKETS:
            assert ops._PREFILL_CHOICE[(m, k, t.bit_length())] == \
                (t <= 4096)
    assert (50, 24, ops.TUNE_BUCKETS[0].bit_length()) \
        not in ops._PREFILL_CHOICE

    # tuner failure -> arm disabled for the run, no raise
    monkeypatch.setattr(ops, "_PREFILL_CHOICE", {})
    monkeypatch.setattr(ops, "_FP8_PREFILL", True)
    def boom(x, w, wq, wsc):
        raise RuntimeError("no fp8 support")
    monkeypatch.setattr(ops, "_tune_prefill", boom)
    ops.tune_prefill_shapes([pair(64, 32)], "cpu")
    assert not ops._FP8_PREFILL
