"""Pure-PyTorch fp32 reference implementations of the HIP kernels.

Used (a) as the CPU fallback so the engine/tests run without a GPU, and
(b) as the numerics reference the GPU tests compare the HIP kernels
against (per-op, fp32 math).
"""
from __future__ import annotations

import torch


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float,
            residual: torch.Tensor | None = None):
    """Returns (out, updated_residual). fp32 accumulation, cast back."""
    orig_dtype = x.dtype
    xf = x.float()
    if residual is not None:
        xf = xf + residual.float()
        residual = xf.to(orig_dtype)
        # the residual stream is carried in the storage dtype (bf16), and
        # the norm reads the stored stream — match the fused kernel
        xf = residual.float()
    var = xf.pow(2).mean(dim=-1, keepdim=True)
    out = (xf * torch.rsqrt(var + eps)) * weight.float()
    return out.to(orig_dtype), residual


def rope_apply(q: torch.Tensor, k: torch.Tensor, cos_sin: torch.Tensor,
               positions: torch.Tensor):
    """Rotate-half RoPE in fp32.

    q: [T, n_q, D], k: [T, n_kv, D]; cos_sin: [max_pos, D] fp32 with
    cos[0:D/2] | sin[0:D/2] per row; positions: [T] int.
    Returns rotated (q, k) in the input dtype.
    """
    D = q.shape[-1]
    half = D // 2
    cs = cos_sin[positions]            # [T, D]
    cos = cs[:, :half].unsqueeze(1)    # [T, 1, half]
    sin = cs[:, half:].unsqueeze(1)

    def rot(x):
        xf = x.float()
        x1, x2 = xf[..., :half], xf[..., half:]
        o1 = x1 * cos - x2 * sin
        o2 = x2 * cos + x1 * sin
        return torch.cat([o1, o2], dim=-1).to(x.dtype)

    return rot(q), rot(k)


def build_cos_sin_table(max_pos: int, head_dim: int, rope_theta: float,
                        device="cpu") -> torch.Tensor:
    """[max_pos, head_dim] fp32: cos(f_i * p) | sin(f_i * p)."""
    half = head_dim // 2
    inv_freq = 1.0 / (
        rope_theta ** (torch.arange(0, half, dtype=torch.float64) / half)
    )
    pos = torch.arange(max_pos, dtype=torch.float64)
    angles = torch.outer(pos, inv_freq)  # [max_pos, half]
    table = torch.cat([angles.cos(), angles.sin()], dim=-1).float()
    return table.to(device)


def swiglu(gateup: torch.Tensor) -> torch.Tensor:
    """gateup: [..., 2*inter] (gate | up) -> silu(gate) * up."""
    inter = gateup.shape[-1] // 2
    g = gateup[..., :inter].float()
    u = gateup[..., inter:].float()
    return (torch.nn.functional.silu(g) * u).to(gateup.dtype)


def kv_append(kcache: torch.Tensor, vcache: torch.Tensor, k: torch.Tensor,
              v: torch.Tensor, slots: torch.Tensor, block_size: int):
    """Scatter k/v rows ([T, n_kv, D]) into paged caches
    ([num_blocks, n_kv, block_size, D]) at flat slot ids."""
    valid = slots >= 0
    idx = slots[valid].long()
    blk = idx // block_size
    off = idx % block_size
    # advanced indices (blk, off) broadcast together; slice dims follow
    kcache[blk, :, off, :] = k[valid]
    vcache[blk, :, off, :] = v[valid]


def quantize_fp8(x: torch.Tensor):
    """Per-(row=last-2-dims-leading) e4m3 quantization used by the FP8
    KV cache: one scale per [..., D] vector, scale = amax/448."""
    amax = x.float().abs().amax(dim=-1, keepdim=True)
    scale = torch.where(amax > 0, amax / 448.0, torch.ones_like(amax))
    q = (x.float() / scale).to(torch.float8_e4m3fn)
    return q.view(torch.uint8), scale.squeeze(-1).float()


def dequant_fp8_cache(cache_u8: torch.Tensor,
                      scale: torch.Tensor) -> torch.Tensor:
    """[nb, n_kv, bs, D] uint8 + [nb, n_kv, bs] scale -> bf16."""
    f = cache_u8.view(torch.float8_e4m3fn).to(torch.float32)
    return (f * scale.unsqueeze(-1)).to(torch.bfloat16) \
        if cache_u8.is_cuda else (f * scale.unsqueeze(-1))


def kv_append_fp8(kcache, vcache, k_scale, v_scale, k, v, slots,
                  block_size):
    valid = slots >= 0
    idx = slots[valid].long()
    blk = idx // block_size
    off = idx % block_size
    kq, ks = quantize_fp8(k[valid])
    vq, vs = quantize_fp8(v[valid])
    kcache[blk, :, off, :] = kq
    vcache[blk, :, off, :] = vq
    k_scale[blk, :, off] = ks
    v_scale[blk, :, off] = vs


def paged_attn_decode(q: torch.Tensor, kcache: torch.Tensor,
                      vcache: torch.Tensor, block_tables: torch.Tensor,
                      seq_lens: torch.Tensor, scale: float) -> torch.Tensor:
    """Reference decode attention over the paged cache, fp32 math.

    q: [S, n_q, D]; caches: [num_blocks, n_kv, block_size, D];
    block_tables: [S, max_blocks] int32; seq_lens: [S].
    Returns [S, n_q, D] in q.dtype.
    """
    S, n_q, D = q.shape
    n_kv = kcache.shape[1]
    block_size = kcache.shape[2]
    group = n_q // n_kv
    out = torch.empty_like(q)
    for s in range(S):
        L = int(seq_lens[s])
        nb = (L + block_size - 1) // block_size
        blks = block_tables[s, :nb].long()
        # gather [L, n_kv, D]
        k = kcache[blks].transpose(1, 2).reshape(nb * block_size, n_kv, D)[:L]
        v = vcache[blks].transpose(1, 2).reshape(nb * block_size, n_kv, D)[:L]
        qf = q[s].float()                       # [n_q, D]
        kf = k.float()                          # [L, n_kv, D]
        vf = v.float()
        for h in range(n_q):
            kvh = h // group
            scores = (kf[:, kvh] @ qf[h]) * scale          # [L]
            w = torch.softmax(scores, dim=0)
            out[s, h] = (w @ vf[:, kvh]).to(q.dtype)
    return out


def quantize_rows_fp8(x: "torch.Tensor"):
    """Per-row e4m3 quantization reference: returns (bytes u8, scale
    f32 [rows]); mirrors the GPU kernels' amax/448 scaling."""
    import torch
    xf = x.float()
    amax = xf.abs().amax(dim=-1)
    scale = torch.where(amax > 0, amax / 448.0, torch.ones_like(amax))
    q = (xf / scale[:, None]).clamp(-448, 448).to(torch.float8_e4m3fn)
    return q.view(torch.uint8).contiguous(), scale.contiguous()


def w8a8_matmul(xq, x_scale, wq, w_scale):
    """Dequantized fp32 reference of the W8A8 GEMM."""
    import torch
    xd = xq.view(torch.float8_e4m3fn).float() * x_scale[:, None]
    wd = wq.view(torch.float8_e4m3fn).float() * w_scale[:, None]
    return xd @ wd.T


# ---------------------------------------------------------------------------
# MXFP4 weights (W4A8): e2m1 nibbles + per-32-value E8M0 block scales in
# the EXACT register layout of v_mfma_scale_f32_16x16x128_f8f6f4 with
# blgp=4, established empirically by benchmarks/mfma128_probe.hip:
#   lane g (= lane>>4) supplies nibble (byte j, half h) holding
#   k_local = (g&1)*64 + (g>>1)*16 + (j>>3)*32 + (j&7)*2 + h
# within each 128-k superblock; its E8M0 scale operand covers exactly
# those 32 k (one MX block per lane per instruction).
# ---------------------------------------------------------------------------

_E2M1_VALS = [0.0, 0.5, 1.0, 1.5, 2.0, 3.0, 4.0, 6.0]


def _fp4_k_local():
    """[4, 16, 2] int64: k_local for (g, byte j, nibble h)."""
    g = torch.arange(4).view(4, 1, 1)
    j = torch.arange(16).view(1, 16, 1)
    h = torch.arange(2).view(1, 1, 2)
    return ((g & 1) * 64 + (g >> 1) * 16 + (j >> 3) * 32 +
            (j & 7) * 2 + h)


def quantize_fp4_weight(w):
    """[M, K] bf16/fp32 -> (w4 u8 [M, K//2], scales u8 [M, K//32]),
    both in the kernel/register layout (NOT plain row-major nibbles):
      w4[(m*S + s)*64 + g*16 + j] = nibbles of k_local(g, j, 0/1)
      scales[(m*(S//4) + c)*16 + u*4 + g] = E8M0 of block (s=c*4+u, g)
    K must be a multiple of 512 (4 superblocks per 512-k chunk)."""
    M, K = w.shape
    assert K % 512 == 0, K
    S = K // 128
    wf = w.float().view(M, S, 128)
    kl = _fp4_k_local().to(w.device)                    # [4,16,2]
    vals = wf[:, :, kl.reshape(-1)].view(M, S, 4, 32)   # [M,S,g,32]
    amax = vals.abs().amax(dim=-1)                      # [M,S,4]
    e = torch.ceil(torch.log2((amax / 6.0).clamp(min=2.0 ** -64)))
    e = e.clamp(-64, 64)
    scale = torch.pow(2.0, e)                           # [M,S,4]
    q = vals / scale.unsqueeze(-1)
    mags = torch.tensor(_E2M1_VALS, device=w.device)
    mids = (mags[1:] + mags[:-1]) / 2                   # 7 midpoints
    code = torch.bucketize(q.abs(), mids)               # [M,S,4,32] 0..7
    nib = (code | ((q < 0).long() << 3)).to(torch.uint8)
    nib = nib.view(M, S, 4, 16, 2)
    packed = nib[..., 0] | (nib[..., 1] << 4)           # [M,S,4,16]
    w4 = packed.reshape(M, K // 2).contiguous()
    ebyte = (e + 127).to(torch.uint8)                   # [M,S,4]
    scales = ebyte.view(M, S // 4, 4, 4).reshape(M, K // 32).contiguous()
    return w4, scales


def dequantize_fp4_weight(w4, scales, K):
    """Inverse of quantize_fp4_weight -> float32 [M, K]."""
    M = w4.shape[0]
    S = K // 128
    packed = w4.view(M, S, 4, 16)
    nib0 = (packed & 15).long()
    nib1 = (packed >> 4).long()
    nib = torch.stack([nib0, nib1], dim=-1)             # [M,S,4,16,2]
    mags = torch.tensor(_E2M1_VALS, device=w4.device)
    v = mags[nib & 7] * torch.where(nib >= 8,
                                    torch.tensor(-1.0), torch.tensor(1.0))
    e = scales.view(M, S // 4, 4, 4).reshape(M, S, 4).float() - 127.0
    v = v.view(M, S, 4, 32) * torch.pow(2.0, e).unsqueeze(-1)
    kl = _fp4_k_local().to(w4.device).reshape(-1)       # [128]
    out = torch.empty(M, S, 128)
    out[:, :, kl] = v.view(M, S, 128)
    return out.view(M, K)


def w4a8_matmul(xq, x_scale, w4, w4_scales):
    """Reference: dequantized fp32 matmul for the W4A8 kernel."""
    K = xq.shape[1]
    xf = xq.view(torch.float8_e4m3fn).float() * x_scale[:, None].float()
    wf = dequantize_fp4_weight(w4.cpu(), w4_scales.cpu(), K)
    return xf.cpu() @ wf.T
