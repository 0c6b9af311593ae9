// llmlb_amd HIP kernels — hand-written for MI355X (gfx950, CDNA4).
//
// These are the inference-backend hot ops for the Llama-3 family served
// behind the gateway (BASELINE.json configs 2-4).  The reference gateway
// (akiojin/llmlb) hosts no model kernels; this file is the
// new MI355X-native backend surface.
//
// Design notes (per /opt/skills/guides/cdna_hip_programming.md):
//  * wave = 64 lanes everywhere; block sizes are multiples of 64.
//  * memory-bound ops (rmsnorm/rope/swiglu/kv-append) load bf16 as 16 B
//    per lane (uint4 = 8 x bf16) — scalar bf16 loads are ~2x slower (G13).
//  * RoPE uses a host-precomputed cos/sin table (f32) — on-device
//    sincosf turns a memory-bound op VALU-bound (Appendix B).
//  * paged decode attention: one workgroup per (seq, kv_head, partition),
//    4 waves; GQA q-heads of the group are split across waves; K/V pages
//    are read directly from global memory (the 4 waves share the page
//    through L1/L2 — LDS staging is not needed for streamed-once data,
//    common-mistake #7); per-wave online softmax in registers with
//    4-lanes-per-token shuffle reduction; fp32 accumulation.
//
// Compile: hipcc --offload-arch=gfx950 -O3 -std=c++17 -shared -fPIC
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp8.h>
#include <math.h>
#include <stdint.h>

#define WAVE 64

typedef __hip_bfloat16 bf16;

__device__ __forceinline__ float bf2f(unsigned short u) {
  union { unsigned int i; float f; } c;
  c.i = ((unsigned int)u) << 16;
  return c.f;
}

// round-to-nearest-even f32 -> bf16 (matches torch's conversion)
__device__ __forceinline__ unsigned short f2bf(float f) {
  union { float f; unsigned int i; } c;
  c.f = f;
  unsigned int x = c.i;
  unsigned int rounding_bias = 0x7FFF + ((x >> 16) & 1);
  x += rounding_bias;
  return (unsigned short)(x >> 16);
}

// unpack a uint4 (8 bf16) into 8 floats
__device__ __forceinline__ void unpack8(const uint4 v, float* f) {
  f[0] = bf2f(v.x & 0xffff);        f[1] = bf2f(v.x >> 16);
  f[2] = bf2f(v.y & 0xffff);        f[3] = bf2f(v.y >> 16);
  f[4] = bf2f(v.z & 0xffff);        f[5] = bf2f(v.z >> 16);
  f[6] = bf2f(v.w & 0xffff);        f[7] = bf2f(v.w >> 16);
}

__device__ __forceinline__ uint4 pack8(const float* f) {
  uint4 v;
  v.x = (unsigned int)f2bf(f[0]) | ((unsigned int)f2bf(f[1]) << 16);
  v.y = (unsigned int)f2bf(f[2]) | ((unsigned int)f2bf(f[3]) << 16);
  v.z = (unsigned int)f2bf(f[4]) | ((unsigned int)f2bf(f[5]) << 16);
  v.w = (unsigned int)f2bf(f[6]) | ((unsigned int)f2bf(f[7]) << 16);
  return v;
}

// ---------------------------------------------------------------------------
// RMSNorm (optionally fused with residual add):
//   if residual != nullptr:  residual += x;  x' = residual  (updated in place)
//   out = x' * rsqrt(mean(x'^2) + eps) * weight
// One 256-thread block per token row; fp32 accumulation; bf16x8 loads.
// ---------------------------------------------------------------------------
template <bool FUSE_RESIDUAL>
__global__ void rmsnorm_kernel(bf16* __restrict__ out,
                               const bf16* __restrict__ in,
                               const bf16* __restrict__ weight,
                               bf16* __restrict__ residual,
                               int hidden, float eps) {
  const int row = blockIdx.x;
  const uint4* inv = reinterpret_cast<const uint4*>(in + (size_t)row * hidden);
  uint4* resv = FUSE_RESIDUAL
      ? reinterpret_cast<uint4*>(residual + (size_t)row * hidden) : nullptr;
  const uint4* wv = reinterpret_cast<const uint4*>(weight);
  uint4* outv = reinterpret_cast<uint4*>(out + (size_t)row * hidden);
  const int nvec = hidden >> 3;  // uint4 = 8 bf16

  float ssq = 0.f;
  // pass 1: (residual add +) sum of squares.  Rows are at most a few K
  // elements (hidden 4096..16384) so keep the row in registers when it
  // fits one iteration, else re-read (L1/L2-resident on pass 2).
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    uint4 v = inv[i];
    float f[8];
    unpack8(v, f);
    if (FUSE_RESIDUAL) {
      float r[8];
      unpack8(resv[i], r);
#pragma unroll
      for (int j = 0; j < 8; ++j) f[j] += r[j];
      uint4 packed = pack8(f);
      resv[i] = packed;  // residual stream carries the sum forward
      // the norm consumes the STORED (bf16-rounded) residual, so the
      // sum of squares must use the rounded values too (pass 2 re-reads
      // resv) — otherwise rrms is computed against different data
      unpack8(packed, f);
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) ssq += f[j] * f[j];
  }
  // wave reduce then cross-wave via LDS
  for (int off = 32; off > 0; off >>= 1) ssq += __shfl_xor(ssq, off, WAVE);
  __shared__ float warp_ssq[4];
  const int wid = threadIdx.x / WAVE;
  if ((threadIdx.x & (WAVE - 1)) == 0) warp_ssq[wid] = ssq;
  __syncthreads();
  const int nw = blockDim.x / WAVE;
  float total = 0.f;
#pragma unroll
  for (int w = 0; w < 4; ++w) total += (w < nw) ? warp_ssq[w] : 0.f;
  const float rrms = rsqrtf(total / hidden + eps);

  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    uint4 v = FUSE_RESIDUAL ? resv[i] : inv[i];
    float f[8], w[8];
    unpack8(v, f);
    unpack8(wv[i], w);
#pragma unroll
    for (int j = 0; j < 8; ++j) f[j] = f[j] * rrms * w[j];
    outv[i] = pack8(f);
  }
}

// ---------------------------------------------------------------------------
// RoPE (GPT-NeoX rotate-half, Llama-3 style), fused for q and k.
//   q: [tokens, n_q * D], k: [tokens, n_kv * D], bf16, updated in place.
//   cos_sin: f32 [max_pos, D]  (cos[0:D/2] then sin[0:D/2] per position)
// One wave-group slice per (token, head); each thread rotates 2 pairs
// via 2x bf16 loads; cos/sin from the table (f32).
// ---------------------------------------------------------------------------
__global__ void rope_kernel(bf16* __restrict__ q, bf16* __restrict__ k,
                            const float* __restrict__ cos_sin,
                            const int* __restrict__ positions,
                            int n_tokens, int n_q, int n_kv, int D,
                            int q_rstride, int k_rstride) {
  const int half = D / 2;
  const int heads = n_q + n_kv;
  // one thread handles 4 rotation pairs (8 bf16 = 16B from each half)
  const int pairs_per_thread = 4;
  const int threads_per_head = half / pairs_per_thread;
  const long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long total = (long long)n_tokens * heads * threads_per_head;
  if (idx >= total) return;
  const int t = idx / (heads * threads_per_head);
  const int rem = idx % (heads * threads_per_head);
  const int h = rem / threads_per_head;
  const int j0 = (rem % threads_per_head) * pairs_per_thread;
  const int pos = positions[t];

  bf16* base = (h < n_q)
      ? (q + (size_t)t * q_rstride + (size_t)h * D)
      : (k + (size_t)t * k_rstride + (size_t)(h - n_q) * D);
  // rotate-half: (x1, x2) at (j, j+half) -> (x1*c - x2*s, x2*c + x1*s)
  float x1[4], x2[4];
  uint2 lo = *reinterpret_cast<uint2*>(base + j0);
  uint2 hi = *reinterpret_cast<uint2*>(base + half + j0);
  x1[0] = bf2f(lo.x & 0xffff); x1[1] = bf2f(lo.x >> 16);
  x1[2] = bf2f(lo.y & 0xffff); x1[3] = bf2f(lo.y >> 16);
  x2[0] = bf2f(hi.x & 0xffff); x2[1] = bf2f(hi.x >> 16);
  x2[2] = bf2f(hi.y & 0xffff); x2[3] = bf2f(hi.y >> 16);
  const float* cs = cos_sin + (size_t)pos * D;
  float4 c = *reinterpret_cast<const float4*>(cs + j0);
  float4 s = *reinterpret_cast<const float4*>(cs + half + j0);
  float o1[4], o2[4];
  o1[0] = x1[0] * c.x - x2[0] * s.x;  o2[0] = x2[0] * c.x + x1[0] * s.x;
  o1[1] = x1[1] * c.y - x2[1] * s.y;  o2[1] = x2[1] * c.y + x1[1] * s.y;
  o1[2] = x1[2] * c.z - x2[2] * s.z;  o2[2] = x2[2] * c.z + x1[2] * s.z;
  o1[3] = x1[3] * c.w - x2[3] * s.w;  o2[3] = x2[3] * c.w + x1[3] * s.w;
  lo.x = (unsigned)f2bf(o1[0]) | ((unsigned)f2bf(o1[1]) << 16);
  lo.y = (unsigned)f2bf(o1[2]) | ((unsigned)f2bf(o1[3]) << 16);
  hi.x = (unsigned)f2bf(o2[0]) | ((unsigned)f2bf(o2[1]) << 16);
  hi.y = (unsigned)f2bf(o2[2]) | ((unsigned)f2bf(o2[3]) << 16);
  *reinterpret_cast<uint2*>(base + j0) = lo;
  *reinterpret_cast<uint2*>(base + half + j0) = hi;
}

// ---------------------------------------------------------------------------
// SwiGLU: out[t, i] = silu(gateup[t, i]) * gateup[t, inter + i]
// gateup: [tokens, 2*inter] (gate | up), bf16.  Grid-stride, bf16x8.
// ---------------------------------------------------------------------------
__global__ void swiglu_kernel(bf16* __restrict__ out,
                              const bf16* __restrict__ gateup,
                              long long n_tokens, int inter) {
  const int nvec = inter >> 3;
  const long long total = n_tokens * nvec;
  for (long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       idx < total; idx += (long long)gridDim.x * blockDim.x) {
    const long long t = idx / nvec;
    const int i = idx % nvec;
    const uint4* row = reinterpret_cast<const uint4*>(gateup + t * 2 * inter);
    uint4 g = row[i];
    uint4 u = row[nvec + i];
    float gf[8], uf[8];
    unpack8(g, gf);
    unpack8(u, uf);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float x = gf[j];
      float silu = x / (1.f + __expf(-x));
      gf[j] = silu * uf[j];
    }
    reinterpret_cast<uint4*>(out + t * inter)[i] = pack8(gf);
  }
}

// ---------------------------------------------------------------------------
// KV append (reshape_and_cache): scatter per-token K/V rows into the paged
// cache.  cache layout: [num_blocks, n_kv, block_size, D] bf16.
// slots[t] = block_id * block_size + offset   (-1 = skip)
// One thread per 8 contiguous elements.
// ---------------------------------------------------------------------------

// ---------------------------------------------------------------------------
// Fused RoPE + KV append: one launch rotates q (in place), rotates k
// (in place AND straight into the paged K cache) and copies v into the
// paged V cache — replaces the rope_kernel + kv_append_kernel pair on
// the decode hot path (two ~5 us launches per layer -> one).
// Thread ranges: [0, R) = rope work (q then k heads), [R, R+V) = v copy.
// ---------------------------------------------------------------------------
__global__ void rope_kv_append_kernel(
    bf16* __restrict__ q, bf16* __restrict__ k,
    const bf16* __restrict__ v, const float* __restrict__ cos_sin,
    const int* __restrict__ positions, bf16* __restrict__ kcache,
    bf16* __restrict__ vcache, const int* __restrict__ slots,
    int n_tokens, int n_q, int n_kv, int D, int block_size,
    int q_rstride, int k_rstride, int v_rstride) {
  const int half = D / 2;
  const int heads = n_q + n_kv;
  const int pairs_per_thread = 4;
  const int tph = half / pairs_per_thread;       // threads per head
  const long long rope_total = (long long)n_tokens * heads * tph;
  const int nvec = D >> 3;
  const long long v_total = (long long)n_tokens * n_kv * nvec;
  const long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= rope_total + v_total) return;

  if (idx >= rope_total) {                       // ---- v copy ----
    const long long vi = idx - rope_total;
    const int t = vi / (n_kv * nvec);
    const int rem = vi % (n_kv * nvec);
    const int h = rem / nvec;
    const int i = rem % nvec;
    const int slot = slots[t];
    if (slot < 0) return;
    const size_t dst = (((size_t)(slot / block_size) * n_kv + h) *
                            block_size +
                        slot % block_size) * D;
    reinterpret_cast<uint4*>(vcache + dst)[i] =
        reinterpret_cast<const uint4*>(
            v + (size_t)t * v_rstride + (size_t)h * D)[i];
    return;
  }
  // ---- rope (q heads, then k heads which also land in the cache) ----
  const int t = idx / (heads * tph);
  const int rem = idx % (heads * tph);
  const int h = rem / tph;
  const int j0 = (rem % tph) * pairs_per_thread;
  const int pos = positions[t];
  const bool is_k = h >= n_q;
  bf16* base = is_k ? (k + (size_t)t * k_rstride +
                       (size_t)(h - n_q) * D)
                    : (q + (size_t)t * q_rstride + (size_t)h * D);
  float x1[4], x2[4];
  uint2 lo = *reinterpret_cast<uint2*>(base + j0);
  uint2 hi = *reinterpret_cast<uint2*>(base + half + j0);
  x1[0] = bf2f(lo.x & 0xffff); x1[1] = bf2f(lo.x >> 16);
  x1[2] = bf2f(lo.y & 0xffff); x1[3] = bf2f(lo.y >> 16);
  x2[0] = bf2f(hi.x & 0xffff); x2[1] = bf2f(hi.x >> 16);
  x2[2] = bf2f(hi.y & 0xffff); x2[3] = bf2f(hi.y >> 16);
  const float* cs = cos_sin + (size_t)pos * D;
  float4 c = *reinterpret_cast<const float4*>(cs + j0);
  float4 sn = *reinterpret_cast<const float4*>(cs + half + j0);
  float o1[4], o2[4];
  o1[0] = x1[0] * c.x - x2[0] * sn.x;  o2[0] = x2[0] * c.x + x1[0] * sn.x;
  o1[1] = x1[1] * c.y - x2[1] * sn.y;  o2[1] = x2[1] * c.y + x1[1] * sn.y;
  o1[2] = x1[2] * c.z - x2[2] * sn.z;  o2[2] = x2[2] * c.z + x1[2] * sn.z;
  o1[3] = x1[3] * c.w - x2[3] * sn.w;  o2[3] = x2[3] * c.w + x1[3] * sn.w;
  lo.x = (unsigned)f2bf(o1[0]) | ((unsigned)f2bf(o1[1]) << 16);
  lo.y = (unsigned)f2bf(o1[2]) | ((unsigned)f2bf(o1[3]) << 16);
  hi.x = (unsigned)f2bf(o2[0]) | ((unsigned)f2bf(o2[1]) << 16);
  hi.y = (unsigned)f2bf(o2[2]) | ((unsigned)f2bf(o2[3]) << 16);
  *reinterpret_cast<uint2*>(base + j0) = lo;
  *reinterpret_cast<uint2*>(base + half + j0) = hi;
  if (is_k) {
    const int slot = slots[t];
    if (slot < 0) return;
    const size_t dst = (((size_t)(slot / block_size) * n_kv +
                         (h - n_q)) * block_size +
                        slot % block_size) * D;
    *reinterpret_cast<uint2*>(kcache + dst + j0) = lo;
    *reinterpret_cast<uint2*>(kcache + dst + half + j0) = hi;
  }
}

__global__ void kv_append_kernel(bf16* __restrict__ kcache,
                                 bf16* __restrict__ vcache,
                                 const bf16* __restrict__ k,
                                 const bf16* __restrict__ v,
                                 const int* __restrict__ slots,
                                 int n_tokens, int n_kv, int D,
                                 int block_size, int k_rstride,
                                 int v_rstride) {
  const int nvec = D >> 3;
  const long long total = (long long)n_tokens * n_kv * nvec;
  const long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= total) return;
  const int t = idx / (n_kv * nvec);
  const int rem = idx % (n_kv * nvec);
  const int h = rem / nvec;
  const int i = rem % nvec;
  const int slot = slots[t];
  if (slot < 0) return;
  const int blk = slot / block_size;
  const int off = slot % block_size;
  const size_t dst = (((size_t)blk * n_kv + h) * block_size + off) * D;
  reinterpret_cast<uint4*>(kcache + dst)[i] =
      reinterpret_cast<const uint4*>(
          k + (size_t)t * k_rstride + (size_t)h * D)[i];
  reinterpret_cast<uint4*>(vcache + dst)[i] =
      reinterpret_cast<const uint4*>(
          v + (size_t)t * v_rstride + (size_t)h * D)[i];
}

// ---------------------------------------------------------------------------
// FP8 (OCP e4m3) KV path: K/V stored as e4m3 bytes with one fp32
// scale per (page, head, token) — half the attention bandwidth and
// double the resident context.  gfx950's FP8 is OCP e4m3fn (the MI300X
// fnuz variant is gone), so bytes round-trip exactly with torch's
// float8_e4m3fn.
// ---------------------------------------------------------------------------
typedef __attribute__((ext_vector_type(2))) float f32x2_t;

// one dword of packed e4m3 -> 4 floats (gfx950 native converts)
__device__ __forceinline__ void fp8x4_to_f32(unsigned int d, float* o) {
  f32x2_t lo = __builtin_amdgcn_cvt_pk_f32_fp8(d, false);
  f32x2_t hi = __builtin_amdgcn_cvt_pk_f32_fp8(d, true);
  o[0] = lo.x; o[1] = lo.y; o[2] = hi.x; o[3] = hi.y;
}


// round-to-nearest-even f32 -> e4m3 byte (input pre-scaled to +-448)
__device__ __forceinline__ unsigned char f32_to_e4m3(float f) {
  __hip_fp8_e4m3 q(f);
  return q.__x;
}

// one wave per (token, head): amax over D, scale = amax/448, quantize
__global__ __launch_bounds__(256)
void kv_append_fp8_kernel(unsigned char* __restrict__ kcache,
                          unsigned char* __restrict__ vcache,
                          float* __restrict__ k_scale,
                          float* __restrict__ v_scale,
                          const bf16* __restrict__ k,
                          const bf16* __restrict__ v,
                          const int* __restrict__ slots,
                          int n_tokens, int n_kv, int D, int block_size,
                          int k_rstride, int v_rstride) {
  const int wave_id = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  if (wave_id >= n_tokens * n_kv) return;
  const int t = wave_id / n_kv;
  const int h = wave_id % n_kv;
  const int slot = slots[t];
  if (slot < 0) return;
  const int blk = slot / block_size;
  const int off = slot % block_size;
  const int per_lane = D / WAVE;               // 2 at D=128
  const size_t dst = (((size_t)blk * n_kv + h) * block_size + off) * D;
  const size_t sidx = ((size_t)blk * n_kv + h) * block_size + off;
#pragma unroll
  for (int which = 0; which < 2; ++which) {
    const bf16* src = which ? v : k;
    const int rstride = which ? v_rstride : k_rstride;
    unsigned char* cache = which ? vcache : kcache;
    float* scale_out = which ? v_scale : k_scale;
    float x[4];
    float amax = 0.f;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      if (i < per_lane) {
        x[i] = bf2f(__builtin_bit_cast(
            unsigned short,
            src[(size_t)t * rstride + (size_t)h * D + lane * per_lane +
                i]));
        amax = fmaxf(amax, fabsf(x[i]));
      }
    }
#pragma unroll
    for (int o = 32; o > 0; o >>= 1)
      amax = fmaxf(amax, __shfl_xor(amax, o, WAVE));
    const float scale = amax > 0.f ? amax / 448.f : 1.f;
    const float inv = 1.f / scale;
#pragma unroll
    for (int i = 0; i < 4; ++i)
      if (i < per_lane)
        cache[dst + lane * per_lane + i] = f32_to_e4m3(x[i] * inv);
    if (lane == 0) scale_out[sidx] = scale;
  }
}

// ---------------------------------------------------------------------------
// Paged decode attention (one query token per sequence), GQA, bf16 KV.
//
//   q:        [num_seqs, n_q, D] bf16
//   kcache:   [num_blocks, n_kv, block_size, D] bf16   (block_size = 16)
//   block_tables: [num_seqs, max_blocks] int32
//   seq_lens: [num_seqs] int32 (length INCLUDING the current token,
//             whose K/V is already appended)
//
// Grid: (num_seqs, n_kv, n_parts); block = 4 waves (256 threads).
// Wave w handles q-heads {w, w+4, ...} of the kv-head's GQA group.
// Within a wave: 4 lanes per token x 16 tokens of a page; each lane
// computes a 32-element slice of q.k^T, shuffle-reduced over the 4-lane
// group; online softmax (m shared, l/acc lane-partial, reduced at end).
//
// When n_parts > 1 partial (m, l, acc) go to f32 workspace and
// paged_attn_reduce combines; n_parts == 1 writes bf16 output directly.
// ---------------------------------------------------------------------------
#define PA_BLOCK 16        // tokens per KV page
#define PA_WAVES 4

__global__ __launch_bounds__(PA_WAVES * WAVE)
void paged_attn_decode_kernel(
    bf16* __restrict__ out,              // [num_seqs, n_q, D]
    float* __restrict__ ws_acc,          // [num_seqs, n_q, n_parts, D]
    float* __restrict__ ws_ml,           // [num_seqs, n_q, n_parts, 2]
    const bf16* __restrict__ q,
    const bf16* __restrict__ kcache,
    const bf16* __restrict__ vcache,
    const int* __restrict__ block_tables,
    const int* __restrict__ seq_lens,
    int n_q, int n_kv, int D, int max_blocks,
    int n_parts, float scale, int q_rstride) {
  const int seq = blockIdx.x;
  const int kvh = blockIdx.y;
  const int part = blockIdx.z;
  const int seq_len = seq_lens[seq];
  const int n_blocks = (seq_len + PA_BLOCK - 1) / PA_BLOCK;
  const int blocks_per_part = (n_blocks + n_parts - 1) / n_parts;
  const int blk_begin = part * blocks_per_part;
  const int blk_end = min(blk_begin + blocks_per_part, n_blocks);
  const int group = n_q / n_kv;                 // GQA group size
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int t_local = lane >> 2;                // token within page (0..15)
  const int p = lane & 3;                       // dim-slice (32 elems each)
  const int DS = D / 4;                         // dim-slice width
  const int* btab = block_tables + (size_t)seq * max_blocks;

  // each wave serves q-heads wid, wid+PA_WAVES, ... of the group
  for (int qh_local = wid; qh_local < group; qh_local += PA_WAVES) {
    const int qh = kvh * group + qh_local;
    // load this lane's q slice (DS elems, fp32), pre-scaled
    float qf[32];
    {
      const uint4* qv = reinterpret_cast<const uint4*>(
          q + (size_t)seq * q_rstride + (size_t)qh * D + p * DS);
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        if (i * 8 < DS) {
          float f[8];
          unpack8(qv[i], f);
#pragma unroll
          for (int j = 0; j < 8; ++j) qf[i * 8 + j] = f[j] * scale;
        }
      }
    }
    float m = -INFINITY;       // running max (shared across wave)
    float l = 0.f;             // sumexp, lane-partial (own tokens)
    float acc[32];             // V-weighted accumulator, lane-partial
#pragma unroll
    for (int i = 0; i < 32; ++i) acc[i] = 0.f;

    for (int b = blk_begin; b < blk_end; ++b) {
      const int blk_id = btab[b];
      const int tok = b * PA_BLOCK + t_local;
      const bool valid = tok < seq_len;
      const bf16* kbase =
          kcache + (((size_t)blk_id * n_kv + kvh) * PA_BLOCK + t_local) * D +
          p * DS;
      // score: dot(q_slice, k_slice) then reduce over the 4-lane group
      float s = 0.f;
      {
        const uint4* kv4 = reinterpret_cast<const uint4*>(kbase);
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          if (i * 8 < DS) {
            float f[8];
            unpack8(kv4[i], f);
#pragma unroll
            for (int j = 0; j < 8; ++j) s += qf[i * 8 + j] * f[j];
          }
        }
      }
      s += __shfl_xor(s, 1, WAVE);
      s += __shfl_xor(s, 2, WAVE);
      if (!valid) s = -INFINITY;
      // block max over the 16 tokens (lane bits 2..5)
      float bm = s;
      bm = fmaxf(bm, __shfl_xor(bm, 4, WAVE));
      bm = fmaxf(bm, __shfl_xor(bm, 8, WAVE));
      bm = fmaxf(bm, __shfl_xor(bm, 16, WAVE));
      bm = fmaxf(bm, __shfl_xor(bm, 32, WAVE));
      const float m_new = fmaxf(m, bm);
      if (m_new != -INFINITY) {
        const float rescale = (m == -INFINITY) ? 0.f : __expf(m - m_new);
        const float w = valid ? __expf(s - m_new) : 0.f;
        l = l * rescale + w;
#pragma unroll
        for (int i = 0; i < 32; ++i) acc[i] *= rescale;
        // accumulate w * V[tok][p*DS..]
        const bf16* vbase =
            vcache +
            (((size_t)blk_id * n_kv + kvh) * PA_BLOCK + t_local) * D + p * DS;
        const uint4* vv4 = reinterpret_cast<const uint4*>(vbase);
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          if (i * 8 < DS) {
            float f[8];
            unpack8(vv4[i], f);
#pragma unroll
            for (int j = 0; j < 8; ++j) acc[i * 8 + j] += w * f[j];
          }
        }
        m = m_new;
      }
    }
    // reduce l and acc across the 16 token groups (butterfly)
#pragma unroll
    for (int off = 4; off <= 32; off <<= 1) {
      l += __shfl_xor(l, off, WAVE);
#pragma unroll
      for (int i = 0; i < 32; ++i)
        if (i < DS) acc[i] += __shfl_xor(acc[i], off, WAVE);
    }
    if (n_parts == 1) {
      // write normalized bf16 output: lanes of token-group 0 hold it all
      if (t_local == 0) {
        const float inv_l = (l > 0.f) ? 1.f / l : 0.f;
        float f[8];
        uint4* ov = reinterpret_cast<uint4*>(
            out + ((size_t)seq * n_q + qh) * D + p * DS);
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          if (i * 8 < DS) {
#pragma unroll
            for (int j = 0; j < 8; ++j) f[j] = acc[i * 8 + j] * inv_l;
            ov[i] = pack8(f);
          }
        }
      }
    } else {
      if (t_local == 0) {
        float* wa = ws_acc +
            (((size_t)seq * n_q + qh) * n_parts + part) * D + p * DS;
#pragma unroll
        for (int i = 0; i < 32; ++i)
          if (i < DS) wa[i] = acc[i];
        if (p == 0) {
          float* wm = ws_ml + (((size_t)seq * n_q + qh) * n_parts + part) * 2;
          wm[0] = m;
          wm[1] = l;
        }
      }
    }
  }
}


// ---------------------------------------------------------------------------
// Paged decode attention v2 — GQA-amortized (any group <= 16 via
// greedy 4/2/1 sub-group passes, D == 128).
//
// v1 gives each wave ONE query head, so the 4 waves of a workgroup
// each stream the same K/V pages (4x redundant L1/L2 traffic — the
// measured ceiling at ctx >= 1k).  v2 gives each wave ALL `group`
// query heads of its kv-head and splits the PAGES across the 4 waves
// instead: every page is read once per workgroup.  Layout: 8 lanes per
// token x 16-element slices (D=128 -> p = lane&7, DS=16); per token
// the lane computes `group` partial dots from one K slice load.  The
// per-wave online-softmax state (m shared, l/acc lane-partial) is
// combined across the 4 waves through LDS at the end (flash-decode
// within the workgroup), then across grid partitions by the shared
// paged_attn_reduce_kernel.
// ---------------------------------------------------------------------------
// GROUP <= 4 uses 8 lanes/token x 16-dim slices; GROUP 5..8 switches
// to 16 lanes/token x 8-dim slices so qf+acc stay ~128 VGPRs and the
// whole group runs in ONE pass over the pages (the r1 fallback ran
// group 8 as two GROUP=4 launches and paid the page reads twice).
template <int GROUP, bool KV8 = false>
__global__ __launch_bounds__(PA_WAVES * WAVE)
void paged_attn_decode_v2_kernel(
    bf16* __restrict__ out,              // [num_seqs, n_q, D]
    float* __restrict__ ws_acc,          // [num_seqs, n_q, n_parts, D]
    float* __restrict__ ws_ml,           // [num_seqs, n_q, n_parts, 2]
    const bf16* __restrict__ q,
    const bf16* __restrict__ kcache,     // e4m3 bytes when KV8
    const bf16* __restrict__ vcache,
    const float* __restrict__ k_scale,   // per (page, head, token); KV8
    const float* __restrict__ v_scale,
    const int* __restrict__ block_tables,
    const int* __restrict__ seq_lens,
    int n_q, int n_kv, int max_blocks, int n_parts, float scale,
    int q_rstride, int group_stride, int g_base) {
  constexpr int D = 128;
  constexpr int LPT = (GROUP > 4) ? 16 : 8;  // lanes per token
  constexpr int DS = D / LPT;            // dims per lane (16 or 8)
  constexpr int TPP = WAVE / LPT;        // tokens per pass (8 or 4)
  const int seq = blockIdx.x;
  const int kvh = blockIdx.y;
  const int part = blockIdx.z;
  const int seq_len = seq_lens[seq];
  const int n_blocks = (seq_len + PA_BLOCK - 1) / PA_BLOCK;
  // page range of this (partition, wave): partition -> waves
  const int slices = n_parts * PA_WAVES;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int slice = part * PA_WAVES + wid;
  const int per_slice = (n_blocks + slices - 1) / slices;
  const int blk_begin = slice * per_slice;
  const int blk_end = min(blk_begin + per_slice, n_blocks);
  const int t_local = lane / LPT;        // token within the pass
  const int p = lane % LPT;              // dim slice
  const int* btab = block_tables + (size_t)seq * max_blocks;

  // q fragments for every head of the group, pre-scaled
  float qf[GROUP][DS];
#pragma unroll
  for (int g = 0; g < GROUP; ++g) {
    const uint4* qv = reinterpret_cast<const uint4*>(
        q + (size_t)seq * q_rstride
        + (size_t)(kvh * group_stride + g_base + g) * D + p * DS);
#pragma unroll
    for (int i = 0; i < DS / 8; ++i) {
      float f[8];
      unpack8(qv[i], f);
#pragma unroll
      for (int j = 0; j < 8; ++j) qf[g][i * 8 + j] = f[j] * scale;
    }
  }
  float m[GROUP], l[GROUP], acc[GROUP][DS];
#pragma unroll
  for (int g = 0; g < GROUP; ++g) {
    m[g] = -INFINITY;
    l[g] = 0.f;
#pragma unroll
    for (int i = 0; i < DS; ++i) acc[g][i] = 0.f;
  }

  for (int b = blk_begin; b < blk_end; ++b) {
    const int blk_id = btab[b];
#pragma unroll
    for (int half = 0; half < PA_BLOCK / TPP; ++half) {
      const int trow = half * TPP + t_local;
      const int tok = b * PA_BLOCK + trow;
      const bool valid = tok < seq_len;
      const size_t row = ((size_t)blk_id * n_kv + kvh) * PA_BLOCK + trow;
      float kf[DS];
      if (KV8) {
        const float sc = k_scale[row];
        const unsigned char* kb =
            reinterpret_cast<const unsigned char*>(kcache) + row * D +
            p * DS;
        if (DS == 16) {
          const uint4 raw = *reinterpret_cast<const uint4*>(kb);
          fp8x4_to_f32(raw.x, kf + 0);
          fp8x4_to_f32(raw.y, kf + 4);
          fp8x4_to_f32(raw.z, kf + 8);
          fp8x4_to_f32(raw.w, kf + 12);
        } else {
          const uint2 raw = *reinterpret_cast<const uint2*>(kb);
          fp8x4_to_f32(raw.x, kf + 0);
          fp8x4_to_f32(raw.y, kf + 4);
        }
#pragma unroll
        for (int j = 0; j < DS; ++j) kf[j] *= sc;
      } else {
        const uint4* kv4 = reinterpret_cast<const uint4*>(
            kcache + row * D + p * DS);
#pragma unroll
        for (int i = 0; i < DS / 8; ++i) {
          float f[8];
          unpack8(kv4[i], f);
#pragma unroll
          for (int j = 0; j < 8; ++j) kf[i * 8 + j] = f[j];
        }
      }
      float s[GROUP];
#pragma unroll
      for (int g = 0; g < GROUP; ++g) {
        float acc_s = 0.f;
#pragma unroll
        for (int j = 0; j < DS; ++j) acc_s += qf[g][j] * kf[j];
        // reduce over the LPT-lane dim group
        acc_s += __shfl_xor(acc_s, 1, WAVE);
        acc_s += __shfl_xor(acc_s, 2, WAVE);
        acc_s += __shfl_xor(acc_s, 4, WAVE);
        if (LPT == 16) acc_s += __shfl_xor(acc_s, 8, WAVE);
        s[g] = valid ? acc_s : -INFINITY;
      }
      // block max over the 8 tokens of this pass (lane bits 3..5)
      float vf[DS];
      if (KV8) {
        const float sc = v_scale[row];
        const unsigned char* vb =
            reinterpret_cast<const unsigned char*>(vcache) + row * D +
            p * DS;
        if (DS == 16) {
          const uint4 raw = *reinterpret_cast<const uint4*>(vb);
          fp8x4_to_f32(raw.x, vf + 0);
          fp8x4_to_f32(raw.y, vf + 4);
          fp8x4_to_f32(raw.z, vf + 8);
          fp8x4_to_f32(raw.w, vf + 12);
        } else {
          const uint2 raw = *reinterpret_cast<const uint2*>(vb);
          fp8x4_to_f32(raw.x, vf + 0);
          fp8x4_to_f32(raw.y, vf + 4);
        }
#pragma unroll
        for (int j = 0; j < DS; ++j) vf[j] *= sc;
      } else {
        const uint4* vv4 = reinterpret_cast<const uint4*>(
            vcache + row * D + p * DS);
#pragma unroll
        for (int i = 0; i < DS / 8; ++i) {
          float f[8];
          unpack8(vv4[i], f);
#pragma unroll
          for (int j = 0; j < 8; ++j) vf[i * 8 + j] = f[j];
        }
      }
#pragma unroll
      for (int g = 0; g < GROUP; ++g) {
        float bm = s[g];
        if (LPT == 8) bm = fmaxf(bm, __shfl_xor(bm, 8, WAVE));
        bm = fmaxf(bm, __shfl_xor(bm, 16, WAVE));
        bm = fmaxf(bm, __shfl_xor(bm, 32, WAVE));
        const float m_new = fmaxf(m[g], bm);
        if (m_new != -INFINITY) {
          const float rescale =
              (m[g] == -INFINITY) ? 0.f : __expf(m[g] - m_new);
          const float wgt = valid ? __expf(s[g] - m_new) : 0.f;
          l[g] = l[g] * rescale + wgt;
#pragma unroll
          for (int i = 0; i < DS; ++i)
            acc[g][i] = acc[g][i] * rescale + wgt * vf[i];
          m[g] = m_new;
        }
      }
    }
  }
  // reduce l and acc across the token groups (butterfly)
#pragma unroll
  for (int off = LPT; off <= 32; off <<= 1) {
#pragma unroll
    for (int g = 0; g < GROUP; ++g) {
      l[g] += __shfl_xor(l[g], off, WAVE);
#pragma unroll
      for (int i = 0; i < DS; ++i)
        acc[g][i] += __shfl_xor(acc[g][i], off, WAVE);
    }
  }
  if (gridDim.z == 1) {
    // single grid partition: the 4 wave-slices live in THIS workgroup,
    // so combine through LDS and write bf16 directly — no global f32
    // slice round trip, no reduce kernel (saves ~8 MB/layer of ws
    // traffic + one launch at the common short-context sizes)
    __shared__ float s_acc[PA_WAVES][GROUP][D];
    __shared__ float s_ml[PA_WAVES][GROUP][2];
    if (t_local == 0) {
#pragma unroll
      for (int g = 0; g < GROUP; ++g) {
#pragma unroll
        for (int i = 0; i < DS; ++i) s_acc[wid][g][p * DS + i] = acc[g][i];
        if (p == 0) {
          s_ml[wid][g][0] = m[g];
          s_ml[wid][g][1] = l[g];
        }
      }
    }
    __syncthreads();
    if (wid == 0 && t_local == 0) {
#pragma unroll
      for (int g = 0; g < GROUP; ++g) {
        float m_tot = -INFINITY;
#pragma unroll
        for (int w2 = 0; w2 < PA_WAVES; ++w2)
          m_tot = fmaxf(m_tot, s_ml[w2][g][0]);
        float l_tot = 0.f, o[DS];
#pragma unroll
        for (int i = 0; i < DS; ++i) o[i] = 0.f;
#pragma unroll
        for (int w2 = 0; w2 < PA_WAVES; ++w2) {
          const float mw = s_ml[w2][g][0];
          const float sc = (mw == -INFINITY) ? 0.f : __expf(mw - m_tot);
          l_tot += s_ml[w2][g][1] * sc;
#pragma unroll
          for (int i = 0; i < DS; ++i)
            o[i] += s_acc[w2][g][p * DS + i] * sc;
        }
        const float inv = l_tot > 0.f ? 1.f / l_tot : 0.f;
        const int qh = kvh * group_stride + g_base + g;
        bf16* ob = out + ((size_t)seq * n_q + qh) * D + p * DS;
#pragma unroll
        for (int i = 0; i < DS; ++i) {
          union { unsigned short u; __hip_bfloat16 b; } cv;
          cv.u = f2bf(o[i] * inv);
          ob[i] = cv.b;
        }
      }
    }
    return;
  }
  // multi-partition: each wave is its own flash-decode slice — write
  // (m, l, acc) to the grid workspace; the reduce kernel combines.
  if (t_local == 0) {
#pragma unroll
    for (int g = 0; g < GROUP; ++g) {
      const int qh = kvh * group_stride + g_base + g;
      float* wa = ws_acc +
          (((size_t)seq * n_q + qh) * slices + slice) * D + p * DS;
#pragma unroll
      for (int i = 0; i < DS; ++i) wa[i] = acc[g][i];
      if (p == 0) {
        float* wm =
            ws_ml + (((size_t)seq * n_q + qh) * slices + slice) * 2;
        wm[0] = m[g];
        wm[1] = l[g];
      }
    }
  }
}

// combine partial (m, l, acc) across partitions: one wave per (seq, q-head)
__global__ __launch_bounds__(WAVE)
void paged_attn_reduce_kernel(bf16* __restrict__ out,
                              const float* __restrict__ ws_acc,
                              const float* __restrict__ ws_ml,
                              const int* __restrict__ seq_lens,
                              int n_q, int D, int n_parts) {
  // n_parts here = total slices in the workspace (grid parts, or
  // grid parts x PA_WAVES for the v2 kernel)
  const int seq = blockIdx.x;
  const int qh = blockIdx.y;
  const int lane = threadIdx.x;
  const float* ml = ws_ml + ((size_t)seq * n_q + qh) * n_parts * 2;
  const int seq_len = seq_lens[seq];
  const int n_blocks = (seq_len + PA_BLOCK - 1) / PA_BLOCK;
  const int blocks_per_part = (n_blocks + n_parts - 1) / n_parts;
  const int used_parts = min(n_parts, (n_blocks + blocks_per_part - 1) /
                                          max(blocks_per_part, 1));
  float m_star = -INFINITY;
  for (int pp = 0; pp < used_parts; ++pp) m_star = fmaxf(m_star, ml[pp * 2]);
  float l_tot = 0.f;
  const int per_lane = (D + WAVE - 1) / WAVE;  // elems per lane (2 for D=128)
  float o[4] = {0.f, 0.f, 0.f, 0.f};
  for (int pp = 0; pp < used_parts; ++pp) {
    const float sc = (ml[pp * 2] == -INFINITY) ? 0.f
                                               : __expf(ml[pp * 2] - m_star);
    l_tot += sc * ml[pp * 2 + 1];
    const float* wa = ws_acc + (((size_t)seq * n_q + qh) * n_parts + pp) * D;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      if (i < per_lane) {
        const int d = lane * per_lane + i;
        if (d < D) o[i] += sc * wa[d];
      }
    }
  }
  const float inv_l = (l_tot > 0.f) ? 1.f / l_tot : 0.f;
  bf16* ob = out + ((size_t)seq * n_q + qh) * D;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    if (i < per_lane) {
      const int d = lane * per_lane + i;
      if (d < D) ob[d] = __float2bfloat16(o[i] * inv_l);
    }
  }
}

// ---------------------------------------------------------------------------
// C API (called from Python via ctypes; raw device pointers + hipStream_t)
// ---------------------------------------------------------------------------
#define LL_CHECK(expr)                                                       \
  do {                                                                       \
    hipError_t _e = (expr);                                                  \
    if (_e != hipSuccess) return (int)_e;                                    \
  } while (0)

extern "C" {

int ll_rmsnorm(void* out, const void* in, const void* weight, void* residual,
               int n_tokens, int hidden, float eps, void* stream) {
  if (hidden % 8 != 0) return -2;
  dim3 grid(n_tokens), block(256);
  hipStream_t s = (hipStream_t)stream;
  if (residual) {
    hipLaunchKernelGGL(rmsnorm_kernel<true>, grid, block, 0, s, (bf16*)out,
                       (const bf16*)in, (const bf16*)weight, (bf16*)residual,
                       hidden, eps);
  } else {
    hipLaunchKernelGGL(rmsnorm_kernel<false>, grid, block, 0, s, (bf16*)out,
                       (const bf16*)in, (const bf16*)weight, nullptr, hidden,
                       eps);
  }
  LL_CHECK(hipGetLastError());
  return 0;
}

int ll_rope(void* q, void* k, const void* cos_sin, const void* positions,
            int n_tokens, int n_q, int n_kv, int D, int q_rstride,
            int k_rstride, void* stream) {
  if (D % 16 != 0) return -2;
  const int threads_per_head = (D / 2) / 4;
  const long long total = (long long)n_tokens * (n_q + n_kv) * threads_per_head;
  const int block = 256;
  const long long grid = (total + block - 1) / block;
  hipLaunchKernelGGL(rope_kernel, dim3((unsigned)grid), dim3(block), 0,
                     (hipStream_t)stream, (bf16*)q, (bf16*)k,
                     (const float*)cos_sin, (const int*)positions, n_tokens,
                     n_q, n_kv, D, q_rstride, k_rstride);
  LL_CHECK(hipGetLastError());
  return 0;
}

int ll_swiglu(void* out, const void* gateup, long long n_tokens, int inter,
              void* stream) {
  if (inter % 8 != 0) return -2;
  const long long total = n_tokens * (inter >> 3);
  const int block = 256;
  long long grid = (total + block - 1) / block;
  if (grid > 2048) grid = 2048;  // grid-stride (G11)
  hipLaunchKernelGGL(swiglu_kernel, dim3((unsigned)grid), dim3(block), 0,
                     (hipStream_t)stream, (bf16*)out, (const bf16*)gateup,
                     n_tokens, inter);
  LL_CHECK(hipGetLastError());
  return 0;
}


int ll_rope_kv_append(void* q, void* k, const void* v,
                      const void* cos_sin, const void* positions,
                      void* kcache, void* vcache, const void* slots,
                      int n_tokens, int n_q, int n_kv, int D,
                      int block_size, int q_rstride, int k_rstride,
                      int v_rstride, void* stream) {
  if (D % 16 != 0) return -2;
  hipStream_t s = (hipStream_t)stream;
  const long long rope_total =
      (long long)n_tokens * (n_q + n_kv) * (D / 8);
  const long long v_total = (long long)n_tokens * n_kv * (D / 8);
  const int block = 256;
  const long long grid = (rope_total + v_total + block - 1) / block;
  hipLaunchKernelGGL(rope_kv_append_kernel, dim3((unsigned)grid),
                     dim3(block), 0, s, (bf16*)q, (bf16*)k,
                     (const bf16*)v, (const float*)cos_sin,
                     (const int*)positions, (bf16*)kcache,
                     (bf16*)vcache, (const int*)slots, n_tokens, n_q,
                     n_kv, D, block_size, q_rstride, k_rstride,
                     v_rstride);
  LL_CHECK(hipGetLastError());
  return 0;
}

int ll_kv_append(void* kcache, void* vcache, const void* k, const void* v,
                 const void* slots, int n_tokens, int n_kv, int D,
                 int block_size, int k_rstride, int v_rstride,
                 void* stream) {
  if (D % 8 != 0) return -2;
  const long long total = (long long)n_tokens * n_kv * (D >> 3);
  const int block = 256;
  const long long grid = (total + block - 1) / block;
  hipLaunchKernelGGL(kv_append_kernel, dim3((unsigned)grid), dim3(block), 0,
                     (hipStream_t)stream, (bf16*)kcache, (bf16*)vcache,
                     (const bf16*)k, (const bf16*)v, (const int*)slots,
                     n_tokens, n_kv, D, block_size, k_rstride, v_rstride);
  LL_CHECK(hipGetLastError());
  return 0;
}

int ll_kv_append_fp8(void* kcache, void* vcache, void* k_scale,
                     void* v_scale, const void* k, const void* v,
                     const void* slots, int n_tokens, int n_kv, int D,
                     int block_size, int k_rstride, int v_rstride,
                     void* stream) {
  if (D != 128) return -2;
  const long long waves = (long long)n_tokens * n_kv;
  const int block = 256;
  const long long grid = (waves * WAVE + block - 1) / block;
  hipLaunchKernelGGL(kv_append_fp8_kernel, dim3((unsigned)grid),
                     dim3(block), 0, (hipStream_t)stream,
                     (unsigned char*)kcache, (unsigned char*)vcache,
                     (float*)k_scale, (float*)v_scale, (const bf16*)k,
                     (const bf16*)v, (const int*)slots, n_tokens, n_kv,
                     D, block_size, k_rstride, v_rstride);
  LL_CHECK(hipGetLastError());
  return 0;
}

int ll_paged_attn_decode(void* out, void* ws_acc, void* ws_ml, const void* q,
                         const void* kcache, const void* vcache,
                         const void* block_tables, const void* seq_lens,
                         int num_seqs, int max_blocks, int n_q, int n_kv,
                         int D, int block_size, int n_parts, float scale,
                         int want_v2, int q_rstride, void* stream) {
  if (block_size != PA_BLOCK || D > 128 || D % 32 != 0) return -2;
  if (n_q % n_kv != 0) return -2;
  hipStream_t s = (hipStream_t)stream;
  const int group = n_q / n_kv;
  // v2 (GQA-amortized, pages split across waves): caller must size the
  // workspace for n_parts*PA_WAVES slices and pass want_v2 explicitly
  const bool v2 = (want_v2 && D == 128 && group <= 16 &&
                   (n_parts == 1 || ws_acc != nullptr));
  if (v2) {
    dim3 grid(num_seqs, n_kv, n_parts), block(PA_WAVES * WAVE);
    const int slices = n_parts * PA_WAVES;
#define LAUNCH_PA_V2(G, GS, GB)                                             \
  hipLaunchKernelGGL((paged_attn_decode_v2_kernel<G>), grid, block, 0, s,   \
                     (bf16*)out, (float*)ws_acc, (float*)ws_ml,             \
                     (const bf16*)q, (const bf16*)kcache,                   \
                     (const bf16*)vcache, nullptr, nullptr,                 \
                     (const int*)block_tables, (const int*)seq_lens, n_q,   \
                     n_kv, max_blocks, n_parts, scale, q_rstride, GS, GB)
    // any GQA group: greedy 8/4/2/1 passes at the same group_stride.
    // GROUP=8 runs SINGLE-PASS (16-lane/token layout) so the 70B
    // shape reads each page once instead of twice (r2; the r1 +16%
    // A/B left that amortization on the table)
    // Default two-pass (max_g 4): the single-pass GROUP=8 kernel
    // measured 40.08 vs 39.63 ms/step on llama-3-70b bs32 ctx1024 —
    // the second pass's K/V re-reads are L2-hot while the 16-lane
    // layout pays ~3x the cross-lane shuffles (profiles/
    // r02_group8_single_pass_ab.md).  LLMLB_PA_MAX_G=8 enables it.
    static const char* mgenv = getenv("LLMLB_PA_MAX_G");
    const int max_g = mgenv ? atoi(mgenv) : 4;
    for (int gb = 0, rem = group; rem > 0;) {
      int g = rem >= 8 ? 8 : (rem >= 4 ? 4 : (rem >= 2 ? 2 : 1));
      if (g > max_g) g = max_g;
      switch (g) {
        case 8: LAUNCH_PA_V2(8, group, gb); break;
        case 4: LAUNCH_PA_V2(4, group, gb); break;
        case 2: LAUNCH_PA_V2(2, group, gb); break;
        default: LAUNCH_PA_V2(1, group, gb); break;
      }
      gb += g;
      rem -= g;
    }
#undef LAUNCH_PA_V2
    LL_CHECK(hipGetLastError());
    if (n_parts > 1) {   // single partition combined in-workgroup
      dim3 rgrid(num_seqs, n_q), rblock(WAVE);
      hipLaunchKernelGGL(paged_attn_reduce_kernel, rgrid, rblock, 0, s,
                         (bf16*)out, (const float*)ws_acc,
                         (const float*)ws_ml, (const int*)seq_lens, n_q,
                         D, slices);
      LL_CHECK(hipGetLastError());
    }
    return 0;
  }
v1_path:
  dim3 grid(num_seqs, n_kv, n_parts), block(PA_WAVES * WAVE);
  hipLaunchKernelGGL(paged_attn_decode_kernel, grid, block, 0, s, (bf16*)out,
                     (float*)ws_acc, (float*)ws_ml, (const bf16*)q,
                     (const bf16*)kcache, (const bf16*)vcache,
                     (const int*)block_tables, (const int*)seq_lens, n_q,
                     n_kv, D, max_blocks, n_parts, scale, q_rstride);
  LL_CHECK(hipGetLastError());
  if (n_parts > 1) {
    dim3 rgrid(num_seqs, n_q), rblock(WAVE);
    hipLaunchKernelGGL(paged_attn_reduce_kernel, rgrid, rblock, 0, s,
                       (bf16*)out, (const float*)ws_acc, (const float*)ws_ml,
                       (const int*)seq_lens, n_q, D, n_parts);
    LL_CHECK(hipGetLastError());
  }
  return 0;
}

// FP8-KV decode attention: v2 kernel only (D=128, group<=16); same
// workspace/reduce and single-partition LDS-combine semantics as bf16
int ll_paged_attn_decode_fp8(void* out, void* ws_acc, void* ws_ml,
                             const void* q, const void* kcache,
                             const void* vcache, const void* k_scale,
                             const void* v_scale,
                             const void* block_tables,
                             const void* seq_lens, int num_seqs,
                             int max_blocks, int n_q, int n_kv, int D,
                             int block_size, int n_parts, float scale,
                             int q_rstride, void* stream) {
  if (block_size != PA_BLOCK || D != 128) return -2;
  if (n_q % n_kv != 0) return -2;
  const int group = n_q / n_kv;
  if (group > 16) return -2;
  if (n_parts > 1 && ws_acc == nullptr) return -2;
  hipStream_t s = (hipStream_t)stream;
  dim3 grid(num_seqs, n_kv, n_parts), block(PA_WAVES * WAVE);
  const int slices = n_parts * PA_WAVES;
#define LAUNCH_V2_FP8(G, GS, GB)                                            \
  hipLaunchKernelGGL((paged_attn_decode_v2_kernel<G, true>), grid, block,   \
                     0, s, (bf16*)out, (float*)ws_acc, (float*)ws_ml,       \
                     (const bf16*)q, (const bf16*)kcache,                   \
                     (const bf16*)vcache, (const float*)k_scale,            \
                     (const float*)v_scale, (const int*)block_tables,       \
                     (const int*)seq_lens, n_q, n_kv, max_blocks,           \
                     n_parts, scale, q_rstride, GS, GB)
  for (int gb = 0, rem = group; rem > 0;) {
    const int g = rem >= 8 ? 8 : (rem >= 4 ? 4 : (rem >= 2 ? 2 : 1));
    switch (g) {
      case 8: LAUNCH_V2_FP8(8, group, gb); break;
      case 4: LAUNCH_V2_FP8(4, group, gb); break;
      case 2: LAUNCH_V2_FP8(2, group, gb); break;
      default: LAUNCH_V2_FP8(1, group, gb); break;
    }
    gb += g;
    rem -= g;
  }
#undef LAUNCH_V2_FP8
  LL_CHECK(hipGetLastError());
  if (n_parts > 1) {
    dim3 rgrid(num_seqs, n_q), rblock(WAVE);
    hipLaunchKernelGGL(paged_attn_reduce_kernel, rgrid, rblock, 0, s,
                       (bf16*)out, (const float*)ws_acc,
                       (const float*)ws_ml, (const int*)seq_lens, n_q, D,
                       slices);
    LL_CHECK(hipGetLastError());
  }
  return 0;
}

int ll_device_sync() {
  LL_CHECK(hipDeviceSynchronize());
  return 0;
}

}  // extern "C"

// ---------------------------------------------------------------------------
// Quantizing producers for the W8A8 decode arm: the activation feeding
// a decode GEMM is emitted as OCP e4m3 bytes + one f32 scale per row
// BY THE KERNEL THAT PRODUCES IT (a standalone quant launch would cost
// ~5 us x 3 per layer and eat the fp8 win, the same failure mode as
// the W8A16 convert pass).  Rounding via the native v_cvt_pk_fp8_f32.
// ---------------------------------------------------------------------------

__device__ __forceinline__ unsigned int pack4_fp8(const float* f,
                                                  float inv_scale) {
  unsigned int lo = __builtin_amdgcn_cvt_pk_fp8_f32(
      f[0] * inv_scale, f[1] * inv_scale, 0, false);
  return __builtin_amdgcn_cvt_pk_fp8_f32(
      f[2] * inv_scale, f[3] * inv_scale, lo, true);
}

// rmsnorm + quant: same row-per-block shape as rmsnorm_kernel; emits
// ONLY the quantized row (the bf16 normed output has no other consumer
// on the decode path).  residual semantics identical to rmsnorm<true>.
template <bool FUSE_RESIDUAL>
__global__ void rmsnorm_q_kernel(unsigned char* __restrict__ out_q,
                                 float* __restrict__ out_scale,
                                 const bf16* __restrict__ in,
                                 const bf16* __restrict__ weight,
                                 bf16* __restrict__ residual,
                                 int hidden, float eps) {
  const int row = blockIdx.x;
  const uint4* inv = reinterpret_cast<const uint4*>(in + (size_t)row * hidden);
  uint4* resv = FUSE_RESIDUAL
      ? reinterpret_cast<uint4*>(residual + (size_t)row * hidden) : nullptr;
  const uint4* wv = reinterpret_cast<const uint4*>(weight);
  const int nvec = hidden >> 3;

  float ssq = 0.f;
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    uint4 v = inv[i];
    float f[8];
    unpack8(v, f);
    if (FUSE_RESIDUAL) {
      float r[8];
      unpack8(resv[i], r);
#pragma unroll
      for (int j = 0; j < 8; ++j) f[j] += r[j];
      uint4 packed = pack8(f);
      resv[i] = packed;
      unpack8(packed, f);
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) ssq += f[j] * f[j];
  }
  for (int off = 32; off > 0; off >>= 1) ssq += __shfl_xor(ssq, off, WAVE);
  __shared__ float warp_red[4];
  const int wid = threadIdx.x / WAVE;
  if ((threadIdx.x & (WAVE - 1)) == 0) warp_red[wid] = ssq;
  __syncthreads();
  const int nw = blockDim.x / WAVE;
  float total = 0.f;
#pragma unroll
  for (int w = 0; w < 4; ++w) total += (w < nw) ? warp_red[w] : 0.f;
  const float rrms = rsqrtf(total / hidden + eps);

  // pass 2a: row amax of the normed values (recomputed pass 2b)
  float amax = 0.f;
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    uint4 v = FUSE_RESIDUAL ? resv[i] : inv[i];
    float f[8], w[8];
    unpack8(v, f);
    unpack8(wv[i], w);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      amax = fmaxf(amax, fabsf(f[j] * rrms * w[j]));
  }
  for (int off = 32; off > 0; off >>= 1)
    amax = fmaxf(amax, __shfl_xor(amax, off, WAVE));
  __syncthreads();   // warp_red reuse
  if ((threadIdx.x & (WAVE - 1)) == 0) warp_red[wid] = amax;
  __syncthreads();
  float amax_tot = 0.f;
#pragma unroll
  for (int w = 0; w < 4; ++w)
    amax_tot = fmaxf(amax_tot, (w < nw) ? warp_red[w] : 0.f);
  const float scale = amax_tot > 0.f ? amax_tot / 448.f : 1.f;
  const float inv_scale = 1.f / scale;
  if (threadIdx.x == 0) out_scale[row] = scale;

  // pass 2b: quantize + store bytes (8 per thread-iter = one dword x2)
  uint2* outv = reinterpret_cast<uint2*>(out_q + (size_t)row * hidden);
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    uint4 v = FUSE_RESIDUAL ? resv[i] : inv[i];
    float f[8], w[8];
    unpack8(v, f);
    unpack8(wv[i], w);
#pragma unroll
    for (int j = 0; j < 8; ++j) f[j] = f[j] * rrms * w[j];
    uint2 q;
    q.x = pack4_fp8(f, inv_scale);
    q.y = pack4_fp8(f + 4, inv_scale);
    outv[i] = q;
  }
}

// swiglu + quant: row per block over gateup [T, 2*inter]
__global__ void swiglu_q_kernel(unsigned char* __restrict__ out_q,
                                float* __restrict__ out_scale,
                                const bf16* __restrict__ gateup,
                                int inter) {
  const int row = blockIdx.x;
  const uint4* gv = reinterpret_cast<const uint4*>(
      gateup + (size_t)row * 2 * inter);
  const uint4* uv = reinterpret_cast<const uint4*>(
      gateup + (size_t)row * 2 * inter + inter);
  const int nvec = inter >> 3;

  float amax = 0.f;
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    float g[8], u[8];
    unpack8(gv[i], g);
    unpack8(uv[i], u);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float a = g[j] / (1.f + __expf(-g[j])) * u[j];
      amax = fmaxf(amax, fabsf(a));
    }
  }
  for (int off = 32; off > 0; off >>= 1)
    amax = fmaxf(amax, __shfl_xor(amax, off, WAVE));
  __shared__ float warp_red[4];
  const int wid = threadIdx.x / WAVE;
  if ((threadIdx.x & (WAVE - 1)) == 0) warp_red[wid] = amax;
  __syncthreads();
  const int nw = blockDim.x / WAVE;
  float amax_tot = 0.f;
#pragma unroll
  for (int w = 0; w < 4; ++w)
    amax_tot = fmaxf(amax_tot, (w < nw) ? warp_red[w] : 0.f);
  const float scale = amax_tot > 0.f ? amax_tot / 448.f : 1.f;
  const float inv_scale = 1.f / scale;
  if (threadIdx.x == 0) out_scale[row] = scale;

  uint2* outv = reinterpret_cast<uint2*>(out_q + (size_t)row * inter);
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    float g[8], u[8], f[8];
    unpack8(gv[i], g);
    unpack8(uv[i], u);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      f[j] = g[j] / (1.f + __expf(-g[j])) * u[j];
    uint2 q;
    q.x = pack4_fp8(f, inv_scale);
    q.y = pack4_fp8(f + 4, inv_scale);
    outv[i] = q;
  }
}

// plain row quantizer: amax + e4m3 pack, no norm — quantizes the
// attention output so o_proj can join the W8A8/W4A8 decode path
// (its producer is the attention kernel, which has no per-token
// amax; this is the one standalone quantize launch in the arm)
__global__ void rows_q_kernel(unsigned char* __restrict__ out_q,
                              float* __restrict__ out_scale,
                              const bf16* __restrict__ in, int hidden) {
  const int row = blockIdx.x;
  const uint4* inv = reinterpret_cast<const uint4*>(
      in + (size_t)row * hidden);
  const int nvec = hidden >> 3;
  float amax = 0.f;
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    float f[8];
    unpack8(inv[i], f);
#pragma unroll
    for (int j = 0; j < 8; ++j) amax = fmaxf(amax, fabsf(f[j]));
  }
  for (int off = 32; off > 0; off >>= 1)
    amax = fmaxf(amax, __shfl_xor(amax, off, WAVE));
  __shared__ float warp_red[4];
  const int wid = threadIdx.x / WAVE;
  if ((threadIdx.x & (WAVE - 1)) == 0) warp_red[wid] = amax;
  __syncthreads();
  const int nw = blockDim.x / WAVE;
  float amax_tot = 0.f;
#pragma unroll
  for (int w = 0; w < 4; ++w)
    amax_tot = fmaxf(amax_tot, (w < nw) ? warp_red[w] : 0.f);
  const float scale = amax_tot > 0.f ? amax_tot / 448.f : 1.f;
  const float inv_scale = 1.f / scale;
  if (threadIdx.x == 0) out_scale[row] = scale;
  uint2* outv = reinterpret_cast<uint2*>(out_q + (size_t)row * hidden);
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    float f[8];
    unpack8(inv[i], f);
    uint2 q;
    q.x = pack4_fp8(f, inv_scale);
    q.y = pack4_fp8(f + 4, inv_scale);
    outv[i] = q;
  }
}

extern "C" {

int ll_rows_q(void* out_q, void* out_scale, const void* in,
              int n_tokens, int hidden, void* stream) {
  if (hidden % 8 != 0) return -2;
  hipStream_t s = (hipStream_t)stream;
  dim3 grid(n_tokens), block(256);
  hipLaunchKernelGGL(rows_q_kernel, grid, block, 0, s,
                     (unsigned char*)out_q, (float*)out_scale,
                     (const bf16*)in, hidden);
  LL_CHECK(hipGetLastError());
  return 0;
}

int ll_rmsnorm_q(void* out_q, void* out_scale, const void* in,
                 const void* weight, void* residual, int n_tokens,
                 int hidden, float eps, void* stream) {
  if (hidden % 8 != 0) return -2;
  hipStream_t s = (hipStream_t)stream;
  dim3 grid(n_tokens), block(256);
  if (residual)
    hipLaunchKernelGGL((rmsnorm_q_kernel<true>), grid, block, 0, s,
                       (unsigned char*)out_q, (float*)out_scale,
                       (const bf16*)in, (const bf16*)weight,
                       (bf16*)residual, hidden, eps);
  else
    hipLaunchKernelGGL((rmsnorm_q_kernel<false>), grid, block, 0, s,
                       (unsigned char*)out_q, (float*)out_scale,
                       (const bf16*)in, (const bf16*)weight, nullptr,
                       hidden, eps);
  LL_CHECK(hipGetLastError());
  return 0;
}

int ll_swiglu_q(void* out_q, void* out_scale, const void* gateup,
                int n_tokens, int inter, void* stream) {
  if (inter % 8 != 0) return -2;
  hipStream_t s = (hipStream_t)stream;
  dim3 grid(n_tokens), block(256);
  hipLaunchKernelGGL(swiglu_q_kernel, grid, block, 0, s,
                     (unsigned char*)out_q, (float*)out_scale,
                     (const bf16*)gateup, inter);
  LL_CHECK(hipGetLastError());
  return 0;
}

}  // extern "C"
