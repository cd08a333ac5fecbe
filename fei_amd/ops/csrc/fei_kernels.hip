// fei_amd gfx950 kernels: RMSNorm, RoPE+KV-append, decode attention
// (split-K online-softmax), SwiGLU, sampling (argmax / Gumbel), position
// advance. Semantics match fei_amd/ops/reference.py; numerics tests compare
// against the fp32 torch reference (tests/test_ops_gpu.py).
//
// Design notes (per /opt/skills/guides/cdna_hip_programming.md):
//  - every memory-bound kernel loads bf16 as 16-byte vectors (Guideline 13)
//  - decode attention reads the length from a DEVICE tensor so the whole
//    decode step is hipGraph-capturable with a static grid
//  - block sizes are multiples of 64 (wave64)
#include "fei_common.h"
#include <cstdlib>  // getenv: FEI_ATTN_KV_NT (decode KV cache policy)

extern "C" {

// ---------------------------------------------------------------------------
// RMSNorm: out[r,:] = x[r,:] * rsqrt(mean(x^2)+eps) * w ; bf16 I/O, f32 acc.
// One block per row (grid-stride); cols % 8 == 0; cols/8 <= 256*MAX_V8.
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(256)
k_rmsnorm(u16* __restrict__ out, const u16* __restrict__ x,
          const u16* __restrict__ w, int rows, int cols, float eps) {
  __shared__ float red[4];
  const int nv = cols >> 3;                       // vec8 per row
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const s16x8* xr = (const s16x8*)(x + (long)row * cols);
    float ss = 0.f;
    for (int i = threadIdx.x; i < nv; i += blockDim.x) {
      s16x8 v = xr[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf2f((u16)v[j]);
        ss = fmaf(f, f, ss);
      }
    }
    ss = block_reduce_sum(ss, red);
    const float inv = rsqrtf(ss / (float)cols + eps);
    s16x8* orow = (s16x8*)(out + (long)row * cols);
    const s16x8* wv = (const s16x8*)w;
    // second read hits L1/L2 (a row is tiny vs 32 MB L2) — avoids a
    // runtime-indexed register array (guide rule 20: scratch).
    for (int i = threadIdx.x; i < nv; i += blockDim.x) {
      s16x8 v = xr[i];
      s16x8 wvv = wv[i];
      s16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o[j] = (short)f2bf(bf2f((u16)v[j]) * inv * bf2f((u16)wvv[j]));
      orow[i] = o;
    }
    __syncthreads();
  }
}

void fei_rmsnorm(void* out, const void* x, const void* w, int rows, int cols,
                 float eps, hipStream_t stream) {
  int grid = rows < 2048 ? rows : 2048;
  hipLaunchKernelGGL(k_rmsnorm, dim3(grid), dim3(256), 0, stream,
                     (u16*)out, (const u16*)x, (const u16*)w, rows, cols, eps);
}

// ---------------------------------------------------------------------------
// Fused residual-add + RMSNorm: res = res + x ; out = rmsnorm(res) * w.
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(256)
k_add_rmsnorm(u16* __restrict__ out, u16* __restrict__ res,
              const u16* __restrict__ x, const u16* __restrict__ w,
              int rows, int cols, float eps) {
  __shared__ float red[4];
  const int nv = cols >> 3;
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const s16x8* xr = (const s16x8*)(x + (long)row * cols);
    s16x8* rr = (s16x8*)(res + (long)row * cols);
    float ss = 0.f;
    for (int i = threadIdx.x; i < nv; i += blockDim.x) {
      s16x8 xv = xr[i];
      s16x8 rv = rr[i];
      s16x8 nr;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf2f((u16)rv[j]) + bf2f((u16)xv[j]);
        u16 b = f2bf(f);
        nr[j] = (short)b;
        f = bf2f(b);                                // match reference rounding
        ss = fmaf(f, f, ss);
      }
      rr[i] = nr;
    }
    ss = block_reduce_sum(ss, red);
    const float inv = rsqrtf(ss / (float)cols + eps);
    s16x8* orow = (s16x8*)(out + (long)row * cols);
    const s16x8* wv = (const s16x8*)w;
    // re-read the just-written residual (L1/L2-hot; rule 20: no runtime-
    // indexed register array).
    for (int i = threadIdx.x; i < nv; i += blockDim.x) {
      s16x8 rv = rr[i];
      s16x8 wvv = wv[i];
      s16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o[j] = (short)f2bf(bf2f((u16)rv[j]) * inv * bf2f((u16)wvv[j]));
      orow[i] = o;
    }
    __syncthreads();
  }
}

void fei_add_rmsnorm(void* out, void* res, const void* x, const void* w,
                     int rows, int cols, float eps, hipStream_t stream) {
  int grid = rows < 2048 ? rows : 2048;
  hipLaunchKernelGGL(k_add_rmsnorm, dim3(grid), dim3(256), 0, stream,
                     (u16*)out, (u16*)res, (const u16*)x, (const u16*)w,
                     rows, cols, eps);
}

// ---------------------------------------------------------------------------
// RoPE (decode) + KV append. q [B,Hq,D] roped in place at pos[b]; k,v
// [B,Hkv,D]: k roped and written with v into caches [B,Hkv,max_seq,D].
// Half-split pairing (Llama): pair (d, d+D/2). cos_sin: [max_seq, D/2, 2] f32.
// Grid: (Hq+Hkv, B); block D/2 threads (<=64 -> one wave for D=128).
// ---------------------------------------------------------------------------
__global__ void k_rope_kv_decode(u16* __restrict__ q, const u16* __restrict__ k,
                                 const u16* __restrict__ v,
                                 u16* __restrict__ kc, u16* __restrict__ vc,
                                 const float* __restrict__ cos_sin,
                                 const int* __restrict__ pos,
                                 int B, int Hq, int Hkv, int D, int max_seq,
                                 long q_bs, long kv_bs) {
  const int h = blockIdx.x;
  const int b = blockIdx.y;
  const int i = threadIdx.x;               // pair index in [0, D/2)
  const int p = pos[b];
  const float c = cos_sin[((long)p * (D / 2) + i) * 2 + 0];
  const float s = cos_sin[((long)p * (D / 2) + i) * 2 + 1];
  if (h < Hq) {
    u16* qp = q + (long)b * q_bs + (long)h * D;
    float x1 = bf2f(qp[i]), x2 = bf2f(qp[i + D / 2]);
    qp[i] = f2bf(x1 * c - x2 * s);
    qp[i + D / 2] = f2bf(x2 * c + x1 * s);
  } else {
    const int hk = h - Hq;
    const u16* kp = k + (long)b * kv_bs + (long)hk * D;
    const u16* vp = v + (long)b * kv_bs + (long)hk * D;
    u16* kcp = kc + (((long)b * Hkv + hk) * max_seq + p) * D;
    u16* vcp = vc + (((long)b * Hkv + hk) * max_seq + p) * D;
    float x1 = bf2f(kp[i]), x2 = bf2f(kp[i + D / 2]);
    kcp[i] = f2bf(x1 * c - x2 * s);
    kcp[i + D / 2] = f2bf(x2 * c + x1 * s);
    vcp[i] = vp[i];
    vcp[i + D / 2] = vp[i + D / 2];
  }
}

void fei_rope_kv_decode(void* q, const void* k, const void* v, void* k_cache,
                        void* v_cache, const float* cos_sin, const int* pos,
                        int B, int Hq, int Hkv, int D, int max_seq,
                        long q_bs, long kv_bs, hipStream_t stream) {
  hipLaunchKernelGGL(k_rope_kv_decode, dim3(Hq + Hkv, B), dim3(D / 2), 0,
                     stream, (u16*)q, (const u16*)k, (const u16*)v,
                     (u16*)k_cache, (u16*)v_cache, cos_sin, pos,
                     B, Hq, Hkv, D, max_seq, q_bs, kv_bs);
}

// ---------------------------------------------------------------------------
// RoPE (prefill) + KV append for S tokens. q [B,S,Hq,D] roped in place;
// k [B,S,Hkv,D] roped + appended with v at pos0[b]+s.
// Grid: (S, Hq+Hkv, B); block D/2.
// ---------------------------------------------------------------------------
__global__ void k_rope_kv_prefill(u16* __restrict__ q, const u16* __restrict__ k,
                                  const u16* __restrict__ v,
                                  u16* __restrict__ kc, u16* __restrict__ vc,
                                  const float* __restrict__ cos_sin,
                                  const int* __restrict__ pos0,
                                  int B, int S, int Hq, int Hkv, int D,
                                  int max_seq, long q_ts, long kv_ts) {
  const int s_idx = blockIdx.x;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int i = threadIdx.x;
  const int p = pos0[b] + s_idx;
  const long tok = (long)b * S + s_idx;
  const float c = cos_sin[((long)p * (D / 2) + i) * 2 + 0];
  const float sn = cos_sin[((long)p * (D / 2) + i) * 2 + 1];
  if (h < Hq) {
    u16* qp = q + tok * q_ts + (long)h * D;
    float x1 = bf2f(qp[i]), x2 = bf2f(qp[i + D / 2]);
    qp[i] = f2bf(x1 * c - x2 * sn);
    qp[i + D / 2] = f2bf(x2 * c + x1 * sn);
  } else {
    const int hk = h - Hq;
    const u16* kp = k + tok * kv_ts + (long)hk * D;
    const u16* vp = v + tok * kv_ts + (long)hk * D;
    u16* kcp = kc + (((long)b * Hkv + hk) * max_seq + p) * D;
    u16* vcp = vc + (((long)b * Hkv + hk) * max_seq + p) * D;
    float x1 = bf2f(kp[i]), x2 = bf2f(kp[i + D / 2]);
    kcp[i] = f2bf(x1 * c - x2 * sn);
    kcp[i + D / 2] = f2bf(x2 * c + x1 * sn);
    vcp[i] = vp[i];
    vcp[i + D / 2] = vp[i + D / 2];
  }
}

void fei_rope_kv_prefill(void* q, const void* k, const void* v, void* k_cache,
                         void* v_cache, const float* cos_sin, const int* pos0,
                         int B, int S, int Hq, int Hkv, int D, int max_seq,
                         long q_ts, long kv_ts, hipStream_t stream) {
  hipLaunchKernelGGL(k_rope_kv_prefill, dim3(S, Hq + Hkv, B), dim3(D / 2), 0,
                     stream, (u16*)q, (const u16*)k, (const u16*)v,
                     (u16*)k_cache, (u16*)v_cache, cos_sin, pos0,
                     B, S, Hq, Hkv, D, max_seq, q_ts, kv_ts);
}

// ---------------------------------------------------------------------------
// Decode attention, split-K partials, GQA-grouped. q [B,Hq,D] bf16 (roped);
// caches [B,Hkv,max_seq,D]. n = pos[b]+1 keys (length read on DEVICE so the
// launch shape is static for hipGraph capture).
// Grid (splits, Hkv, B), block 256 = 4 waves. One block serves ALL G=Hq/Hkv
// q-heads of its kv-head, so each K/V row is read ONCE per block (the
// one-block-per-q-head version re-read K/V G times — profiles/r01).
//   phase A: thread-per-key; dot against the G q-vectors staged in LDS,
//            G-wide block online-softmax state
//   phase B: threads as (D/2 pairs x key-groups) accumulate P*V for all G
// Partials: part_o [B,Hq,splits,D] f32; part_ml [B,Hq,splits,2] f32.
// ---------------------------------------------------------------------------
#define DEC_TILE 256
#define DEC_DMAX 128
#define DEC_GMAX 8

}  // extern "C" (templates need C++ linkage)

namespace {

// Reduce G values at once (one barrier pair for all heads).
template <int OP, int G>  // OP: 0 = max, 1 = sum
__device__ __forceinline__ void block_reduce_vec(float* v,
                                                 float red[DEC_GMAX][4]) {
  const int nw = blockDim.x >> 6;
  const int wid = threadIdx.x >> 6;
  #pragma unroll
    for (int g = 0; g < G; ++g) {
    float x = v[g];
#pragma unroll
    for (int off = 32; off; off >>= 1)
      x = OP == 0 ? fmaxf(x, __shfl_xor(x, off)) : x + __shfl_xor(x, off);
    if ((threadIdx.x & 63) == 0) red[g][wid] = x;
    v[g] = x;
  }
  __syncthreads();
  #pragma unroll
    for (int g = 0; g < G; ++g) {
    float x = red[g][0];
    for (int i = 1; i < nw; ++i)
      x = OP == 0 ? fmaxf(x, red[g][i]) : x + red[g][i];
    v[g] = x;
  }
  __syncthreads();
}

// ROPE=true additionally fuses the per-step RoPE + KV-append: q is the RAW
// qkv-buffer view (roped into LDS during staging), the new token's k/v are
// roped in-block from kin/vin (every split computes them — the new key may
// fall in any split's chunk) and the split that OWNS key n-1 writes the
// caches. Readers never read cache[n-1]; they use the LDS copy, so there is
// no cross-workgroup ordering (placement-independent by construction).
// sc1 write-through hand-off helpers for the fused split-K combine (guide
// §6 G16 cheap variant): relaxed agent-scope stores/loads lower to sc1 at
// 4-8 B widths — visibility comes from the write-through to the coherence
// point, so NO release/acquire fences are needed (the fence-based recipe's
// per-WG buffer_wbl2 + reducer-side L2 invalidate measured 9% SLOWER
// end-to-end than the separate combine launch it replaced).
__device__ inline void store2_sc1(float* p, float a, float b) {
  float2 v = make_float2(a, b);
  unsigned long long bits;
  __builtin_memcpy(&bits, &v, 8);
  __hip_atomic_store((unsigned long long*)p, bits, __ATOMIC_RELAXED,
                     __HIP_MEMORY_SCOPE_AGENT);
}
__device__ inline float load_sc1(const float* p) {
  return __hip_atomic_load(p, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
}
__device__ inline void store_i_sc1(int* p, int v) {
  __hip_atomic_store(p, v, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
}
__device__ inline int load_i_sc1(const int* p) {
  return __hip_atomic_load(p, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
}

// When `flags` is non-null the split-K combine is FUSED: every split
// workgroup of a (b, hkv) pair publishes its partials with sc1
// write-through stores, drains them (vmcnt), then lane 0 writes a per-split
// DONE flag (sc1, DISTINCT address per split — a shared arrival counter's
// same-address fetch_adds serialize at the coherence point, ~300 ns each x
// 32 splits ≈ 10 us/layer, which made the counter form 8-9% slower
// end-to-end). Split 0 (never empty: its chunk starts at key 0) is the
// fixed reducer: it polls the 32 flags in parallel (one lane each), then
// reduces every split's partials for its G heads into `outf` (bf16) with
// sc1 loads. Flags carry a (step, layer) tag — no reset, no poisoning on
// replay: a flag matches only when THIS step's layer instance wrote it.
// This removes the separate k_attn_decode_combine launch (32/step, ~4.9 us
// + ~1.2 us boundary each — profiles/r02_plainchain_kernel_shares).
template <int G, bool ROPE, bool KVNT = false>
__global__ void __launch_bounds__(256)
k_attn_decode(const u16* __restrict__ q, u16* __restrict__ kc,
              u16* __restrict__ vc, float* __restrict__ part_o,
              float* __restrict__ part_ml, const int* __restrict__ pos,
              int B, int Hq, int Hkv, int D, int max_seq, int splits,
              float scale, long q_bs,
              const u16* __restrict__ kin, const u16* __restrict__ vin,
              const float* __restrict__ cos_sin, long kv_bs,
              u16* __restrict__ outf, int* __restrict__ arrive, int layer) {
  const int split = blockIdx.x;
  const int hkv = blockIdx.y;
  const int b = blockIdx.z;
  const int tid = threadIdx.x;

  __shared__ float qs[DEC_GMAX][DEC_DMAX];
  __shared__ float pl[DEC_GMAX][DEC_TILE];
  __shared__ float red[DEC_GMAX][4];
  // key-group partial-o staging: [kg * dvecs + dv][element] — kgroups *
  // dvecs == blockDim == DEC_TILE for every D
  __shared__ float osh[DEC_TILE][8];

  const int n = pos[b] + 1;
  const int chunk = (n + splits - 1) / splits;
  const int start = split * chunk;
  const int end = min(start + chunk, n);

  if (start >= end) {
    #pragma unroll
    for (int g = 0; g < G; ++g) {
      const int hq = hkv * G + g;
      float* po = part_o + (((long)b * Hq + hq) * splits + split) * D;
      float* pml = part_ml + (((long)b * Hq + hq) * splits + split) * 2;
      if (arrive != nullptr) {
        for (int d = tid * 2; d < D; d += blockDim.x * 2)
          store2_sc1(&po[d], 0.f, 0.f);
        if (tid == 0) store2_sc1(pml, -1.0f / 0.0f, 0.f);
      } else {
        for (int d = tid; d < D; d += blockDim.x) po[d] = 0.f;
        if (tid == 0) { pml[0] = -1.0f / 0.0f; pml[1] = 0.f; }
      }
    }
    if (arrive == nullptr) return;
  } else {

  const int half = D / 2;
  const int p_new = n - 1;
  if (ROPE) {
    // stage roped+scaled q
    for (int i = tid; i < G * half; i += blockDim.x) {
      const int g = i / half, d = i % half;
      const float c = cos_sin[((long)p_new * half + d) * 2 + 0];
      const float sn = cos_sin[((long)p_new * half + d) * 2 + 1];
      const u16* qp = q + (long)b * q_bs + (long)(hkv * G + g) * D;
      const float x1 = bf2f(qp[d]);
      const float x2 = bf2f(qp[d + half]);
      qs[g][d] = (x1 * c - x2 * sn) * scale;
      qs[g][d + half] = (x2 * c + x1 * sn) * scale;
    }
    // the split whose chunk contains key p_new ropes + appends the new
    // token's k/v BEFORE this block's barrier, then reads it back from the
    // cache like any other key (same-workgroup store->load visibility).
    // No other split's [start,end) range ever touches key p_new.
    if (p_new >= start && p_new < end) {
      u16* kcp = kc + (((long)b * Hkv + hkv) * max_seq + p_new) * D;
      u16* vcp = vc + (((long)b * Hkv + hkv) * max_seq + p_new) * D;
      for (int d = tid; d < half; d += blockDim.x) {
        const float c = cos_sin[((long)p_new * half + d) * 2 + 0];
        const float sn = cos_sin[((long)p_new * half + d) * 2 + 1];
        const u16* kp = kin + (long)b * kv_bs + (long)hkv * D;
        const u16* vp = vin + (long)b * kv_bs + (long)hkv * D;
        const float x1 = bf2f(kp[d]);
        const float x2 = bf2f(kp[d + half]);
        kcp[d] = f2bf(x1 * c - x2 * sn);
        kcp[d + half] = f2bf(x2 * c + x1 * sn);
        vcp[d] = vp[d];
        vcp[d + half] = vp[d + half];
      }
    }
  } else {
    // q pre-roped; stage scaled
    for (int i = tid; i < G * D; i += blockDim.x) {
      const int g = i / D, d = i % D;
      qs[g][d] = bf2f(q[(long)b * q_bs + (long)(hkv * G + g) * D + d]) * scale;
    }
  }
  __syncthreads();

  const u16* kbase = kc + ((long)b * Hkv + hkv) * max_seq * D;
  const u16* vbase = vc + ((long)b * Hkv + hkv) * max_seq * D;

  // V pass lane maps. WIDE (chunk >= 64): 16 B of one V row per lane —
  // fewer, full-width loads win once every key-group has work (+1% at
  // 2k-8k context). PAIR (short chunks): 4 B per lane over D/2 lanes —
  // 4x more active lanes when the chunk covers few keys (the wide map
  // measured -3% on the 512-ctx headline: only 16 of 256 lanes active).
  // Ablation: experimental/decode_ablate.hip, profiles/r02.
  const int dvecs = D / 8;                        // 16 for D=128
  const int kgroups = blockDim.x / dvecs;         // 16
  const int keys_per_group = DEC_TILE / kgroups;  // 16
  const int dv = tid % dvecs;
  const int kg = tid / dvecs;
  const bool wide = chunk >= 64;
  const int dpairs = D / 2;                       // pair-mode roles
  const int kgroupsP = blockDim.x / dpairs;
  const int keysP = DEC_TILE / kgroupsP;
  const int dp = tid % dpairs;
  const int kgP = tid / dpairs;

  float m[DEC_GMAX], l[DEC_GMAX], sc[DEC_GMAX];
  float oa[DEC_GMAX][8];
  #pragma unroll
    for (int g = 0; g < G; ++g) {
    m[g] = -1.0f / 0.0f; l[g] = 0.f;
#pragma unroll
    for (int e = 0; e < 8; ++e) oa[g][e] = 0.f;
  }

  for (int tile = start; tile < end; tile += DEC_TILE) {
    const int kk = tile + tid;
    // scores for this key against every q-head of the group
    if (kk < end) {
      const s16x8* krow = (const s16x8*)(kbase + (long)kk * D);
      #pragma unroll
    for (int g = 0; g < G; ++g) sc[g] = 0.f;
      // (a #pragma unroll 4 here measured exactly neutral at every
      // context length — the backend already pipelines the key-row loads
      // despite the runtime D trip count)
      for (int i = 0; i < D / 8; ++i) {
        s16x8 kv8 = KVNT ? __builtin_nontemporal_load(&krow[i]) : krow[i];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const float kf = bf2f((u16)kv8[j]);
          #pragma unroll
    for (int g = 0; g < G; ++g)
            sc[g] = fmaf(qs[g][i * 8 + j], kf, sc[g]);
        }
      }
    } else {
      #pragma unroll
    for (int g = 0; g < G; ++g) sc[g] = -1.0f / 0.0f;
    }
    float tile_m[DEC_GMAX];
    #pragma unroll
    for (int g = 0; g < G; ++g) tile_m[g] = sc[g];
    block_reduce_vec<0, G>(tile_m, red);
    float alpha[DEC_GMAX], pv[DEC_GMAX];
    #pragma unroll
    for (int g = 0; g < G; ++g) {
      const float m_new = fmaxf(m[g], tile_m[g]);
      alpha[g] = __expf(m[g] - m_new);             // 0 on first tile
      pv[g] = (kk < end) ? __expf(sc[g] - m_new) : 0.f;
      pl[g][tid] = pv[g];
      m[g] = m_new;
    }
    float tile_sum[DEC_GMAX];
    #pragma unroll
    for (int g = 0; g < G; ++g) tile_sum[g] = pv[g];
    block_reduce_vec<1, G>(tile_sum, red);         // also makes pl visible
    #pragma unroll
    for (int g = 0; g < G; ++g) l[g] = l[g] * alpha[g] + tile_sum[g];

    // P*V: V row read once (16 B/lane), used by every head
    #pragma unroll
    for (int g = 0; g < G; ++g) {
#pragma unroll
      for (int e = 0; e < 8; ++e) oa[g][e] *= alpha[g];
    }
    // Break-free trip count so hipcc can unroll and keep several V loads
    // in flight (a data-dependent break serialises the loop into dependent
    // L2 round trips — measured 41 us/kernel at seq 640).
    const int kmax = min(DEC_TILE, end - tile);
    if (wide) {
      const int kbase_local = kg * keys_per_group;
      const int iters = min(keys_per_group, max(0, kmax - kbase_local));
#pragma unroll 2
      for (int j = 0; j < iters; ++j) {
        const int kl = kbase_local + j;
        const s16x8* vrow =
            (const s16x8*)(vbase + (long)(tile + kl) * D) + dv;
        const s16x8 v8 = KVNT ? __builtin_nontemporal_load(vrow) : *vrow;
        float vf[8];
        #pragma unroll
        for (int e = 0; e < 8; ++e) vf[e] = bf2f((u16)v8[e]);
        #pragma unroll
    for (int g = 0; g < G; ++g) {
          const float p = pl[g][kl];
          #pragma unroll
          for (int e = 0; e < 8; ++e) oa[g][e] = fmaf(p, vf[e], oa[g][e]);
        }
      }
    } else {
      const int kbase_local = kgP * keysP;
      const int iters = min(keysP, max(0, kmax - kbase_local));
#pragma unroll 4
      for (int j = 0; j < iters; ++j) {
        const int kl = kbase_local + j;
        const u16* vrow = vbase + (long)(tile + kl) * D + dp * 2;
        const u32 vpair = KVNT
            ? __builtin_nontemporal_load((const u32*)vrow)
            : *(const u32*)vrow;
        const float v0 = bf2f((u16)(vpair & 0xffff));
        const float v1 = bf2f((u16)(vpair >> 16));
        #pragma unroll
    for (int g = 0; g < G; ++g) {
          const float p = pl[g][kl];
          oa[g][0] = fmaf(p, v0, oa[g][0]);
          oa[g][1] = fmaf(p, v1, oa[g][1]);
        }
      }
    }
    __syncthreads();                               // pl reuse next tile
  }

  // combine the key-groups' partial o through LDS, one head at a time
  // epilogue: each key-group writes its own LDS slot; ONE barrier, then
  // key-group 0 sums (a sequential per-group accumulate cost 16 barriers).
  #pragma unroll
    for (int g = 0; g < G; ++g) {
    const int hq = hkv * G + g;
    float* po = part_o + (((long)b * Hq + hq) * splits + split) * D;
    float* pml = part_ml + (((long)b * Hq + hq) * splits + split) * 2;
    if (wide) {
#pragma unroll
      for (int e = 0; e < 8; ++e) osh[kg * dvecs + dv][e] = oa[g][e];
      __syncthreads();
      if (kg == 0) {
        float s[8];
#pragma unroll
        for (int e = 0; e < 8; ++e) s[e] = osh[dv][e];
        for (int gg = 1; gg < kgroups; ++gg) {
#pragma unroll
          for (int e = 0; e < 8; ++e) s[e] += osh[gg * dvecs + dv][e];
        }
        if (arrive != nullptr) {
#pragma unroll
          for (int e = 0; e < 8; e += 2)
            store2_sc1(&po[dv * 8 + e], s[e], s[e + 1]);
        } else {
#pragma unroll
          for (int e = 0; e < 8; ++e) po[dv * 8 + e] = s[e];
        }
      }
    } else {
      osh[kgP * dpairs + dp][0] = oa[g][0];
      osh[kgP * dpairs + dp][1] = oa[g][1];
      __syncthreads();
      if (kgP == 0) {
        float s0 = osh[dp][0], s1 = osh[dp][1];
        for (int gg = 1; gg < kgroupsP; ++gg) {
          s0 += osh[gg * dpairs + dp][0];
          s1 += osh[gg * dpairs + dp][1];
        }
        if (arrive != nullptr) store2_sc1(&po[dp * 2], s0, s1);
        else { po[dp * 2] = s0; po[dp * 2 + 1] = s1; }
      }
    }
    if (tid == 0) {
      if (arrive != nullptr) store2_sc1(pml, m[g], l[g]);
      else { pml[0] = m[g]; pml[1] = l[g]; }
    }
    __syncthreads();
  }
  }  // start < end

  if (arrive == nullptr) return;
  // --- fused combine hand-off: partials are at the coherence point (sc1
  // write-through stores above); drain them (vmcnt), barrier, then lane 0
  // publishes this split's DONE flag. Tag (step, layer)-unique => no reset.
  {
    const int tag = 0x40000000 | (n * 64 + (layer & 63));
    int* flags = arrive + ((long)b * Hkv + hkv) * splits;
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    if (tid == 0) store_i_sc1(&flags[split], tag);
    if (split >= G) return;
    // splits 0..G-1 are reducers, ONE HEAD EACH (same combine parallelism
    // as the separate kernel — a single split-0 reducer walking G heads
    // read 4x the bytes per WG and lost ~3.6 us/layer). Each polls the
    // flags one lane per split (distinct addresses — parallel sc1 loads),
    // bounded spin (a lost producer makes this give up and produce stale
    // output rather than wedge the GPU).
    int ready = 0;
    for (int spin = 0; spin < (1 << 18) && !ready; ++spin) {
      const int mine =
          (tid < splits) ? (load_i_sc1(&flags[tid]) == tag) : 1;
      ready = __syncthreads_and(mine);
    }
    // One agent-scope ACQUIRE, then PLAIN cached loads: relaxed-atomic
    // (sc1) reducer loads stay in program order and bypass the caches, so
    // per-thread re-reads of the same 2*splits m/l values would serialize
    // ~192 uncached round trips per thread (the first reducer forms lost
    // ~10 us/layer exactly here). The fence orders the loads after the
    // observed flags and invalidates the local caches, so plain loads see
    // the written-through partials (placement-independent, guide G16).
    if (tid == 0) __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
    __syncthreads();
    float* pmlsh = &pl[0][0];                     // dead after main loop
    for (int g = split; g < G; g += splits) {     // usually exactly one
      const int hq = hkv * G + g;
      const float* pml0 = part_ml + ((long)b * Hq + hq) * splits * 2;
      for (int i = tid; i < splits * 2; i += blockDim.x)
        pmlsh[i] = pml0[i];
      __syncthreads();
      const float* po = part_o + ((long)b * Hq + hq) * splits * D;
      for (int d = tid; d < D; d += blockDim.x) {
        float mm = -1.0f / 0.0f;
        for (int s = 0; s < splits; ++s) mm = fmaxf(mm, pmlsh[s * 2]);
        float ll = 0.f, oo = 0.f;
#pragma unroll 8
        for (int s = 0; s < splits; ++s) {
          const float w = __expf(pmlsh[s * 2] - mm);
          ll = fmaf(w, pmlsh[s * 2 + 1], ll);
          oo = fmaf(w, po[s * D + d], oo);
        }
        outf[((long)b * Hq + hq) * D + d] = f2bf(ll > 0.f ? oo / ll : 0.f);
      }
      __syncthreads();                            // pmlsh reuse next g
    }
  }
}

}  // namespace

extern "C" {

void fei_attn_decode(const void* q, const void* k_cache, const void* v_cache,
                     float* part_o, float* part_ml, const int* pos,
                     int B, int Hq, int Hkv, int D, int max_seq, int splits,
                     float scale, long q_bs,
                     const void* kin, const void* vin, const float* cos_sin,
                     long kv_bs, void* out_fused, int* arrive, int layer,
                     hipStream_t stream) {
  const int G = Hq / Hkv;
  dim3 grid(splits, Hkv, B);
  const int rope = cos_sin != nullptr;
  static int kvnt = -1;
  if (kvnt < 0) {
    const char* e = getenv("FEI_ATTN_KV_NT");
    kvnt = (e && e[0] == '1') ? 1 : 0;
  }
#define LAUNCH_DEC(GV, RV, NTV) \
  hipLaunchKernelGGL((k_attn_decode<GV, RV, NTV>), grid, dim3(256), 0, \
                     stream, (const u16*)q, (u16*)k_cache, (u16*)v_cache, \
                     part_o, part_ml, pos, B, Hq, Hkv, D, max_seq, splits, \
                     scale, q_bs, (const u16*)kin, (const u16*)vin, cos_sin, \
                     kv_bs, (u16*)out_fused, arrive, layer)
#define LAUNCH_DEC_R(GV) do { \
    if (rope) { if (kvnt) LAUNCH_DEC(GV, true, true); \
                else LAUNCH_DEC(GV, true, false); } \
    else { if (kvnt) LAUNCH_DEC(GV, false, true); \
           else LAUNCH_DEC(GV, false, false); } } while (0)
  switch (G) {
    case 1: LAUNCH_DEC_R(1); break;
    case 2: LAUNCH_DEC_R(2); break;
    case 4: LAUNCH_DEC_R(4); break;
    case 8: LAUNCH_DEC_R(8); break;
    default: break;  // unsupported group size: wrapper validates
  }
#undef LAUNCH_DEC_R
#undef LAUNCH_DEC
}

// Combine: out[b,hq,:] = sum_s exp(m_s-m*) o_s / (sum_s exp(m_s-m*) l_s).
// Grid (Hq, B), block D threads. All split partials are loaded in PARALLEL
// (LDS-staged m/l, unrolled independent o loads) — a serial per-split loop
// of dependent global loads cost ~7 us at splits=16 (profiles/r01).
#define CMB_SMAX 64

__global__ void k_attn_decode_combine(u16* __restrict__ out,
                                      const float* __restrict__ part_o,
                                      const float* __restrict__ part_ml,
                                      int B, int Hq, int D, int splits) {
  const int hq = blockIdx.x;
  const int b = blockIdx.y;
  const int d = threadIdx.x;
  __shared__ float sml[CMB_SMAX][2];
  const float* pml = part_ml + ((long)b * Hq + hq) * splits * 2;
  if (d < 2 * splits)
    ((float*)sml)[d] = pml[d];                    // parallel m/l fetch
  __syncthreads();
  float m = -1.0f / 0.0f;
  for (int s = 0; s < splits; ++s) m = fmaxf(m, sml[s][0]);
  const float* po = part_o + ((long)b * Hq + hq) * splits * D;
  float l = 0.f, o = 0.f;
  int s = 0;
  for (; s + 4 <= splits; s += 4) {               // independent load quads
    const float o0 = po[(s + 0) * D + d];
    const float o1 = po[(s + 1) * D + d];
    const float o2 = po[(s + 2) * D + d];
    const float o3 = po[(s + 3) * D + d];
    const float w0 = __expf(sml[s + 0][0] - m);
    const float w1 = __expf(sml[s + 1][0] - m);
    const float w2 = __expf(sml[s + 2][0] - m);
    const float w3 = __expf(sml[s + 3][0] - m);
    l += w0 * sml[s + 0][1] + w1 * sml[s + 1][1] +
         w2 * sml[s + 2][1] + w3 * sml[s + 3][1];
    o += w0 * o0 + w1 * o1 + w2 * o2 + w3 * o3;
  }
  for (; s < splits; ++s) {
    const float w = __expf(sml[s][0] - m);
    l += w * sml[s][1];
    o += w * po[s * D + d];
  }
  out[((long)b * Hq + hq) * D + d] = f2bf(l > 0.f ? o / l : 0.f);
}

void fei_attn_decode_combine(void* out, const float* part_o,
                             const float* part_ml, int B, int Hq, int D,
                             int splits, hipStream_t stream) {
  hipLaunchKernelGGL(k_attn_decode_combine, dim3(Hq, B), dim3(D), 0, stream,
                     (u16*)out, part_o, part_ml, B, Hq, D, splits);
}

// ---------------------------------------------------------------------------
// SwiGLU: gate_up [rows, 2*inter] -> out [rows, inter]; silu(g)*u, f32 math.
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(256)
k_swiglu(u16* __restrict__ out, const u16* __restrict__ gu, long rows,
         int inter) {
  const long nvec = rows * (inter >> 3);
  const int iv = inter >> 3;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < nvec;
       idx += (long)gridDim.x * blockDim.x) {
    const long row = idx / iv;
    const int col = (int)(idx % iv);
    const s16x8* gp = (const s16x8*)(gu + row * 2 * inter) + col;
    const s16x8* up = (const s16x8*)(gu + row * 2 * inter + inter) + col;
    s16x8 g8 = *gp, u8 = *up;
    s16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = bf2f((u16)g8[j]);
      float u = bf2f((u16)u8[j]);
      float silu = g / (1.f + __expf(-g));
      o[j] = (short)f2bf(silu * u);
    }
    ((s16x8*)(out + row * inter))[col] = o;
  }
}

void fei_swiglu(void* out, const void* gate_up, long rows, int inter,
                hipStream_t stream) {
  long nvec = rows * (inter >> 3);
  int grid = (int)min((nvec + 255) / 256, (long)2048);
  if (grid < 1) grid = 1;
  hipLaunchKernelGGL(k_swiglu, dim3(grid), dim3(256), 0, stream,
                     (u16*)out, (const u16*)gate_up, rows, inter);
}

// ---------------------------------------------------------------------------
// Sampling. Stage 1: per (b, chunk) partial argmax of logits/T + Gumbel
// noise (temperature<=0 -> greedy, no noise). Stage 2: final reduce, write
// token[b], out_tokens[b, step]. `step` lives on device so the sequence of
// kernels is hipGraph-replayable.
// ws: [B, nchunks, 2] f32 (value, idx-as-float-bits via int store).
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(256)
k_sample_partial(const u16* __restrict__ logits, float* __restrict__ ws,
                 int B, int vocab, int nchunks, float temperature,
                 u64 seed, const int* __restrict__ step) {
  const int chunk = blockIdx.x;
  const int b = blockIdx.y;
  const int chunk_size = (vocab + nchunks - 1) / nchunks;
  const int start = chunk * chunk_size;
  const int end = min(start + chunk_size, vocab);
  const u64 st = (u64)(*step);
  float best = -1.0f / 0.0f;
  int besti = start;
  const float invT = temperature > 0.f ? 1.f / temperature : 1.f;
  for (int i = start + threadIdx.x; i < end; i += blockDim.x) {
    float v = bf2f(logits[(long)b * vocab + i]) * invT;
    if (temperature > 0.f) {
      float u = hash_uniform(seed ^ (st * 0x51ed27f1ull) ^ ((u64)b << 40) ^ (u64)i);
      v += -__logf(-__logf(u));
    }
    if (v > best) { best = v; besti = i; }
  }
  // wave then block reduce of (best, besti)
#pragma unroll
  for (int off = 32; off; off >>= 1) {
    float ov = __shfl_xor(best, off);
    int oi = __shfl_xor(besti, off);
    if (ov > best || (ov == best && oi < besti)) { best = ov; besti = oi; }
  }
  __shared__ float rv[4];
  __shared__ int ri[4];
  const int wid = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) { rv[wid] = best; ri[wid] = besti; }
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int i = 1; i < (int)(blockDim.x >> 6); ++i)
      if (rv[i] > best || (rv[i] == best && ri[i] < besti)) { best = rv[i]; besti = ri[i]; }
    ws[((long)b * nchunks + chunk) * 2] = best;
    ((int*)ws)[((long)b * nchunks + chunk) * 2 + 1] = besti;
  }
}

__global__ void k_sample_final(const float* __restrict__ ws,
                               int* __restrict__ token,
                               int* __restrict__ out_tokens,
                               const int* __restrict__ step,
                               int B, int nchunks, int max_new) {
  // one wave, parallel chunk loads + shfl argmax (a serial thread-0 loop
  // cost ~8 us at nchunks=64)
  const int b = blockIdx.x;
  const int lane = threadIdx.x & 63;
  float best = -1.0f / 0.0f;
  int besti = 0x7fffffff;
  for (int c = lane; c < nchunks; c += 64) {
    const float v = ws[((long)b * nchunks + c) * 2];
    const int i = ((const int*)ws)[((long)b * nchunks + c) * 2 + 1];
    if (v > best || (v == best && i < besti)) { best = v; besti = i; }
  }
#pragma unroll
  for (int off = 32; off; off >>= 1) {
    const float ov = __shfl_xor(best, off);
    const int oi = __shfl_xor(besti, off);
    if (ov > best || (ov == best && oi < besti)) { best = ov; besti = oi; }
  }
  if (lane == 0) {
    token[b] = besti;
    const int st = *step;
    if (out_tokens && st < max_new) out_tokens[(long)b * max_new + st] = besti;
  }
}

void fei_sample(const void* logits, int* token, int* out_tokens,
                const int* step, float* ws, int B, int vocab, int nchunks,
                float temperature, u64 seed, int max_new,
                hipStream_t stream) {
  hipLaunchKernelGGL(k_sample_partial, dim3(nchunks, B), dim3(256), 0, stream,
                     (const u16*)logits, ws, B, vocab, nchunks, temperature,
                     seed, step);
  hipLaunchKernelGGL(k_sample_final, dim3(B), dim3(64), 0, stream,
                     ws, token, out_tokens, step, B, nchunks, max_new);
}

// advance: pos[b]+=1 ; step+=1  (runs after sampling inside the graph)
// max_pos clamps the position at cache capacity: a caller decoding past
// max_seq would otherwise index the RoPE table / KV cache out of bounds
// from inside the graph (memory corruption, not an exception). Clamped
// steps overwrite the last slot — wrong results but memory-safe; the
// python loops stop at capacity before this matters.
__global__ void k_advance(int* pos, int* step, int B, int max_pos) {
  if ((int)threadIdx.x < B)
    pos[threadIdx.x] = min(pos[threadIdx.x] + 1, max_pos);
  if (threadIdx.x == 0 && step) *step += 1;
}

void fei_advance(int* pos, int* step, int B, int max_pos,
                 hipStream_t stream) {
  hipLaunchKernelGGL(k_advance, dim3(1), dim3(max(B, 64)), 0, stream,
                     pos, step, B, max_pos);
}

}  // extern "C"

// ---------------------------------------------------------------------------
// One-pass sampler for small B: grid (B), block 1024 = 16 waves streaming
// the whole vocab row; replaces the partial+final pair (one node, no
// workspace round trip).
// ---------------------------------------------------------------------------
extern "C" {

__global__ void __launch_bounds__(1024)
k_sample_onepass(const u16* __restrict__ logits, int* __restrict__ token,
                 int* __restrict__ out_tokens, const int* __restrict__ step,
                 int B, int vocab, float temperature, u64 seed, int max_new) {
  const int b = blockIdx.x;
  const int tid = threadIdx.x;
  const u64 st = (u64)(*step);
  const float invT = temperature > 0.f ? 1.f / temperature : 1.f;
  float best = -1.0f / 0.0f;
  int besti = 0x7fffffff;
  const s16x8* row = (const s16x8*)(logits + (long)b * vocab);
  const int nv = vocab >> 3;
  for (int i = tid; i < nv; i += blockDim.x) {
    s16x8 v8 = row[i];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int idx = i * 8 + j;
      float v = bf2f((u16)v8[j]) * invT;
      if (temperature > 0.f) {
        float u = hash_uniform(seed ^ (st * 0x51ed27f1ull) ^ ((u64)b << 40) ^ (u64)idx);
        v += -__logf(-__logf(u));
      }
      if (v > best || (v == best && idx < besti)) { best = v; besti = idx; }
    }
  }
  // tail (vocab % 8)
  for (int idx = nv * 8 + tid; idx < vocab; idx += blockDim.x) {
    float v = bf2f(logits[(long)b * vocab + idx]) * invT;
    if (temperature > 0.f) {
      float u = hash_uniform(seed ^ (st * 0x51ed27f1ull) ^ ((u64)b << 40) ^ (u64)idx);
      v += -__logf(-__logf(u));
    }
    if (v > best || (v == best && idx < besti)) { best = v; besti = idx; }
  }
#pragma unroll
  for (int off = 32; off; off >>= 1) {
    const float ov = __shfl_xor(best, off);
    const int oi = __shfl_xor(besti, off);
    if (ov > best || (ov == best && oi < besti)) { best = ov; besti = oi; }
  }
  __shared__ float rv[16];
  __shared__ int ri[16];
  const int wid = tid >> 6;
  if ((tid & 63) == 0) { rv[wid] = best; ri[wid] = besti; }
  __syncthreads();
  if (tid == 0) {
    const int nw = blockDim.x >> 6;
    for (int i = 1; i < nw; ++i)
      if (rv[i] > best || (rv[i] == best && ri[i] < besti)) {
        best = rv[i]; besti = ri[i];
      }
    token[b] = besti;
    if (out_tokens && (int)st < max_new)
      out_tokens[(long)b * max_new + (int)st] = besti;
  }
}

void fei_sample_onepass(const void* logits, int* token, int* out_tokens,
                        const int* step, int B, int vocab, float temperature,
                        u64 seed, int max_new, hipStream_t stream) {
  hipLaunchKernelGGL(k_sample_onepass, dim3(B), dim3(1024), 0, stream,
                     (const u16*)logits, token, out_tokens, step, B, vocab,
                     temperature, seed, max_new);
}

// Shard sampler for tensor-parallel decode: each rank samples over its OWN
// lm_head shard [B, vocab_local] and emits (best value, best GLOBAL index)
// into out[b] = {f32 val, i32 idx}; the ranks then all-gather 8 bytes per
// sequence instead of the full vocab row (replaces the 513 KB logits
// all-gather per step at Llama-3 vocab). Gumbel noise is keyed by the
// GLOBAL index, so the sharded sample is bit-identical to the full-vocab
// k_sample_onepass on the gathered logits.
__global__ void __launch_bounds__(1024)
k_sample_shard(const u16* __restrict__ logits, float* __restrict__ out,
               const int* __restrict__ step, int B, int vocab_local,
               int v_offset, float temperature, u64 seed) {
  const int b = blockIdx.x;
  const int tid = threadIdx.x;
  const u64 st = (u64)(*step);
  const float invT = temperature > 0.f ? 1.f / temperature : 1.f;
  float best = -1.0f / 0.0f;
  int besti = 0x7fffffff;
  const s16x8* row = (const s16x8*)(logits + (long)b * vocab_local);
  const int nv = vocab_local >> 3;
  for (int i = tid; i < nv; i += blockDim.x) {
    s16x8 v8 = row[i];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int idx = v_offset + i * 8 + j;
      float v = bf2f((u16)v8[j]) * invT;
      if (temperature > 0.f) {
        float u = hash_uniform(seed ^ (st * 0x51ed27f1ull) ^ ((u64)b << 40) ^ (u64)idx);
        v += -__logf(-__logf(u));
      }
      if (v > best || (v == best && idx < besti)) { best = v; besti = idx; }
    }
  }
  for (int k = nv * 8 + tid; k < vocab_local; k += blockDim.x) {
    const int idx = v_offset + k;
    float v = bf2f(logits[(long)b * vocab_local + k]) * invT;
    if (temperature > 0.f) {
      float u = hash_uniform(seed ^ (st * 0x51ed27f1ull) ^ ((u64)b << 40) ^ (u64)idx);
      v += -__logf(-__logf(u));
    }
    if (v > best || (v == best && idx < besti)) { best = v; besti = idx; }
  }
#pragma unroll
  for (int off = 32; off; off >>= 1) {
    const float ov = __shfl_xor(best, off);
    const int oi = __shfl_xor(besti, off);
    if (ov > best || (ov == best && oi < besti)) { best = ov; besti = oi; }
  }
  __shared__ float rv[16];
  __shared__ int ri[16];
  const int wid = tid >> 6;
  if ((tid & 63) == 0) { rv[wid] = best; ri[wid] = besti; }
  __syncthreads();
  if (tid == 0) {
    const int nw = blockDim.x >> 6;
    for (int i = 1; i < nw; ++i)
      if (rv[i] > best || (rv[i] == best && ri[i] < besti)) {
        best = rv[i]; besti = ri[i];
      }
    out[(long)b * 2] = best;
    ((int*)out)[(long)b * 2 + 1] = besti;
  }
}

void fei_sample_shard(const void* logits, float* out, const int* step,
                      int B, int vocab_local, int v_offset, float temperature,
                      u64 seed, hipStream_t stream) {
  hipLaunchKernelGGL(k_sample_shard, dim3(B), dim3(1024), 0, stream,
                     (const u16*)logits, out, step, B, vocab_local, v_offset,
                     temperature, seed);
}

}  // extern "C"

// ---------------------------------------------------------------------------
// Paged decode attention: K/V live in a block pool [num_blocks, Hkv, BS, D]
// and a per-sequence block table maps logical key blocks to physical pool
// blocks (engine/kv_cache.py's PagedKVPool). Same split-K online-softmax
// structure as k_attn_decode (no in-kernel rope here: serving stacks rope
// at the scheduler level); writes the same part_o/part_ml partials, reused
// combine kernel. BS is a power of two (bs_log).
// ---------------------------------------------------------------------------

namespace {

// ROPE=true fuses the per-step RoPE + paged KV-append exactly like the
// plain kernel (kin/vin raw, cos_sin table; the split that OWNS key n-1
// ropes and writes the pool row via the block table before its own
// tile pass — no other split touches that key). Replaces the serving
// path's ~15 torch-op launches per layer (apply_rope x2, contiguous,
// 2 advanced-index writes), which were ~30% of the sessions step.
template <int G, bool ROPE = false>
__global__ void __launch_bounds__(256)
k_attn_decode_paged(const u16* __restrict__ q, u16* __restrict__ kp,
                    u16* __restrict__ vp,
                    const int* __restrict__ block_table,
                    float* __restrict__ part_o, float* __restrict__ part_ml,
                    const int* __restrict__ pos,
                    int B, int Hq, int Hkv, int D, int bs_log,
                    int max_blocks, int splits, float scale, long q_bs,
                    const u16* __restrict__ kin, const u16* __restrict__ vin,
                    const float* __restrict__ cos_sin, long kv_bs) {
  const int split = blockIdx.x;
  const int hkv = blockIdx.y;
  const int b = blockIdx.z;
  const int tid = threadIdx.x;

  __shared__ float qs[DEC_GMAX][DEC_DMAX];
  __shared__ float pl[DEC_GMAX][DEC_TILE];
  __shared__ float red[DEC_GMAX][4];
  __shared__ float osh[DEC_TILE][8];   // [kg * dvecs + dv][element]

  const int n = pos[b] + 1;
  const int chunk = (n + splits - 1) / splits;
  const int start = split * chunk;
  const int end = min(start + chunk, n);
  const int BS = 1 << bs_log;
  const int* bt = block_table + (long)b * max_blocks;

  if (start >= end) {
#pragma unroll
    for (int g = 0; g < G; ++g) {
      const int hq = hkv * G + g;
      float* po = part_o + (((long)b * Hq + hq) * splits + split) * D;
      float* pml = part_ml + (((long)b * Hq + hq) * splits + split) * 2;
      for (int d = tid; d < D; d += blockDim.x) po[d] = 0.f;
      if (tid == 0) { pml[0] = -1.0f / 0.0f; pml[1] = 0.f; }
    }
    return;
  }

  auto row_ptr = [&](const u16* pool, int kk) -> const u16* {
    const int blk = bt[kk >> bs_log];
    const int within = kk & (BS - 1);
    return pool + (((long)blk * Hkv + hkv) * BS + within) * D;
  };

  const int half = D / 2;
  const int p_new = n - 1;
  if (ROPE) {
    // stage roped+scaled q (same math as the plain kernel's ROPE path)
    for (int i = tid; i < G * half; i += blockDim.x) {
      const int g = i / half, d = i % half;
      const float c = cos_sin[((long)p_new * half + d) * 2 + 0];
      const float sn = cos_sin[((long)p_new * half + d) * 2 + 1];
      const u16* qp = q + (long)b * q_bs + (long)(hkv * G + g) * D;
      const float x1 = bf2f(qp[d]);
      const float x2 = bf2f(qp[d + half]);
      qs[g][d] = (x1 * c - x2 * sn) * scale;
      qs[g][d + half] = (x2 * c + x1 * sn) * scale;
    }
    if (p_new >= start && p_new < end) {
      u16* kcp = (u16*)row_ptr(kp, p_new);
      u16* vcp = (u16*)row_ptr(vp, p_new);
      for (int d = tid; d < half; d += blockDim.x) {
        const float c = cos_sin[((long)p_new * half + d) * 2 + 0];
        const float sn = cos_sin[((long)p_new * half + d) * 2 + 1];
        const u16* kq = kin + (long)b * kv_bs + (long)hkv * D;
        const u16* vq = vin + (long)b * kv_bs + (long)hkv * D;
        const float x1 = bf2f(kq[d]);
        const float x2 = bf2f(kq[d + half]);
        kcp[d] = f2bf(x1 * c - x2 * sn);
        kcp[d + half] = f2bf(x2 * c + x1 * sn);
        vcp[d] = vq[d];
        vcp[d + half] = vq[d + half];
      }
    }
  } else {
    for (int i = tid; i < G * D; i += blockDim.x) {
      const int g = i / D, d = i % D;
      qs[g][d] = bf2f(q[(long)b * q_bs + (long)(hkv * G + g) * D + d]) * scale;
    }
  }
  __syncthreads();

  // chunk-adaptive V lane maps — same scheme as k_attn_decode (wide
  // 16 B/lane once a split's chunk covers >= 64 keys; pair otherwise)
  const int dvecs = D / 8;
  const int kgroups = blockDim.x / dvecs;
  const int keys_per_group = DEC_TILE / kgroups;
  const int dv = tid % dvecs;
  const int kg = tid / dvecs;
  const bool wide = chunk >= 64;
  const int dpairs = D / 2;
  const int kgroupsP = blockDim.x / dpairs;
  const int keysP = DEC_TILE / kgroupsP;
  const int dp = tid % dpairs;
  const int kgP = tid / dpairs;

  float m[DEC_GMAX], l[DEC_GMAX], sc[DEC_GMAX];
  float oa[DEC_GMAX][8];
#pragma unroll
  for (int g = 0; g < G; ++g) {
    m[g] = -1.0f / 0.0f; l[g] = 0.f;
#pragma unroll
    for (int e = 0; e < 8; ++e) oa[g][e] = 0.f;
  }

  for (int tile = start; tile < end; tile += DEC_TILE) {
    const int kk = tile + tid;
    if (kk < end) {
      const s16x8* krow = (const s16x8*)row_ptr(kp, kk);
#pragma unroll
      for (int g = 0; g < G; ++g) sc[g] = 0.f;
#pragma unroll 4
      for (int i = 0; i < D / 8; ++i) {
        s16x8 kv8 = krow[i];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const float kf = bf2f((u16)kv8[j]);
#pragma unroll
          for (int g = 0; g < G; ++g)
            sc[g] = fmaf(qs[g][i * 8 + j], kf, sc[g]);
        }
      }
    } else {
#pragma unroll
      for (int g = 0; g < G; ++g) sc[g] = -1.0f / 0.0f;
    }
    float tile_m[DEC_GMAX];
#pragma unroll
    for (int g = 0; g < G; ++g) tile_m[g] = sc[g];
    block_reduce_vec<0, G>(tile_m, red);
    float alpha[DEC_GMAX], pv[DEC_GMAX];
#pragma unroll
    for (int g = 0; g < G; ++g) {
      const float m_new = fmaxf(m[g], tile_m[g]);
      alpha[g] = __expf(m[g] - m_new);
      pv[g] = (kk < end) ? __expf(sc[g] - m_new) : 0.f;
      pl[g][tid] = pv[g];
      m[g] = m_new;
    }
    float tile_sum[DEC_GMAX];
#pragma unroll
    for (int g = 0; g < G; ++g) tile_sum[g] = pv[g];
    block_reduce_vec<1, G>(tile_sum, red);
#pragma unroll
    for (int g = 0; g < G; ++g) l[g] = l[g] * alpha[g] + tile_sum[g];

#pragma unroll
    for (int g = 0; g < G; ++g) {
#pragma unroll
      for (int e = 0; e < 8; ++e) oa[g][e] *= alpha[g];
    }
    const int kmax = min(DEC_TILE, end - tile);
    if (wide) {
      const int kbase_local = kg * keys_per_group;
      const int iters = min(keys_per_group, max(0, kmax - kbase_local));
#pragma unroll 2
      for (int j = 0; j < iters; ++j) {
        const int kl = kbase_local + j;
        const s16x8 v8 = *((const s16x8*)row_ptr(vp, tile + kl) + dv);
        float vf[8];
#pragma unroll
        for (int e = 0; e < 8; ++e) vf[e] = bf2f((u16)v8[e]);
#pragma unroll
        for (int g = 0; g < G; ++g) {
          const float p = pl[g][kl];
#pragma unroll
          for (int e = 0; e < 8; ++e) oa[g][e] = fmaf(p, vf[e], oa[g][e]);
        }
      }
    } else {
      const int kbase_local = kgP * keysP;
      const int iters = min(keysP, max(0, kmax - kbase_local));
#pragma unroll 4
      for (int j = 0; j < iters; ++j) {
        const int kl = kbase_local + j;
        const u16* vrow = row_ptr(vp, tile + kl) + dp * 2;
        const float v0 = bf2f(vrow[0]);
        const float v1 = bf2f(vrow[1]);
#pragma unroll
        for (int g = 0; g < G; ++g) {
          const float p = pl[g][kl];
          oa[g][0] = fmaf(p, v0, oa[g][0]);
          oa[g][1] = fmaf(p, v1, oa[g][1]);
        }
      }
    }
    __syncthreads();
  }

#pragma unroll
  for (int g = 0; g < G; ++g) {
    const int hq = hkv * G + g;
    float* po = part_o + (((long)b * Hq + hq) * splits + split) * D;
    float* pml = part_ml + (((long)b * Hq + hq) * splits + split) * 2;
    if (wide) {
#pragma unroll
      for (int e = 0; e < 8; ++e) osh[kg * dvecs + dv][e] = oa[g][e];
      __syncthreads();
      if (kg == 0) {
        float s[8];
#pragma unroll
        for (int e = 0; e < 8; ++e) s[e] = osh[dv][e];
        for (int gg = 1; gg < kgroups; ++gg) {
#pragma unroll
          for (int e = 0; e < 8; ++e) s[e] += osh[gg * dvecs + dv][e];
        }
#pragma unroll
        for (int e = 0; e < 8; ++e) po[dv * 8 + e] = s[e];
      }
    } else {
      osh[kgP * dpairs + dp][0] = oa[g][0];
      osh[kgP * dpairs + dp][1] = oa[g][1];
      __syncthreads();
      if (kgP == 0) {
        float s0 = osh[dp][0], s1 = osh[dp][1];
        for (int gg = 1; gg < kgroupsP; ++gg) {
          s0 += osh[gg * dpairs + dp][0];
          s1 += osh[gg * dpairs + dp][1];
        }
        po[dp * 2] = s0;
        po[dp * 2 + 1] = s1;
      }
    }
    if (tid == 0) { pml[0] = m[g]; pml[1] = l[g]; }
    __syncthreads();
  }
}

}  // namespace

extern "C" {

void fei_attn_decode_paged(const void* q, void* k_pool,
                           void* v_pool, const int* block_table,
                           float* part_o, float* part_ml, const int* pos,
                           int B, int Hq, int Hkv, int D, int bs_log,
                           int max_blocks, int splits, float scale,
                           long q_bs,
                           const void* kin, const void* vin,
                           const float* cos_sin, long kv_bs,
                           hipStream_t stream) {
  const int G = Hq / Hkv;
  dim3 grid(splits, Hkv, B);
  const int rope = cos_sin != nullptr;
#define LP(GV, RV) hipLaunchKernelGGL((k_attn_decode_paged<GV, RV>), grid, \
    dim3(256), 0, stream, (const u16*)q, (u16*)k_pool, (u16*)v_pool, \
    block_table, part_o, part_ml, pos, B, Hq, Hkv, D, bs_log, max_blocks, \
    splits, scale, q_bs, (const u16*)kin, (const u16*)vin, cos_sin, kv_bs)
#define LPR(GV) do { if (rope) LP(GV, true); else LP(GV, false); } while (0)
  switch (G) {
    case 1: LPR(1); break;
    case 2: LPR(2); break;
    case 4: LPR(4); break;
    case 8: LPR(8); break;
    default: break;
  }
#undef LPR
#undef LP
}

}  // extern "C"

// ---------------------------------------------------------------------------
// LayerNorm (+ optional fused residual add) and bias+GELU — the encoder-side
// kernels (BERT/bge architecture: post-LN residual blocks, GELU MLP).
// bf16 I/O, f32 statistics. One block per row.
// ---------------------------------------------------------------------------
extern "C" {

__global__ void __launch_bounds__(256)
k_add_layernorm(u16* __restrict__ out, const u16* __restrict__ x,
                const u16* __restrict__ res, const u16* __restrict__ w,
                const u16* __restrict__ bias, int rows, int cols, float eps,
                int has_res) {
  __shared__ float red[4];
  const int nv = cols >> 3;
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const s16x8* xr = (const s16x8*)(x + (long)row * cols);
    const s16x8* rr = (const s16x8*)(res + (long)row * cols);
    float sum = 0.f;
    for (int i = threadIdx.x; i < nv; i += blockDim.x) {
      s16x8 xv = xr[i];
      s16x8 rv = has_res ? rr[i] : xv;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf2f((u16)xv[j]) + (has_res ? bf2f((u16)rv[j]) : 0.f);
        sum += f;
      }
    }
    sum = block_reduce_sum(sum, red);
    const float mean = sum / (float)cols;
    float var = 0.f;
    for (int i = threadIdx.x; i < nv; i += blockDim.x) {
      s16x8 xv = xr[i];
      s16x8 rv = has_res ? rr[i] : xv;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf2f((u16)xv[j]) + (has_res ? bf2f((u16)rv[j]) : 0.f) - mean;
        var = fmaf(f, f, var);
      }
    }
    var = block_reduce_sum(var, red);
    const float inv = rsqrtf(var / (float)cols + eps);
    s16x8* orow = (s16x8*)(out + (long)row * cols);
    const s16x8* wv8 = (const s16x8*)w;
    const s16x8* bv8 = (const s16x8*)bias;
    for (int i = threadIdx.x; i < nv; i += blockDim.x) {
      s16x8 xv = xr[i];
      s16x8 rv = has_res ? rr[i] : xv;
      s16x8 wv = wv8[i];
      s16x8 bv = bv8[i];
      s16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf2f((u16)xv[j]) + (has_res ? bf2f((u16)rv[j]) : 0.f);
        o[j] = (short)f2bf((f - mean) * inv * bf2f((u16)wv[j]) +
                           bf2f((u16)bv[j]));
      }
      orow[i] = o;
    }
    __syncthreads();
  }
}

void fei_add_layernorm(void* out, const void* x, const void* res,
                       const void* w, const void* bias, int rows, int cols,
                       float eps, int has_res, hipStream_t stream) {
  int grid = rows < 2048 ? rows : 2048;
  hipLaunchKernelGGL(k_add_layernorm, dim3(grid), dim3(256), 0, stream,
                     (u16*)out, (const u16*)x, (const u16*)res,
                     (const u16*)w, (const u16*)bias, rows, cols, eps,
                     has_res);
}

// exact GELU (erf form, matching torch's default): x * 0.5 * (1 + erf(x/sqrt(2)))
__global__ void __launch_bounds__(256)
k_gelu(u16* __restrict__ out, const u16* __restrict__ x, long n8) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (long)gridDim.x * blockDim.x) {
    s16x8 v = ((const s16x8*)x)[i];
    s16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float f = bf2f((u16)v[j]);
      o[j] = (short)f2bf(f * 0.5f * (1.f + erff(f * 0.70710678f)));
    }
    ((s16x8*)out)[i] = o;
  }
}

void fei_gelu(void* out, const void* x, long n, hipStream_t stream) {
  long n8 = n >> 3;
  int grid = (int)((n8 + 255) / 256);
  if (grid > 2048) grid = 2048;
  if (grid < 1) grid = 1;
  hipLaunchKernelGGL(k_gelu, dim3(grid), dim3(256), 0, stream,
                     (u16*)out, (const u16*)x, n8);
}

}  // extern "C"
