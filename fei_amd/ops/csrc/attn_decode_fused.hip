// Fused decode attention for agent-length sequences (gfx950).
// ONE kernel per layer replaces {rope_kv_decode, attn_decode, combine}:
//   grid (Hkv, B), block 256. Each workgroup serves a whole GQA group:
//   1. stages the group's q vectors into LDS, applying RoPE + scale
//   2. ropes the new token's k, writes k/v into the caches at pos[b]
//      (same block reads them back after __syncthreads — workgroup-scope
//      visibility on one CU)
//   3. online-softmax attention over n = pos[b]+1 keys (thread-per-key
//      scores, (D/2 x key-group) P*V), bf16 output — no split partials.
// Removes 3 graph nodes per layer and the split/combine latency; the
// split-K kernel (k_attn_decode) remains for long-context (engine picks
// by max_seq_len).
#include "fei_common.h"

#define DEC_TILE 256
#define DEC_DMAX 128
#define DEC_GMAX 8

namespace {

template <int OP, int G>  // OP: 0 = max, 1 = sum
__device__ __forceinline__ void f_block_reduce_vec(float* v,
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

template <int G>
__global__ void __launch_bounds__(256)
k_attn_decode_fused(const u16* __restrict__ q, const u16* __restrict__ kin,
                    const u16* __restrict__ vin, u16* __restrict__ kc,
                    u16* __restrict__ vc, u16* __restrict__ out,
                    const float* __restrict__ cos_sin,
                    const int* __restrict__ pos,
                    int B, int Hq, int Hkv, int D, int max_seq, float scale,
                    long q_bs, long kv_bs) {
  const int hkv = blockIdx.x;
  const int b = blockIdx.y;
  const int tid = threadIdx.x;

  __shared__ float qs[DEC_GMAX][DEC_DMAX];
  __shared__ float pl[DEC_GMAX][DEC_TILE];
  __shared__ float red[DEC_GMAX][4];
  __shared__ float osh[8][DEC_DMAX / 2][2];

  const int p = pos[b];
  const int n = p + 1;
  const int half = D / 2;

  // ---- stage roped+scaled q for the whole group -------------------------
  for (int i = tid; i < G * half; i += blockDim.x) {
    const int g = i / half, d = i % half;
    const float c = cos_sin[((long)p * half + d) * 2 + 0];
    const float s = cos_sin[((long)p * half + d) * 2 + 1];
    const u16* qp = q + (long)b * q_bs + (long)(hkv * G + g) * D;
    const float x1 = bf2f(qp[d]);
    const float x2 = bf2f(qp[d + half]);
    qs[g][d] = (x1 * c - x2 * s) * scale;
    qs[g][d + half] = (x2 * c + x1 * s) * scale;
  }
  // ---- rope + append the new token's k/v --------------------------------
  u16* kcp = kc + (((long)b * Hkv + hkv) * max_seq + p) * D;
  u16* vcp = vc + (((long)b * Hkv + hkv) * max_seq + p) * D;
  for (int d = tid; d < half; d += blockDim.x) {
    const float c = cos_sin[((long)p * half + d) * 2 + 0];
    const float s = cos_sin[((long)p * half + d) * 2 + 1];
    const u16* kp = kin + (long)b * kv_bs + (long)hkv * D;
    const u16* vp = vin + (long)b * kv_bs + (long)hkv * D;
    const float x1 = bf2f(kp[d]);
    const float x2 = bf2f(kp[d + half]);
    kcp[d] = f2bf(x1 * c - x2 * s);
    kcp[d + half] = f2bf(x2 * c + x1 * s);
    vcp[d] = vp[d];
    vcp[d + half] = vp[d + half];
  }
  __syncthreads();   // qs ready; cache writes visible within this block

  const u16* kbase = kc + ((long)b * Hkv + hkv) * max_seq * D;
  const u16* vbase = vc + ((long)b * Hkv + hkv) * max_seq * D;

  const int dpairs = D / 2;
  const int kgroups = blockDim.x / dpairs;
  const int keys_per_group = DEC_TILE / kgroups;
  const int dp = tid % dpairs;
  const int kg = tid / dpairs;

  float m[DEC_GMAX], l[DEC_GMAX], sc[DEC_GMAX];
  float o0[DEC_GMAX], o1[DEC_GMAX];
#pragma unroll
  for (int g = 0; g < G; ++g) {
    m[g] = -1.0f / 0.0f; l[g] = 0.f; o0[g] = 0.f; o1[g] = 0.f;
  }

  for (int tile = 0; tile < n; tile += DEC_TILE) {
    const int kk = tile + tid;
    if (kk < n) {
      const s16x8* krow = (const s16x8*)(kbase + (long)kk * D);
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
    f_block_reduce_vec<0, G>(tile_m, red);
    float alpha[DEC_GMAX], pv[DEC_GMAX];
#pragma unroll
    for (int g = 0; g < G; ++g) {
      const float m_new = fmaxf(m[g], tile_m[g]);
      alpha[g] = __expf(m[g] - m_new);
      pv[g] = (kk < n) ? __expf(sc[g] - m_new) : 0.f;
      pl[g][tid] = pv[g];
      m[g] = m_new;
    }
    float tile_sum[DEC_GMAX];
#pragma unroll
    for (int g = 0; g < G; ++g) tile_sum[g] = pv[g];
    f_block_reduce_vec<1, G>(tile_sum, red);
#pragma unroll
    for (int g = 0; g < G; ++g) l[g] = l[g] * alpha[g] + tile_sum[g];

#pragma unroll
    for (int g = 0; g < G; ++g) { o0[g] *= alpha[g]; o1[g] *= alpha[g]; }
    const int kbase_local = kg * keys_per_group;
    const int kmax = min(DEC_TILE, n - tile);
    const int iters = min(keys_per_group, max(0, kmax - kbase_local));
#pragma unroll 4
    for (int j = 0; j < iters; ++j) {
      const int kl = kbase_local + j;
      const u16* vrow = vbase + (long)(tile + kl) * D + dp * 2;
      const float v0 = bf2f(vrow[0]);
      const float v1 = bf2f(vrow[1]);
#pragma unroll
      for (int g = 0; g < G; ++g) {
        const float pp = pl[g][kl];
        o0[g] = fmaf(pp, v0, o0[g]);
        o1[g] = fmaf(pp, v1, o1[g]);
      }
    }
    __syncthreads();
  }

  // ---- epilogue: combine key-groups, write bf16 -------------------------
#pragma unroll
  for (int g = 0; g < G; ++g) {
    osh[kg][dp][0] = o0[g];
    osh[kg][dp][1] = o1[g];
    __syncthreads();
    if (kg == 0) {
      float s0 = osh[0][dp][0], s1 = osh[0][dp][1];
      for (int gg = 1; gg < kgroups; ++gg) {
        s0 += osh[gg][dp][0];
        s1 += osh[gg][dp][1];
      }
      const float inv_l = l[g] > 0.f ? 1.f / l[g] : 0.f;
      u16* orow = out + ((long)b * Hq + hkv * G + g) * D;
      orow[dp * 2] = f2bf(s0 * inv_l);
      orow[dp * 2 + 1] = f2bf(s1 * inv_l);
    }
    __syncthreads();
  }
}

}  // namespace

extern "C" {

void fei_attn_decode_fused(const void* q, const void* kin, const void* vin,
                           void* k_cache, void* v_cache, void* out,
                           const float* cos_sin, const int* pos,
                           int B, int Hq, int Hkv, int D, int max_seq,
                           float scale, long q_bs, long kv_bs,
                           hipStream_t stream) {
  const int G = Hq / Hkv;
  dim3 grid(Hkv, B);
#define LF(GV) hipLaunchKernelGGL(k_attn_decode_fused<GV>, grid, dim3(256), \
                                  0, stream, (const u16*)q, (const u16*)kin, \
                                  (const u16*)vin, (u16*)k_cache, \
                                  (u16*)v_cache, (u16*)out, cos_sin, pos, \
                                  B, Hq, Hkv, D, max_seq, scale, q_bs, kv_bs)
  switch (G) {
    case 1: LF(1); break;
    case 2: LF(2); break;
    case 4: LF(4); break;
    case 8: LF(8); break;
    default: break;   // wrapper validates
  }
#undef LF
}

}  // extern "C"
