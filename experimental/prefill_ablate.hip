// Prefill-attention ablation (guide §5.4 rule 8: ablate before tuning).
// Standalone copy of ops/csrc/attn_prefill.hip's k_attn_prefill<128> with
// template<int CUT> phase stubs, each dead value kept live with an empty
// asm (rule 17 — otherwise DCE deletes the upstream phase too):
//   CUT 0 FULL        1 NOSM (skip mask+softmax VALU)
//   2 NOPV (skip PV: LDS scalar V reads + MFMAs)   3 NOQK (skip QK MFMAs)
//   4 NOSTAGE (skip K/V LDS staging: garbage data, timing only)
// Llama-3-8B prefill shape: B1, S2048, Hq32, Hkv8, D128. Prints ms and
// attributed share per phase.
#include <hip/hip_runtime.h>
#include <cstdio>

typedef unsigned short u16;
typedef unsigned int u32;
typedef __attribute__((ext_vector_type(8))) short s16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

__device__ __forceinline__ float bf2f(u16 u) {
  union { float f; u32 i; } cv; cv.i = ((u32)u) << 16; return cv.f;
}
__device__ __forceinline__ u16 f2bf(float f) {
  union { float f; u32 i; } cv; cv.f = f;
  u32 x = cv.i; u32 lsb = (x >> 16) & 1u; x += 0x7fffu + lsb;
  return (u16)(x >> 16);
}
template <typename T>
__device__ __forceinline__ void keep(T& v) { asm volatile("" : "+v"(v)); }

constexpr int D = 128, DC = D / 16, KS = D / 32, KVT = 64, NC = 4, KA = 2;

template <int CUT>
__global__ void __launch_bounds__(256)
k_ablate(const u16* __restrict__ q, const u16* __restrict__ kc,
         const u16* __restrict__ vc, u16* __restrict__ out,
         int S, int Hq, int Hkv, int max_seq, float scale, long q_ts) {
  const int qt = blockIdx.x, hq = blockIdx.y;
  const int hkv = hq / (Hq / Hkv);
  const int tid = threadIdx.x, w = tid >> 6, lane = tid & 63;
  const int l15 = lane & 15, lg = lane >> 4;
  __shared__ u16 kt[64][D];
  __shared__ u16 vt[64][D];
  __shared__ u16 p_lds[4][16][64];
#define SWZ16(row, col8) ((col8) ^ ((row) & 7))
  const int q_hi = min(qt * 64 + 64, S);
  const int kv_end = q_hi;                        // causal, pos0 = 0
  const int qrow_ld = min(qt * 64 + w * 16 + l15, S - 1);
  s16x8 a_q[KS];
  {
    const u16* qp = q + (long)qrow_ld * q_ts + (long)hq * D + 8 * lg;
#pragma unroll
    for (int ks = 0; ks < KS; ++ks) a_q[ks] = *(const s16x8*)(qp + ks * 32);
  }
  float m_row[4], l_row[4];
  f32x4 o_acc[DC];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m_row[r] = -1.0f / 0.0f; l_row[r] = 0.f; }
#pragma unroll
  for (int dc = 0; dc < DC; ++dc) o_acc[dc] = f32x4{0.f, 0.f, 0.f, 0.f};
  const u16* kbase = kc + (long)hkv * max_seq * D;
  const u16* vbase = vc + (long)hkv * max_seq * D;
  const int ntiles = (kv_end + KVT - 1) / KVT;
  for (int t = 0; t < ntiles; ++t) {
    __syncthreads();
    if (CUT != 4) {
      const int nv8 = KVT * D / 8;
      for (int i = tid; i < nv8; i += 256) {
        const int key = i / (D / 8), col8 = i % (D / 8);
        const int kk = t * KVT + key;
        const int dst = key * (D / 8) + SWZ16(key, col8);
        s16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
        if (kk < kv_end) {
          ((s16x8*)kt)[dst] = *(const s16x8*)(kbase + (long)kk * D + col8 * 8);
          ((s16x8*)vt)[dst] = *(const s16x8*)(vbase + (long)kk * D + col8 * 8);
        } else { ((s16x8*)kt)[dst] = z; ((s16x8*)vt)[dst] = z; }
      }
    }
    __syncthreads();
    f32x4 sfrag[NC];
    if (CUT != 3) {
#pragma unroll
      for (int c = 0; c < NC; ++c) {
        f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int ks = 0; ks < KS; ++ks) {
          const int krow = c * 16 + l15;
          const int kcol8 = SWZ16(krow, ks * 4 + lg);
          s16x8 b_k = *(const s16x8*)(&((s16x8*)kt)[krow * (D / 8) + kcol8]);
          acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_q[ks], b_k, acc,
                                                        0, 0, 0);
        }
        sfrag[c] = acc;
      }
    } else {
#pragma unroll
      for (int c = 0; c < NC; ++c) {
        sfrag[c] = f32x4{0.1f, 0.2f, 0.3f, 0.4f};
        keep(sfrag[c]);
      }
    }
    float p_val[NC][4];
    float alpha[4];
    if (CUT != 1) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow_abs = qt * 64 + w * 16 + 4 * lg + r;
        float sv[NC]; float mx = -1.0f / 0.0f;
#pragma unroll
        for (int c = 0; c < NC; ++c) {
          sv[c] = sfrag[c][r] * scale;
          const int key = t * KVT + c * 16 + l15;
          if (key >= kv_end || key > qrow_abs) sv[c] = -1.0f / 0.0f;
          mx = fmaxf(mx, sv[c]);
        }
#pragma unroll
        for (int off = 1; off < 16; off <<= 1)
          mx = fmaxf(mx, __shfl_xor(mx, off));
        const float m_new = fmaxf(m_row[r], mx);
        alpha[r] = __expf(m_row[r] - m_new);
        m_row[r] = m_new;
        float psum = 0.f;
#pragma unroll
        for (int c = 0; c < NC; ++c) {
          p_val[c][r] = (sv[c] == -1.0f / 0.0f) ? 0.f : __expf(sv[c] - m_new);
          psum += p_val[c][r];
        }
#pragma unroll
        for (int off = 1; off < 16; off <<= 1) psum += __shfl_xor(psum, off);
        l_row[r] = l_row[r] * alpha[r] + psum;
      }
    } else {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        alpha[r] = 1.f;
#pragma unroll
        for (int c = 0; c < NC; ++c) {
          p_val[c][r] = sfrag[c][r];      // keeps QK live without softmax
          keep(p_val[c][r]);
        }
      }
    }
    if (CUT != 2) {
#pragma unroll
      for (int dc = 0; dc < DC; ++dc)
#pragma unroll
        for (int r = 0; r < 4; ++r) o_acc[dc][r] *= alpha[r];
#pragma unroll
      for (int c = 0; c < NC; ++c)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          p_lds[w][4 * lg + r][c * 16 + l15] = f2bf(p_val[c][r]);
      s16x8 a_p[KA];
#pragma unroll
      for (int ka = 0; ka < KA; ++ka)
        a_p[ka] = *(const s16x8*)(&p_lds[w][l15][ka * 32 + 8 * lg]);
#pragma unroll
      for (int dc = 0; dc < DC; ++dc) {
#pragma unroll
        for (int ka = 0; ka < KA; ++ka) {
          s16x8 b_v;
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            const int vrow = ka * 32 + 8 * lg + j;
            const int col = dc * 16 + l15;
            const int vcol8 = SWZ16(vrow, col >> 3);
            b_v[j] = (short)((u16*)vt)[vrow * D + vcol8 * 8 + (col & 7)];
          }
          o_acc[dc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_p[ka], b_v, o_acc[dc], 0, 0, 0);
        }
      }
    } else {
#pragma unroll
      for (int c = 0; c < NC; ++c)
#pragma unroll
        for (int r = 0; r < 4; ++r) keep(p_val[c][r]);
#pragma unroll
      for (int dc = 0; dc < DC; ++dc) {
        o_acc[dc][0] += 1e-7f;
        keep(o_acc[dc]);
      }
    }
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int srow = qt * 64 + w * 16 + 4 * lg + r;
    if (srow >= S) continue;
    const float inv_l = l_row[r] > 0.f ? 1.f / l_row[r] : 1.f;
    u16* orow = out + ((long)srow * Hq + hq) * D;
#pragma unroll
    for (int dc = 0; dc < DC; ++dc)
      orow[dc * 16 + l15] = f2bf(o_acc[dc][r] * inv_l);
  }
}


// v2: V stored transposed [col][key] with 8-key block swizzle; PV
// B-fragment = ONE 16-byte vector read (the shipped r02 layout)
__global__ void __launch_bounds__(256)
k_ablate_v2(const u16* __restrict__ q, const u16* __restrict__ kc,
            const u16* __restrict__ vc, u16* __restrict__ out,
            int S, int Hq, int Hkv, int max_seq, float scale, long q_ts) {
  const int qt = blockIdx.x, hq = blockIdx.y;
  const int hkv = hq / (Hq / Hkv);
  const int tid = threadIdx.x, w = tid >> 6, lane = tid & 63;
  const int l15 = lane & 15, lg = lane >> 4;
  __shared__ u16 kt[64][D];
  __shared__ u16 vtT[D][64];
  __shared__ u16 p_lds[4][16][64];
#define VT_OFF(col, key) \
  ((col) * 64 + \
   ((((key) >> 3) ^ ((col) & 7) ^ (((col) >> 3) & 7)) << 3) + ((key) & 7))
  const int q_hi = min(qt * 64 + 64, S);
  const int kv_end = q_hi;
  const int qrow_ld = min(qt * 64 + w * 16 + l15, S - 1);
  s16x8 a_q[KS];
  {
    const u16* qp = q + (long)qrow_ld * q_ts + (long)hq * D + 8 * lg;
#pragma unroll
    for (int ks = 0; ks < KS; ++ks) a_q[ks] = *(const s16x8*)(qp + ks * 32);
  }
  float m_row[4], l_row[4];
  f32x4 o_acc[DC];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m_row[r] = -1.0f / 0.0f; l_row[r] = 0.f; }
#pragma unroll
  for (int dc = 0; dc < DC; ++dc) o_acc[dc] = f32x4{0.f, 0.f, 0.f, 0.f};
  const u16* kbase = kc + (long)hkv * max_seq * D;
  const u16* vbase = vc + (long)hkv * max_seq * D;
  const int ntiles = (kv_end + KVT - 1) / KVT;
  for (int t = 0; t < ntiles; ++t) {
    __syncthreads();
    {
      const int nv8 = KVT * D / 8;
      for (int i = tid; i < nv8; i += 256) {
        const int key = i / (D / 8), col8 = i % (D / 8);
        const int kk = t * KVT + key;
        const int dst = key * (D / 8) + SWZ16(key, col8);
        s16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
        s16x8 v8 = z;
        if (kk < kv_end) {
          ((s16x8*)kt)[dst] = *(const s16x8*)(kbase + (long)kk * D + col8 * 8);
          v8 = *(const s16x8*)(vbase + (long)kk * D + col8 * 8);
        } else { ((s16x8*)kt)[dst] = z; }
#pragma unroll
        for (int j = 0; j < 8; ++j)
          ((u16*)vtT)[VT_OFF(col8 * 8 + j, key)] = (u16)v8[j];
      }
    }
    __syncthreads();
    f32x4 sfrag[NC];
#pragma unroll
    for (int c = 0; c < NC; ++c) {
      f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ks = 0; ks < KS; ++ks) {
        const int krow = c * 16 + l15;
        const int kcol8 = SWZ16(krow, ks * 4 + lg);
        s16x8 b_k = *(const s16x8*)(&((s16x8*)kt)[krow * (D / 8) + kcol8]);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_q[ks], b_k, acc,
                                                      0, 0, 0);
      }
      sfrag[c] = acc;
    }
    float p_val[NC][4];
    float alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qrow_abs = qt * 64 + w * 16 + 4 * lg + r;
      float sv[NC]; float mx = -1.0f / 0.0f;
#pragma unroll
      for (int c = 0; c < NC; ++c) {
        sv[c] = sfrag[c][r] * scale;
        const int key = t * KVT + c * 16 + l15;
        if (key >= kv_end || key > qrow_abs) sv[c] = -1.0f / 0.0f;
        mx = fmaxf(mx, sv[c]);
      }
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        mx = fmaxf(mx, __shfl_xor(mx, off));
      const float m_new = fmaxf(m_row[r], mx);
      alpha[r] = __expf(m_row[r] - m_new);
      m_row[r] = m_new;
      float psum = 0.f;
#pragma unroll
      for (int c = 0; c < NC; ++c) {
        p_val[c][r] = (sv[c] == -1.0f / 0.0f) ? 0.f : __expf(sv[c] - m_new);
        psum += p_val[c][r];
      }
#pragma unroll
      for (int off = 1; off < 16; off <<= 1) psum += __shfl_xor(psum, off);
      l_row[r] = l_row[r] * alpha[r] + psum;
    }
#pragma unroll
    for (int dc = 0; dc < DC; ++dc)
#pragma unroll
      for (int r = 0; r < 4; ++r) o_acc[dc][r] *= alpha[r];
#pragma unroll
    for (int c = 0; c < NC; ++c)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        p_lds[w][4 * lg + r][c * 16 + l15] = f2bf(p_val[c][r]);
    s16x8 a_p[KA];
#pragma unroll
    for (int ka = 0; ka < KA; ++ka)
      a_p[ka] = *(const s16x8*)(&p_lds[w][l15][ka * 32 + 8 * lg]);
#pragma unroll
    for (int dc = 0; dc < DC; ++dc) {
      const int col = dc * 16 + l15;
#pragma unroll
      for (int ka = 0; ka < KA; ++ka) {
        const s16x8 b_v = *(const s16x8*)(
            &((u16*)vtT)[VT_OFF(col, ka * 32 + 8 * lg)]);
        o_acc[dc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_p[ka], b_v, o_acc[dc], 0, 0, 0);
      }
    }
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int srow = qt * 64 + w * 16 + 4 * lg + r;
    if (srow >= S) continue;
    const float inv_l = l_row[r] > 0.f ? 1.f / l_row[r] : 1.f;
    u16* orow = out + ((long)srow * Hq + hq) * D;
#pragma unroll
    for (int dc = 0; dc < DC; ++dc)
      orow[dc * 16 + l15] = f2bf(o_acc[dc][r] * inv_l);
  }
#undef VT_OFF
}

static double run_v2(const u16* q, const u16* kc, const u16* vc, u16* out,
                     int S, int Hq, int Hkv, int max_seq, long q_ts) {
  dim3 grid((S + 63) / 64, Hq, 1);
  hipEvent_t a, b;
  (void)hipEventCreate(&a);
  (void)hipEventCreate(&b);
  for (int r = 0; r < 2; ++r)
    hipLaunchKernelGGL(k_ablate_v2, grid, dim3(256), 0, 0, q, kc, vc, out,
                       S, Hq, Hkv, max_seq, 0.0883883f, q_ts);
  (void)hipEventRecord(a);
  for (int r = 0; r < 10; ++r)
    hipLaunchKernelGGL(k_ablate_v2, grid, dim3(256), 0, 0, q, kc, vc, out,
                       S, Hq, Hkv, max_seq, 0.0883883f, q_ts);
  (void)hipEventRecord(b);
  (void)hipDeviceSynchronize();
  float ms = 0;
  (void)hipEventElapsedTime(&ms, a, b);
  printf("%-10s %8.3f ms\n", "FULL_v2", ms / 10.0);
  return ms / 10.0;
}

template <int CUT>
static double run(const u16* q, const u16* kc, const u16* vc, u16* out,
                  int S, int Hq, int Hkv, int max_seq, long q_ts,
                  const char* name) {
  dim3 grid((S + 63) / 64, Hq, 1);
  hipEvent_t a, b;
  (void)hipEventCreate(&a);
  (void)hipEventCreate(&b);
  for (int r = 0; r < 2; ++r)
    hipLaunchKernelGGL(k_ablate<CUT>, grid, dim3(256), 0, 0, q, kc, vc, out,
                       S, Hq, Hkv, max_seq, 0.0883883f, q_ts);
  (void)hipEventRecord(a);
  for (int r = 0; r < 10; ++r)
    hipLaunchKernelGGL(k_ablate<CUT>, grid, dim3(256), 0, 0, q, kc, vc, out,
                       S, Hq, Hkv, max_seq, 0.0883883f, q_ts);
  (void)hipEventRecord(b);
  (void)hipDeviceSynchronize();
  float ms = 0;
  (void)hipEventElapsedTime(&ms, a, b);
  const double per = ms / 10.0;
  printf("%-10s %8.3f ms\n", name, per);
  return per;
}

int main() {
  const int S = 2048, Hq = 32, Hkv = 8, max_seq = 2304;
  const long q_ts = (long)Hq * D;
  u16 *q, *kc, *vc, *out;
  (void)hipMalloc(&q, (long)S * q_ts * 2);
  (void)hipMalloc(&kc, (long)Hkv * max_seq * D * 2);
  (void)hipMalloc(&vc, (long)Hkv * max_seq * D * 2);
  (void)hipMalloc(&out, (long)S * Hq * D * 2);
  (void)hipMemset(q, 0x2e, (long)S * q_ts * 2);     // random-ish bf16 bytes
  (void)hipMemset(kc, 0x31, (long)Hkv * max_seq * D * 2);
  (void)hipMemset(vc, 0x33, (long)Hkv * max_seq * D * 2);
  const double full = run<0>(q, kc, vc, out, S, Hq, Hkv, max_seq, q_ts, "FULL");
  run_v2(q, kc, vc, out, S, Hq, Hkv, max_seq, q_ts);
  run_v2(q, kc, vc, out, S, Hq, Hkv, max_seq, q_ts);
  const double nosm = run<1>(q, kc, vc, out, S, Hq, Hkv, max_seq, q_ts, "NOSM");
  const double nopv = run<2>(q, kc, vc, out, S, Hq, Hkv, max_seq, q_ts, "NOPV");
  const double noqk = run<3>(q, kc, vc, out, S, Hq, Hkv, max_seq, q_ts, "NOQK");
  const double nost = run<4>(q, kc, vc, out, S, Hq, Hkv, max_seq, q_ts, "NOSTAGE");
  printf("attribution: sm %.0f%%  pv %.0f%%  qk %.0f%%  stage %.0f%%\n",
         100 * (full - nosm) / full, 100 * (full - nopv) / full,
         100 * (full - noqk) / full, 100 * (full - nost) / full);
  return 0;
}
