// Skinny-M GEMV kernels for the decode hot path (gfx950).
// out[M, N] = x[M, K] @ W[N, K]^T  — torch F.linear weight layout (W row-
// major over K). M <= 8 (decode batch), K % 512 == 0.
//
// Design: two W rows per wave, streamed with 16 B/lane coalesced loads
// (64 lanes x 16 B = 2 cachelines per instruction; 2 rows = 2 independent
// load streams for memory-level parallelism); x read straight through
// L1/L2 (8-28 KB, cache-hot — LDS staging only added startup latency);
// per-lane f32 dot then wave shfl reduce. This is pure HBM streaming —
// rocBLAS's tiled GEMM kernels reach only ~4 TB/s on M=1 shapes
// (profiles/r01), streaming reaches the flat-read ceiling.
//
// fei_gemv_swiglu fuses the MLP gate/up pair: wave computes both dots
// (rows n and n+N) and writes silu(g)*u — kills the separate swiglu
// kernel and the 2*I-wide intermediate.
#include "fei_common.h"

namespace {

template <int M, bool NT>
__global__ void __launch_bounds__(256)
k_gemv(u16* __restrict__ out, const u16* __restrict__ x,
       const u16* __restrict__ w, int N, int K) {
  // 4 waves per block, TWO W rows per wave: two independent load streams
  // per wave double the in-flight memory requests (the fused swiglu kernel
  // streams 2 rows/wave and measures ~25 % more bandwidth than 1 row/wave).
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int n0 = blockIdx.x * 8 + wid * 2;
  if (n0 >= N) return;
  const u16* xs = x;                          // L1/L2-hot, no staging
  const bool two = (n0 + 1) < N;

  const s16x8* wrow0 = (const s16x8*)(w + (long)n0 * K);
  const s16x8* wrow1 = (const s16x8*)(w + (long)(n0 + (two ? 1 : 0)) * K);
  float acc0[M], acc1[M];
#pragma unroll
  for (int m = 0; m < M; ++m) { acc0[m] = 0.f; acc1[m] = 0.f; }
  const int nv = K >> 3;                      // vec8 per row
  for (int i = lane; i < nv; i += 64) {
    s16x8 wv0, wv1;
    if (NT) {
      wv0 = __builtin_nontemporal_load(&wrow0[i]);
      wv1 = __builtin_nontemporal_load(&wrow1[i]);
    } else {
      wv0 = wrow0[i];
      wv1 = wrow1[i];
    }
#pragma unroll
    for (int m = 0; m < M; ++m) {
      s16x8 xv = ((const s16x8*)(xs + m * K))[i];
      acc0[m] += dot8_bf16(xv, wv0);
      acc1[m] += dot8_bf16(xv, wv1);
    }
  }
#pragma unroll
  for (int m = 0; m < M; ++m) {
    float v0 = wave_reduce_sum(acc0[m]);
    float v1 = wave_reduce_sum(acc1[m]);
    if (lane == 0) {
      out[(long)m * N + n0] = f2bf(v0);
      if (two) out[(long)m * N + n0 + 1] = f2bf(v1);
    }
  }
}

// gate/up + SwiGLU fused: W = [gate rows (N) ; up rows (N)] stacked, out[M,N]
template <int M, bool NT>
__global__ void __launch_bounds__(256)
k_gemv_swiglu(u16* __restrict__ out, const u16* __restrict__ x,
              const u16* __restrict__ w, int N, int K) {
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int n = blockIdx.x * 4 + wid;
  if (n >= N) return;
  const u16* xs = x;                          // L1/L2-hot, no staging

  const s16x8* grow = (const s16x8*)(w + (long)n * K);
  const s16x8* urow = (const s16x8*)(w + (long)(n + N) * K);
  float accg[M], accu[M];
#pragma unroll
  for (int m = 0; m < M; ++m) { accg[m] = 0.f; accu[m] = 0.f; }
  const int nv = K >> 3;
  for (int i = lane; i < nv; i += 64) {
    s16x8 gv = NT ? __builtin_nontemporal_load(&grow[i]) : grow[i];
    s16x8 uv = NT ? __builtin_nontemporal_load(&urow[i]) : urow[i];
#pragma unroll
    for (int m = 0; m < M; ++m) {
      s16x8 xv = ((const s16x8*)(xs + m * K))[i];
      accg[m] += dot8_bf16(xv, gv);
      accu[m] += dot8_bf16(xv, uv);
    }
  }
#pragma unroll
  for (int m = 0; m < M; ++m) {
    float g = wave_reduce_sum(accg[m]);
    float u = wave_reduce_sum(accu[m]);
    if (lane == 0) {
      const float silu = g / (1.f + __expf(-g));
      out[(long)m * N + n] = f2bf(silu * u);
    }
  }
}

}  // namespace

extern "C" {

void fei_gemv(void* out, const void* x, const void* w, int M, int N, int K,
              int nontemporal, hipStream_t stream) {
  dim3 grid((N + 7) / 8);
  const size_t lds = 0;
#define LG(MV, NTV) hipLaunchKernelGGL((k_gemv<MV, NTV>), grid, dim3(256), \
                                       lds, stream, (u16*)out, \
                                       (const u16*)x, (const u16*)w, N, K)
#define LGD(MV) do { if (nontemporal) LG(MV, true); else LG(MV, false); } while (0)
  switch (M) {
    case 1: LGD(1); break;
    case 2: LGD(2); break;
    case 4: LGD(4); break;
    case 8: LGD(8); break;
    default: break;   // wrapper validates
  }
#undef LGD
#undef LG
}

void fei_gemv_swiglu(void* out, const void* x, const void* w, int M, int N,
                     int K, int nontemporal, hipStream_t stream) {
  dim3 grid((N + 3) / 4);
  const size_t lds = 0;
#define LG(MV, NTV) hipLaunchKernelGGL((k_gemv_swiglu<MV, NTV>), grid, \
                                  dim3(256), lds, stream, (u16*)out, \
                                  (const u16*)x, (const u16*)w, N, K)
#define LGD(MV) do { if (nontemporal) LG(MV, true); else LG(MV, false); } while (0)
  switch (M) {
    case 1: LGD(1); break;
    case 2: LGD(2); break;
    case 4: LGD(4); break;
    case 8: LGD(8); break;
    default: break;
  }
#undef LGD
#undef LG
}

}  // extern "C"

// ---------------------------------------------------------------------------
// Fused-chain GEMV variants (tp=1 decode): the residual stream never leaves
// the GEMV kernels —
//   k_gemv_res:  res[m,n] += x[m,:] @ W[n,:]      (epilogue residual add)
//   k_gemv_norm: out = rmsnorm(res)*wn @ W^T      (norm prologue; every wave
//                recomputes the row's sumsq from L1/L2-hot res — cheaper
//                than a separate kernel + intermediate round trip)
//   k_gemv_swiglu_norm: norm prologue + fused gate/up + SwiGLU
// ---------------------------------------------------------------------------

namespace {

template <int M>
__device__ __forceinline__ void norm_factors(const u16* __restrict__ res,
                                             int K, float eps, int lane,
                                             float inv[M]) {
#pragma unroll
  for (int m = 0; m < M; ++m) {
    const s16x8* row = (const s16x8*)(res + (long)m * K);
    float ss = 0.f;
    const int nv = K >> 3;
    for (int i = lane; i < nv; i += 64) {
      s16x8 v = row[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float f = bf2f((u16)v[j]);
        ss = fmaf(f, f, ss);
      }
    }
    ss = wave_reduce_sum(ss);
    inv[m] = rsqrtf(ss / (float)K + eps);
  }
}

// ssq (nullable): per-row sum-of-squares of the UPDATED residual,
// accumulated across workgroups (LDS reduce + one atomicAdd per WG) so the
// NEXT norm-prologue GEMV can skip its sumsq pass over res (the prologue
// stalls every wave on an L2-latency-bound read of the whole row before any
// W bytes stream — measured ~7-13 us per fused GEMV, profiles/r01).
// Caller zeroes ssq before the step; kernels on one stream order the
// producer before the consumer. Squares are taken of the bf16-ROUNDED
// stored value so the result matches what a separate k_rmsnorm would see.
template <int M, bool NT>
__global__ void __launch_bounds__(256)
k_gemv_res(u16* __restrict__ res, const u16* __restrict__ x,
           const u16* __restrict__ w, int N, int K, float* __restrict__ ssq) {
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int n0 = blockIdx.x * 8 + wid * 2;
  const bool active = n0 < N;   // inactive waves still reach the barrier
  const bool two = (n0 + 1) < N;
  const s16x8* wrow0 = (const s16x8*)(w + (long)(active ? n0 : 0) * K);
  const s16x8* wrow1 = (const s16x8*)(w + (long)(active ? n0 + (two ? 1 : 0) : 0) * K);
  float acc0[M], acc1[M], mysq[M];
#pragma unroll
  for (int m = 0; m < M; ++m) { acc0[m] = 0.f; acc1[m] = 0.f; mysq[m] = 0.f; }
  const int nv = K >> 3;
  if (active) {
    for (int i = lane; i < nv; i += 64) {
      s16x8 wv0 = NT ? __builtin_nontemporal_load(&wrow0[i]) : wrow0[i];
      s16x8 wv1 = NT ? __builtin_nontemporal_load(&wrow1[i]) : wrow1[i];
#pragma unroll
      for (int m = 0; m < M; ++m) {
        s16x8 xv = ((const s16x8*)(x + (long)m * K))[i];
        acc0[m] += dot8_bf16(xv, wv0);
        acc1[m] += dot8_bf16(xv, wv1);
      }
    }
  }
#pragma unroll
  for (int m = 0; m < M; ++m) {
    const float v0 = wave_reduce_sum(acc0[m]);
    const float v1 = wave_reduce_sum(acc1[m]);
    if (lane == 0 && active) {
      u16* r = res + (long)m * N + n0;
      const u16 b0 = f2bf(bf2f(r[0]) + v0);
      r[0] = b0;
      const float f0 = bf2f(b0);
      mysq[m] = f0 * f0;
      if (two) {
        const u16 b1 = f2bf(bf2f(r[1]) + v1);
        r[1] = b1;
        const float f1 = bf2f(b1);
        mysq[m] += f1 * f1;
      }
    }
  }
  if (ssq) {                         // uniform across the block
    __shared__ float sred[4][M];
    if (lane == 0) {
#pragma unroll
      for (int m = 0; m < M; ++m) sred[wid][m] = mysq[m];
    }
    __syncthreads();
    if (tid == 0) {
#pragma unroll
      for (int m = 0; m < M; ++m)
        atomicAdd(&ssq[m], sred[0][m] + sred[1][m] + sred[2][m] + sred[3][m]);
    }
  }
}

template <int M, bool NT>
__global__ void __launch_bounds__(256)
k_gemv_norm(u16* __restrict__ out, const u16* __restrict__ res,
            const u16* __restrict__ wn, const u16* __restrict__ w,
            int N, int K, float eps, const float* __restrict__ ssq) {
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int n0 = blockIdx.x * 8 + wid * 2;
  if (n0 >= N) return;
  float inv[M];
  if (ssq) {     // sumsq precomputed by the producing k_gemv_res epilogue
#pragma unroll
    for (int m = 0; m < M; ++m)
      inv[m] = rsqrtf(ssq[m] / (float)K + eps);
  } else {
    norm_factors<M>(res, K, eps, lane, inv);
  }
  const bool two = (n0 + 1) < N;
  const s16x8* wrow0 = (const s16x8*)(w + (long)n0 * K);
  const s16x8* wrow1 = (const s16x8*)(w + (long)(n0 + (two ? 1 : 0)) * K);
  float acc0[M], acc1[M];
#pragma unroll
  for (int m = 0; m < M; ++m) { acc0[m] = 0.f; acc1[m] = 0.f; }
  const int nv = K >> 3;
  for (int i = lane; i < nv; i += 64) {
    s16x8 wv0 = NT ? __builtin_nontemporal_load(&wrow0[i]) : wrow0[i];
    s16x8 wv1 = NT ? __builtin_nontemporal_load(&wrow1[i]) : wrow1[i];
    s16x8 wnv = ((const s16x8*)wn)[i];
#pragma unroll
    for (int m = 0; m < M; ++m) {
      s16x8 xv = ((const s16x8*)(res + (long)m * K))[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        // match k_rmsnorm's rounding: the normed activation is quantised
        // to bf16 before the dot (reference semantics)
        const float xn = bf2f(f2bf(bf2f((u16)xv[j]) * inv[m] * bf2f((u16)wnv[j])));
        acc0[m] = fmaf(xn, bf2f((u16)wv0[j]), acc0[m]);
        acc1[m] = fmaf(xn, bf2f((u16)wv1[j]), acc1[m]);
      }
    }
  }
#pragma unroll
  for (int m = 0; m < M; ++m) {
    const float v0 = wave_reduce_sum(acc0[m]);
    const float v1 = wave_reduce_sum(acc1[m]);
    if (lane == 0) {
      out[(long)m * N + n0] = f2bf(v0);
      if (two) out[(long)m * N + n0 + 1] = f2bf(v1);
    }
  }
}

template <int M, bool NT>
__global__ void __launch_bounds__(256)
k_gemv_swiglu_norm(u16* __restrict__ out, const u16* __restrict__ res,
                   const u16* __restrict__ wn, const u16* __restrict__ w,
                   int N, int K, float eps, const float* __restrict__ ssq) {
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int n = blockIdx.x * 4 + wid;
  if (n >= N) return;
  float inv[M];
  if (ssq) {
#pragma unroll
    for (int m = 0; m < M; ++m)
      inv[m] = rsqrtf(ssq[m] / (float)K + eps);
  } else {
    norm_factors<M>(res, K, eps, lane, inv);
  }
  const s16x8* grow = (const s16x8*)(w + (long)n * K);
  const s16x8* urow = (const s16x8*)(w + (long)(n + N) * K);
  float accg[M], accu[M];
#pragma unroll
  for (int m = 0; m < M; ++m) { accg[m] = 0.f; accu[m] = 0.f; }
  const int nv = K >> 3;
  for (int i = lane; i < nv; i += 64) {
    s16x8 gv = NT ? __builtin_nontemporal_load(&grow[i]) : grow[i];
    s16x8 uv = NT ? __builtin_nontemporal_load(&urow[i]) : urow[i];
    s16x8 wnv = ((const s16x8*)wn)[i];
#pragma unroll
    for (int m = 0; m < M; ++m) {
      s16x8 xv = ((const s16x8*)(res + (long)m * K))[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float xn = bf2f(f2bf(bf2f((u16)xv[j]) * inv[m] * bf2f((u16)wnv[j])));
        accg[m] = fmaf(xn, bf2f((u16)gv[j]), accg[m]);
        accu[m] = fmaf(xn, bf2f((u16)uv[j]), accu[m]);
      }
    }
  }
#pragma unroll
  for (int m = 0; m < M; ++m) {
    const float g = wave_reduce_sum(accg[m]);
    const float u = wave_reduce_sum(accu[m]);
    if (lane == 0)
      out[(long)m * N + n] = f2bf(g / (1.f + __expf(-g)) * u);
  }
}

// Cache prefetch: stream a weight region through normal (cache-filling)
// loads so a LATER kernel's reads hit L2/L3 instead of HBM. Launched on a
// side stream during low-bandwidth phases (decode attention) to convert
// idle HBM cycles into prefill of the next GEMV's operand. The impossible
// store defeats dead-code elimination; nothing is written in practice.
__global__ void __launch_bounds__(256)
k_prefetch(const u16* __restrict__ w, long n_vec8, float* __restrict__ sink) {
  float acc = 0.f;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n_vec8;
       i += (long)gridDim.x * blockDim.x) {
    const s16x8 v = ((const s16x8*)w)[i];
    acc += (float)(short)v[0];
  }
  if (acc == 1.0f / 0.0f) sink[0] = acc;
}

}  // namespace

extern "C" {

void fei_prefetch(const void* w, long bytes, void* sink, int n_blocks,
                  hipStream_t stream) {
  const long n_vec8 = bytes / 16;
  if (n_vec8 <= 0) return;
  hipLaunchKernelGGL(k_prefetch, dim3(n_blocks), dim3(256), 0, stream,
                     (const u16*)w, n_vec8, (float*)sink);
}

#define DISPATCH_M(FN, ...) \
  switch (M) { \
    case 1: FN(1, __VA_ARGS__); break; \
    case 2: FN(2, __VA_ARGS__); break; \
    case 4: FN(4, __VA_ARGS__); break; \
    case 8: FN(8, __VA_ARGS__); break; \
    default: break; \
  }

void fei_gemv_res(void* res, const void* x, const void* w, int M, int N,
                  int K, int nontemporal, void* ssq, hipStream_t stream) {
  dim3 grid((N + 7) / 8);
#define LR(MV, NTV) hipLaunchKernelGGL((k_gemv_res<MV, NTV>), grid, \
    dim3(256), 0, stream, (u16*)res, (const u16*)x, (const u16*)w, N, K, \
    (float*)ssq)
#define LRD(MV, _ignored) do { if (nontemporal) LR(MV, true); else LR(MV, false); } while (0)
  DISPATCH_M(LRD, 0)
#undef LRD
#undef LR
}

void fei_gemv_norm(void* out, const void* res, const void* wn, const void* w,
                   int M, int N, int K, float eps, int nontemporal,
                   const void* ssq, hipStream_t stream) {
  dim3 grid((N + 7) / 8);
#define LN(MV, NTV) hipLaunchKernelGGL((k_gemv_norm<MV, NTV>), grid, \
    dim3(256), 0, stream, (u16*)out, (const u16*)res, (const u16*)wn, \
    (const u16*)w, N, K, eps, (const float*)ssq)
#define LND(MV, _ignored) do { if (nontemporal) LN(MV, true); else LN(MV, false); } while (0)
  DISPATCH_M(LND, 0)
#undef LND
#undef LN
}

void fei_gemv_swiglu_norm(void* out, const void* res, const void* wn,
                          const void* w, int M, int N, int K, float eps,
                          int nontemporal, const void* ssq,
                          hipStream_t stream) {
  dim3 grid((N + 3) / 4);
#define LS(MV, NTV) hipLaunchKernelGGL((k_gemv_swiglu_norm<MV, NTV>), grid, \
    dim3(256), 0, stream, (u16*)out, (const u16*)res, (const u16*)wn, \
    (const u16*)w, N, K, eps, (const float*)ssq)
#define LSD(MV, _ignored) do { if (nontemporal) LS(MV, true); else LS(MV, false); } while (0)
  DISPATCH_M(LSD, 0)
#undef LSD
#undef LS
}

#undef DISPATCH_M

}  // extern "C"

// ---------------------------------------------------------------------------
// MFMA M<=8 GEMM: out[M, N] = x[M, K] @ W[N, K]^T for the decode batch
// path. Four VALU-kernel variants measured slower than the pipelined
// k_gemv<8> (profiles/r02_batch_attention.md); this MFMA tile beats it
// on every llama shape (qkv 1.45x, o 1.43x, down 1.27x, stacked gate/up
// 1.65x — experimental/gemv_mfma.hip, numerics exact vs the VALU form).
// Structure: 16 n-rows per workgroup, each wave owns a K-QUARTER with
// wave-private double-buffered LDS staging (register prefetch overlaps
// the next tile's loads with the current tile's MFMAs; same-wave ds
// ordering needs no barriers), one end-of-kernel cross-wave C reduce.
// A-fragment rows clamp to M-1: the duplicate rows only produce C rows
// >= M, which are never written. Requires K % 1024 == 0, N % 16 == 0
// (the wrapper routes other shapes to k_gemv).
#define GEMM_NT 16
#define GEMM_KC 128
#define GSWZ(row, col8) ((col8) ^ ((row) & 7))

namespace {

typedef __attribute__((ext_vector_type(4))) float f32x4_g;

template <bool NTW>
__global__ void __launch_bounds__(256)
k_gemm_m8(u16* __restrict__ out, const u16* __restrict__ x,
          const u16* __restrict__ w, int M, int N, int K) {
  const int n0 = blockIdx.x * GEMM_NT;
  const int tid = threadIdx.x;
  const int wv = tid >> 6;
  const int lane = tid & 63;
  const int l15 = lane & 15, lg = lane >> 4;
  __shared__ u16 wt[4][2][GEMM_NT * GEMM_KC];
  __shared__ float cred[4][16][16];

  f32x4_g acc = {0.f, 0.f, 0.f, 0.f};
  const int kq = K / 4;
  const int nv8 = GEMM_KC / 8;
  const int k_lo = wv * kq, k_hi = (wv + 1) * kq;
  const int arow = l15 < M ? l15 : M - 1;
  const int srow = lane / 4, scol8 = (lane % 4) * 4;
  auto fetch = [&](int k0, s16x8 r[4]) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const s16x8* p = (const s16x8*)(w + (long)(n0 + srow) * K + k0 +
                                      (scol8 + j) * 8);
      r[j] = NTW ? __builtin_nontemporal_load(p) : *p;
    }
  };
  auto put = [&](int buf, s16x8 r[4]) {
#pragma unroll
    for (int j = 0; j < 4; ++j)
      ((s16x8*)wt[wv][buf])[srow * nv8 + GSWZ(srow, scol8 + j)] = r[j];
  };
  s16x8 pre[4];
  fetch(k_lo, pre);
  put(0, pre);
  int buf = 0;
  for (int k0 = k_lo; k0 < k_hi; k0 += GEMM_KC) {
    if (k0 + GEMM_KC < k_hi) fetch(k0 + GEMM_KC, pre);
#pragma unroll
    for (int kb = 0; kb < GEMM_KC / 32; ++kb) {
      const s16x8 a_frag =
          *(const s16x8*)(x + (long)arow * K + k0 + kb * 32 + lg * 8);
      const int col8 = kb * 4 + lg;
      const s16x8 b_frag =
          ((s16x8*)wt[wv][buf])[l15 * nv8 + GSWZ(l15, col8)];
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b_frag, acc,
                                                    0, 0, 0);
    }
    if (k0 + GEMM_KC < k_hi) put(buf ^ 1, pre);
    buf ^= 1;
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) cred[wv][lg * 4 + r][l15] = acc[r];
  __syncthreads();
  if (wv == 0) {
    for (int i = lane; i < M * 16; i += 64) {
      const int m = i / 16, n = i % 16;
      const float v = cred[0][m][n] + cred[1][m][n] + cred[2][m][n] +
                      cred[3][m][n];
      out[(long)m * N + n0 + n] = f2bf(v);
    }
  }
}

}  // namespace

extern "C" void fei_gemm_m8(void* out, const void* x, const void* w, int M,
                            int N, int K, int nontemporal,
                            hipStream_t stream) {
  dim3 grid(N / GEMM_NT);
  if (nontemporal)
    hipLaunchKernelGGL(k_gemm_m8<true>, grid, dim3(256), 0, stream,
                       (u16*)out, (const u16*)x, (const u16*)w, M, N, K);
  else
    hipLaunchKernelGGL(k_gemm_m8<false>, grid, dim3(256), 0, stream,
                       (u16*)out, (const u16*)x, (const u16*)w, M, N, K);
}
