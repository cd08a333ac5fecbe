// FP8 (OCP e4m3fn) weight-quantized decode GEMVs for gfx950.
// Weights stored as u8 with one f32 scale per OUTPUT ROW (absmax/448);
// activations and outputs stay bf16, dots accumulate f32. Halves the
// per-token weight traffic (the decode roofline) at ~0.5 % relative
// weight error. gfx950's FP8 is OCP e4m3fn (NOT MI300X fnuz) — the LDS
// lookup table below is the e4m3fn decode.
//
// Only the fused-chain ops the tp=1 decode path uses get fp8 variants:
//   k_gemv_norm_fp8        (QKV, lm_head: norm prologue)
//   k_gemv_res_fp8         (O, down: residual epilogue)
//   k_gemv_swiglu_norm_fp8 (gate/up + SwiGLU + norm prologue)
// Dequant via the HARDWARE converter: v_cvt_pk_f32_fp8 (2 elems/instr)
// into packed bf16 + v_dot2 f32 accumulation — measured 277.9 tok/s vs
// 272.8 for a 256-entry LDS lookup table (docs/BENCHMARKS.md; the LUT
// attempt lives in git history). These kernels are VALU-issue-bound, not
// HBM-bound like their bf16 siblings.
//
// Quantization itself: k_quant_fp8_rows (one block per row: absmax ->
// scale -> round-to-nearest-even encode), done once at load time.
#include "fei_common.h"

namespace {

// e4m3fn decode (OCP): e=0 subnormal m/8 * 2^-6; 0x7f/0xff = NaN.
__device__ __forceinline__ float e4m3_decode(int b) {
  const int s = (b >> 7) & 1;
  const int e = (b >> 3) & 0xF;
  const int m = b & 7;
  float f;
  if (e == 0) {
    f = (float)m * 0.001953125f;            // m/8 * 2^-6
  } else if (e == 15 && m == 7) {
    f = __builtin_nanf("");
  } else {
    f = ldexpf(1.0f + (float)m * 0.125f, e - 7);
  }
  return s ? -f : f;
}

__device__ __forceinline__ void lut_init(float* lut) {
  for (int i = threadIdx.x; i < 256; i += blockDim.x)
    lut[i] = e4m3_decode(i);
  __syncthreads();
}

typedef __attribute__((ext_vector_type(4))) unsigned int u32x4;
typedef __attribute__((ext_vector_type(2))) float f32x2;
typedef __attribute__((ext_vector_type(2))) __bf16 bf16x2;

// dot of 16 fp8 weights (one 16-B load) against 16 bf16 x elems:
// hardware e4m3fn converter (v_cvt_pk_f32_fp8, 2 elems/instr) -> packed
// bf16 (lossless: e4m3's 3-bit mantissa fits bf16's 7; the compiler emits
// v_cvt_pk_bf16_f32 from the scalar casts) -> v_dot2_f32_bf16 (2 MACs +
// the accumulate per instruction, and x stays packed — no per-element
// bf16->f32 converts). ~24 VALU ops per 16 elems vs 40 for the scalar-fma
// form and ~56 for an LDS lookup table.
__device__ __forceinline__ float dot16_fp8(const float* /*unused*/,
                                           const u16* xs, long xoff,
                                           u32x4 w16) {
  float acc = 0.f;
  const bf16x2* xp = (const bf16x2*)(xs + xoff);
#pragma unroll
  for (int q = 0; q < 4; ++q) {
    const f32x2 lo = __builtin_amdgcn_cvt_pk_f32_fp8(w16[q], false);
    const f32x2 hi = __builtin_amdgcn_cvt_pk_f32_fp8(w16[q], true);
    const bf16x2 wlo = {(__bf16)lo[0], (__bf16)lo[1]};
    const bf16x2 whi = {(__bf16)hi[0], (__bf16)hi[1]};
    acc = __builtin_amdgcn_fdot2_f32_bf16(xp[q * 2], wlo, acc, false);
    acc = __builtin_amdgcn_fdot2_f32_bf16(xp[q * 2 + 1], whi, acc, false);
  }
  return acc;
}

template <int M>
__device__ __forceinline__ void norm_factors8(const u16* __restrict__ res,
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

// norm-prologue fp8 GEMV: out = (rmsnorm(res)*wn) @ W8^T * row_scale
template <int M>
__global__ void __launch_bounds__(256)
k_gemv_norm_fp8(u16* __restrict__ out, const u16* __restrict__ res,
                const u16* __restrict__ wn, const unsigned char* __restrict__ w8,
                const float* __restrict__ wscale, int N, int K, float eps) {
  extern __shared__ __attribute__((aligned(16))) u16 xs[];   // M*K bf16
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  float inv[M];
  norm_factors8<M>(res, K, eps, lane, inv);
  // stage the normed-x rows once per block (bf16-rounded, k_rmsnorm parity)
  for (int i = tid; i < M * K; i += blockDim.x) {
    const int m = i / K, kk = i % K;
    xs[i] = f2bf(bf2f(res[(long)m * K + kk]) * inv[m] * bf2f(wn[kk]));
  }
  __syncthreads();

  const int n0 = blockIdx.x * 8 + wid * 2;
  if (n0 >= N) return;
  const bool two = (n0 + 1) < N;
  const u32x4* wrow0 = (const u32x4*)(w8 + (long)n0 * K);
  const u32x4* wrow1 = (const u32x4*)(w8 + (long)(n0 + (two ? 1 : 0)) * K);
  float acc0[M], acc1[M];
#pragma unroll
  for (int m = 0; m < M; ++m) { acc0[m] = 0.f; acc1[m] = 0.f; }
  const int nv = K >> 4;                    // 16 elems per 16-B load
  for (int i = lane; i < nv; i += 64) {
    const u32x4 wv0 = __builtin_nontemporal_load(&wrow0[i]);
    const u32x4 wv1 = __builtin_nontemporal_load(&wrow1[i]);
#pragma unroll
    for (int m = 0; m < M; ++m) {
      acc0[m] += dot16_fp8(nullptr, xs, (long)m * K + i * 16, wv0);
      acc1[m] += dot16_fp8(nullptr, xs, (long)m * K + i * 16, wv1);
    }
  }
#pragma unroll
  for (int m = 0; m < M; ++m) {
    const float v0 = wave_reduce_sum(acc0[m]) * wscale[n0];
    const float v1 = wave_reduce_sum(acc1[m]) * wscale[two ? n0 + 1 : n0];
    if (lane == 0) {
      out[(long)m * N + n0] = f2bf(v0);
      if (two) out[(long)m * N + n0 + 1] = f2bf(v1);
    }
  }
}

// residual-epilogue fp8 GEMV: res += x @ W8^T * row_scale
template <int M>
__global__ void __launch_bounds__(256)
k_gemv_res_fp8(u16* __restrict__ res, const u16* __restrict__ x,
               const unsigned char* __restrict__ w8,
               const float* __restrict__ wscale, int N, int K) {
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int n0 = blockIdx.x * 8 + wid * 2;
  if (n0 >= N) return;
  const bool two = (n0 + 1) < N;
  const u32x4* wrow0 = (const u32x4*)(w8 + (long)n0 * K);
  const u32x4* wrow1 = (const u32x4*)(w8 + (long)(n0 + (two ? 1 : 0)) * K);
  float acc0[M], acc1[M];
#pragma unroll
  for (int m = 0; m < M; ++m) { acc0[m] = 0.f; acc1[m] = 0.f; }
  const int nv = K >> 4;
  for (int i = lane; i < nv; i += 64) {
    const u32x4 wv0 = __builtin_nontemporal_load(&wrow0[i]);
    const u32x4 wv1 = __builtin_nontemporal_load(&wrow1[i]);
#pragma unroll
    for (int m = 0; m < M; ++m) {
      acc0[m] += dot16_fp8(nullptr, x, (long)m * K + i * 16, wv0);
      acc1[m] += dot16_fp8(nullptr, x, (long)m * K + i * 16, wv1);
    }
  }
#pragma unroll
  for (int m = 0; m < M; ++m) {
    const float v0 = wave_reduce_sum(acc0[m]) * wscale[n0];
    const float v1 = wave_reduce_sum(acc1[m]) * wscale[two ? n0 + 1 : n0];
    if (lane == 0) {
      u16* r = res + (long)m * N + n0;
      r[0] = f2bf(bf2f(r[0]) + v0);
      if (two) r[1] = f2bf(bf2f(r[1]) + v1);
    }
  }
}

// norm-prologue + gate/up + SwiGLU, fp8 weights
template <int M>
__global__ void __launch_bounds__(256)
k_gemv_swiglu_norm_fp8(u16* __restrict__ out, const u16* __restrict__ res,
                       const u16* __restrict__ wn,
                       const unsigned char* __restrict__ w8,
                       const float* __restrict__ wscale, int N, int K,
                       float eps) {
  extern __shared__ __attribute__((aligned(16))) u16 xs[];   // M*K bf16
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  float inv[M];
  norm_factors8<M>(res, K, eps, lane, inv);
  for (int i = tid; i < M * K; i += blockDim.x) {
    const int m = i / K, kk = i % K;
    xs[i] = f2bf(bf2f(res[(long)m * K + kk]) * inv[m] * bf2f(wn[kk]));
  }
  __syncthreads();

  const int n = blockIdx.x * 4 + wid;
  if (n >= N) return;
  const u32x4* grow = (const u32x4*)(w8 + (long)n * K);
  const u32x4* urow = (const u32x4*)(w8 + (long)(n + N) * K);
  float accg[M], accu[M];
#pragma unroll
  for (int m = 0; m < M; ++m) { accg[m] = 0.f; accu[m] = 0.f; }
  const int nv = K >> 4;
  for (int i = lane; i < nv; i += 64) {
    const u32x4 gv = __builtin_nontemporal_load(&grow[i]);
    const u32x4 uv = __builtin_nontemporal_load(&urow[i]);
#pragma unroll
    for (int m = 0; m < M; ++m) {
      accg[m] += dot16_fp8(nullptr, xs, (long)m * K + i * 16, gv);
      accu[m] += dot16_fp8(nullptr, xs, (long)m * K + i * 16, uv);
    }
  }
#pragma unroll
  for (int m = 0; m < M; ++m) {
    const float g = wave_reduce_sum(accg[m]) * wscale[n];
    const float u = wave_reduce_sum(accu[m]) * wscale[n + N];
    if (lane == 0)
      out[(long)m * N + n] = f2bf(g / (1.f + __expf(-g)) * u);
  }
}

// quantize bf16 rows -> e4m3fn + per-row scale (absmax/448), RNE on the
// mantissa via float rounding: encode(w / scale).
__device__ __forceinline__ unsigned char e4m3_encode(float f) {
  if (f != f) return 0x7F;                  // NaN
  const unsigned int bits = __float_as_uint(f);
  const unsigned int s = bits >> 31;
  float a = fabsf(f);
  if (a > 448.f) a = 448.f;                 // saturate (no inf in e4m3fn)
  // round to the e4m3 grid: scale into [1,2), round mantissa to 3 bits
  if (a < 0.001953125f * 0.5f) return (unsigned char)(s << 7);  // -> 0
  int e;
  float mant = frexpf(a, &e);               // a = mant * 2^e, mant in [0.5,1)
  mant *= 2.f; e -= 1;                      // mant in [1,2)
  if (e < -6) {                             // subnormal: units of 2^-9
    const int mi = (int)rintf(a * 512.f);   // a / 2^-9, RNE
    if (mi > 7)                             // rounds up to the min normal
      return (unsigned char)((s << 7) | (1 << 3));
    return (unsigned char)((s << 7) | mi);
  }
  int mi = (int)rintf((mant - 1.f) * 8.f);  // 3-bit mantissa, RNE via rintf
  if (mi == 8) { mi = 0; e += 1; }
  if (e > 8) { e = 8; mi = 6; }             // clamp to 448 = 1.75*2^8
  return (unsigned char)((s << 7) | ((e + 7) << 3) | mi);
}

__global__ void __launch_bounds__(256)
k_quant_fp8_rows(unsigned char* __restrict__ w8, float* __restrict__ wscale,
                 const u16* __restrict__ w, int N, int K) {
  __shared__ float red[4];
  for (int n = blockIdx.x; n < N; n += gridDim.x) {
    const u16* row = w + (long)n * K;
    float amax = 0.f;
    for (int i = threadIdx.x; i < K; i += blockDim.x)
      amax = fmaxf(amax, fabsf(bf2f(row[i])));
    amax = block_reduce_max(amax, red);
    const float scale = amax > 0.f ? amax / 448.f : 1.f;
    const float inv = 1.f / scale;
    if (threadIdx.x == 0) wscale[n] = scale;
    for (int i = threadIdx.x; i < K; i += blockDim.x)
      w8[(long)n * K + i] = e4m3_encode(bf2f(row[i]) * inv);
    __syncthreads();
  }
}

}  // namespace

extern "C" {

#define DISPATCH_M8(FN) \
  switch (M) { case 1: FN(1); break; case 2: FN(2); break; \
               case 4: FN(4); break; case 8: FN(8); break; default: break; }

void fei_gemv_norm_fp8(void* out, const void* res, const void* wn,
                       const void* w8, const float* wscale, int M, int N,
                       int K, float eps, hipStream_t stream) {
  dim3 grid((N + 7) / 8);
  const size_t lds = (size_t)M * K * 2;
#define LNF(MV) hipLaunchKernelGGL(k_gemv_norm_fp8<MV>, grid, dim3(256), lds, \
    stream, (u16*)out, (const u16*)res, (const u16*)wn, \
    (const unsigned char*)w8, wscale, N, K, eps)
  DISPATCH_M8(LNF)
#undef LNF
}

void fei_gemv_res_fp8(void* res, const void* x, const void* w8,
                      const float* wscale, int M, int N, int K,
                      hipStream_t stream) {
  dim3 grid((N + 7) / 8);
#define LRF(MV) hipLaunchKernelGGL(k_gemv_res_fp8<MV>, grid, dim3(256), 0, \
    stream, (u16*)res, (const u16*)x, (const unsigned char*)w8, wscale, N, K)
  DISPATCH_M8(LRF)
#undef LRF
}

void fei_gemv_swiglu_norm_fp8(void* out, const void* res, const void* wn,
                              const void* w8, const float* wscale, int M,
                              int N, int K, float eps, hipStream_t stream) {
  dim3 grid((N + 3) / 4);
  const size_t lds = (size_t)M * K * 2;
#define LSF(MV) hipLaunchKernelGGL(k_gemv_swiglu_norm_fp8<MV>, grid, \
    dim3(256), lds, stream, (u16*)out, (const u16*)res, (const u16*)wn, \
    (const unsigned char*)w8, wscale, N, K, eps)
  DISPATCH_M8(LSF)
#undef LSF
}

void fei_quant_fp8_rows(void* w8, float* wscale, const void* w, int N, int K,
                        hipStream_t stream) {
  int grid = N < 2048 ? N : 2048;
  hipLaunchKernelGGL(k_quant_fp8_rows, dim3(grid), dim3(256), 0, stream,
                     (unsigned char*)w8, wscale, (const u16*)w, N, K);
}

#undef DISPATCH_M8

}  // extern "C"
