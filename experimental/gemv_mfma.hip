// M=8 GEMM on MFMA — microbench for the batch-decode GEMV family
// (profiles/r02_batch_attention.md §4: at batch 8 the VALU GEMV kernels
// are 70% of the step at ~2.6-3.4 TB/s; four register/LDS/unroll
// variants of the VALU form all measured slower than the compiler's
// deep pipeline). This probes the documented alternative: a real MFMA
// tile — x [8, K] padded to 16 rows as A-fragments, W staged through
// LDS in [64 n x KC k] tiles as B-fragments, C written from the
// standard 16x16x32 map (row = (lane>>4)*4 + reg, col = lane&15).
// Fragment lane maps follow ops/csrc/attn_prefill.hip (validated by the
// 52 prefill numerics tests + guide §3).
//
// out[8, N] = x16[0:8, K] @ W[N, K]^T   (x16 rows 8..15 are zero pad)
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <cmath>

typedef unsigned short u16;
typedef unsigned int u32;
typedef __attribute__((ext_vector_type(8))) short s16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

__host__ __device__ __forceinline__ float bf2f(u16 u) {
  union { float f; u32 i; } cv; cv.i = ((u32)u) << 16; return cv.f;
}
__host__ __device__ __forceinline__ u16 f2bf(float f) {
  union { float f; u32 i; } cv; cv.f = f;
  u32 x = cv.i; u32 lsb = (x >> 16) & 1u; x += 0x7fffu + lsb;
  return (u16)(x >> 16);
}

constexpr int KC = 256;                  // k elements staged per tile
constexpr int NT = 64;                   // n rows per workgroup
#define SWZ(row, col8) ((col8) ^ ((row) & 7))

__global__ void __launch_bounds__(256)
k_gemm_m8(u16* __restrict__ out, const u16* __restrict__ x16,
          const u16* __restrict__ w, int N, int K) {
  const int n0 = blockIdx.x * NT;
  const int tid = threadIdx.x;
  const int wv = tid >> 6;               // wave -> 16-n subtile
  const int lane = tid & 63;
  const int l15 = lane & 15, lg = lane >> 4;
  __shared__ u16 wt[NT][KC];             // 32 KB, row-swizzled vec8

  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  const int nv8 = KC / 8;                // vec8 per staged row
  for (int k0 = 0; k0 < K; k0 += KC) {
    __syncthreads();
    for (int i = tid; i < NT * nv8; i += 256) {
      const int row = i / nv8, col8 = i % nv8;
      ((s16x8*)wt)[row * nv8 + SWZ(row, col8)] =
          *(const s16x8*)(w + (long)(n0 + row) * K + k0 + col8 * 8);
    }
    __syncthreads();
#pragma unroll
    for (int kb = 0; kb < KC / 32; ++kb) {
      // A: x16 row = l15 (rows 8..15 zero pad), k = k0 + kb*32 + lg*8
      const s16x8 a_frag =
          *(const s16x8*)(x16 + (long)l15 * K + k0 + kb * 32 + lg * 8);
      // B: W row (n) = wave subtile + l15, same k chunk, from LDS
      const int wrow = wv * 16 + l15;
      const int col8 = kb * 4 + lg;
      const s16x8 b_frag = ((s16x8*)wt)[wrow * nv8 + SWZ(wrow, col8)];
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b_frag, acc,
                                                    0, 0, 0);
    }
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int m = lg * 4 + r;            // C row
    if (m < 8) {
      const int n = n0 + wv * 16 + l15;  // C col
      out[(long)m * N + n] = f2bf(acc[r]);
    }
  }
}


// v2: NT=16 n per WG (grid x4 — v1's N/64 grid left 2/3 of the chip
// idle on the square shapes), waves own K-QUARTERS with wave-private
// LDS staging (no per-stage barriers; same-wave ds ordering is
// compiler-tracked), one end-of-kernel LDS reduce across the 4 partial
// C tiles. Requires K % 1024 == 0 (all llama shapes).
constexpr int NT2 = 16;
constexpr int KC2 = 256;

__global__ void __launch_bounds__(256)
k_gemm_m8v2(u16* __restrict__ out, const u16* __restrict__ x16,
            const u16* __restrict__ w, int N, int K) {
  const int n0 = blockIdx.x * NT2;
  const int tid = threadIdx.x;
  const int wv = tid >> 6;               // wave -> k quarter
  const int lane = tid & 63;
  const int l15 = lane & 15, lg = lane >> 4;
  __shared__ u16 wt[4][NT2 * KC2];       // 4 x 8 KB, wave-private
  __shared__ float cred[4][16][16];      // partial C tiles

  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  const int kq = K / 4;
  const int nv8 = KC2 / 8;
  for (int k0 = wv * kq; k0 < (wv + 1) * kq; k0 += KC2) {
    for (int i = lane; i < NT2 * nv8; i += 64) {
      const int row = i / nv8, col8 = i % nv8;
      ((s16x8*)wt[wv])[row * nv8 + SWZ(row, col8)] =
          *(const s16x8*)(w + (long)(n0 + row) * K + k0 + col8 * 8);
    }
#pragma unroll
    for (int kb = 0; kb < KC2 / 32; ++kb) {
      const s16x8 a_frag =
          *(const s16x8*)(x16 + (long)l15 * K + k0 + kb * 32 + lg * 8);
      const int col8 = kb * 4 + lg;
      const s16x8 b_frag = ((s16x8*)wt[wv])[l15 * nv8 + SWZ(l15, col8)];
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b_frag, acc,
                                                    0, 0, 0);
    }
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) cred[wv][lg * 4 + r][l15] = acc[r];
  __syncthreads();
  if (wv == 0) {
    for (int i = lane; i < 8 * 16; i += 64) {
      const int m = i / 16, n = i % 16;
      const float v = cred[0][m][n] + cred[1][m][n] + cred[2][m][n] +
                      cred[3][m][n];
      out[(long)m * N + n0 + n] = f2bf(v);
    }
  }
}


// v3: v2 + per-wave DOUBLE-BUFFERED staging with register prefetch —
// the v2 wave serializes stage(8 KB) -> lgkmcnt drain -> mfma each
// tile; here tile j+1 is loaded into registers while tile j's MFMAs
// run, then written to the other LDS half (same-wave ordering, still
// no barriers). KC3=128 keeps LDS at 4 waves x 2 x 4 KB + cred = 36 KB.
constexpr int KC3 = 128;

__global__ void __launch_bounds__(256)
k_gemm_m8v3(u16* __restrict__ out, const u16* __restrict__ x16,
            const u16* __restrict__ w, int N, int K) {
  const int n0 = blockIdx.x * NT2;
  const int tid = threadIdx.x;
  const int wv = tid >> 6;
  const int lane = tid & 63;
  const int l15 = lane & 15, lg = lane >> 4;
  __shared__ u16 wt[4][2][NT2 * KC3];
  __shared__ float cred[4][16][16];

  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  const int kq = K / 4;
  const int nv8 = KC3 / 8;               // 16
  const int k_lo = wv * kq, k_hi = (wv + 1) * kq;
  // each lane stages NT2*nv8/64 = 4 vec8 per tile
  const int srow = lane / 4, scol8 = (lane % 4) * 4;
  auto fetch = [&](int k0, s16x8 r[4]) {
#pragma unroll
    for (int j = 0; j < 4; ++j)
      r[j] = *(const s16x8*)(w + (long)(n0 + srow) * K + k0 +
                             (scol8 + j) * 8);
  };
  auto put = [&](int buf, s16x8 r[4]) {
#pragma unroll
    for (int j = 0; j < 4; ++j)
      ((s16x8*)wt[wv][buf])[srow * nv8 + SWZ(srow, scol8 + j)] = r[j];
  };
  s16x8 pre[4];
  fetch(k_lo, pre);
  put(0, pre);
  int buf = 0;
  for (int k0 = k_lo; k0 < k_hi; k0 += KC3) {
    if (k0 + KC3 < k_hi) fetch(k0 + KC3, pre);   // overlap next tile
#pragma unroll
    for (int kb = 0; kb < KC3 / 32; ++kb) {
      const s16x8 a_frag =
          *(const s16x8*)(x16 + (long)l15 * K + k0 + kb * 32 + lg * 8);
      const int col8 = kb * 4 + lg;
      const s16x8 b_frag =
          ((s16x8*)wt[wv][buf])[l15 * nv8 + SWZ(l15, col8)];
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b_frag, acc,
                                                    0, 0, 0);
    }
    if (k0 + KC3 < k_hi) put(buf ^ 1, pre);
    buf ^= 1;
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) cred[wv][lg * 4 + r][l15] = acc[r];
  __syncthreads();
  if (wv == 0) {
    for (int i = lane; i < 8 * 16; i += 64) {
      const int m = i / 16, n = i % 16;
      const float v = cred[0][m][n] + cred[1][m][n] + cred[2][m][n] +
                      cred[3][m][n];
      out[(long)m * N + n0 + n] = f2bf(v);
    }
  }
}

// current-champion VALU kernel (copy of ops/csrc/gemv.hip k_gemv<8, true>)
__device__ __forceinline__ float dot8_bf16(s16x8 a, s16x8 b) {
  float acc = 0.f;
#pragma unroll
  for (int j = 0; j < 8; ++j)
    acc += bf2f((u16)a[j]) * bf2f((u16)b[j]);
  return acc;
}
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off; off >>= 1) v += __shfl_xor(v, off);
  return v;
}

__global__ void __launch_bounds__(256)
k_gemv8(u16* __restrict__ out, const u16* __restrict__ x,
        const u16* __restrict__ w, int N, int K) {
  constexpr int M = 8;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int n0 = blockIdx.x * 8 + wid * 2;
  if (n0 >= N) return;
  const bool two = (n0 + 1) < N;
  const s16x8* wrow0 = (const s16x8*)(w + (long)n0 * K);
  const s16x8* wrow1 = (const s16x8*)(w + (long)(n0 + (two ? 1 : 0)) * K);
  float acc0[M], acc1[M];
#pragma unroll
  for (int m = 0; m < M; ++m) { acc0[m] = 0.f; acc1[m] = 0.f; }
  const int nv = K >> 3;
  for (int i = lane; i < nv; i += 64) {
    s16x8 wv0 = __builtin_nontemporal_load(&wrow0[i]);
    s16x8 wv1 = __builtin_nontemporal_load(&wrow1[i]);
#pragma unroll
    for (int m = 0; m < M; ++m) {
      s16x8 xv = ((const s16x8*)(x + m * K))[i];
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

#define CHK(x) do { hipError_t e = (x); if (e != hipSuccess) { \
  printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__); exit(1); } } while (0)

static float bench(void (*launch)(u16*, const u16*, const u16*, int, int),
                   u16* out, const u16* x, const u16* w, int N, int K) {
  for (int i = 0; i < 10; ++i) launch(out, x, w, N, K);
  hipEvent_t a, b;
  (void)hipEventCreate(&a);
  (void)hipEventCreate(&b);
  (void)hipEventRecord(a);
  const int REP = 100;
  for (int i = 0; i < REP; ++i) launch(out, x, w, N, K);
  (void)hipEventRecord(b);
  (void)hipEventSynchronize(b);
  float ms;
  (void)hipEventElapsedTime(&ms, a, b);
  return ms * 1000.f / REP;              // us
}

static void launch_mfma(u16* out, const u16* x, const u16* w, int N, int K) {
  hipLaunchKernelGGL(k_gemm_m8, dim3(N / NT), dim3(256), 0, 0, out, x, w, N, K);
}
static void launch_valu(u16* out, const u16* x, const u16* w, int N, int K) {
  hipLaunchKernelGGL(k_gemv8, dim3((N + 7) / 8), dim3(256), 0, 0, out, x, w, N, K);
}
static void launch_mfma2(u16* out, const u16* x, const u16* w, int N, int K) {
  hipLaunchKernelGGL(k_gemm_m8v2, dim3(N / NT2), dim3(256), 0, 0, out, x, w, N, K);
}
static void launch_mfma3(u16* out, const u16* x, const u16* w, int N, int K) {
  hipLaunchKernelGGL(k_gemm_m8v3, dim3(N / NT2), dim3(256), 0, 0, out, x, w, N, K);
}

int main() {
  struct Shape { const char* name; int N, K; };
  Shape shapes[] = {{"qkv 6144x4096", 6144, 4096},
                    {"o 4096x4096", 4096, 4096},
                    {"down 4096x14336", 4096, 14336},
                    {"gateup-stack 28672x4096", 28672, 4096}};
  for (auto& s : shapes) {
    const int N = s.N, K = s.K, M = 8;
    u16 *x16, *w, *o1, *o2;
    CHK(hipMalloc(&x16, 16L * K * 2));
    CHK(hipMalloc(&w, (long)N * K * 2));
    CHK(hipMalloc(&o1, (long)M * N * 2));
    CHK(hipMalloc(&o2, (long)M * N * 2));
    // host init: small asymmetric values (A=I-style check impossible at
    // M=8; asymmetric x and W catch transposed maps)
    u16* hx = (u16*)calloc(16L * K, 2);
    u16* hw = (u16*)malloc((long)N * K * 2);
    for (long i = 0; i < 8L * K; ++i)
      hx[i] = f2bf(0.01f * (float)((i * 37 + 11) % 97) - 0.45f);
    for (long i = 0; i < (long)N * K; ++i)
      hw[i] = f2bf(0.01f * (float)((i * 53 + 7) % 101) - 0.5f);
    CHK(hipMemcpy(x16, hx, 16L * K * 2, hipMemcpyHostToDevice));
    CHK(hipMemcpy(w, hw, (long)N * K * 2, hipMemcpyHostToDevice));

    launch_mfma(o1, x16, w, N, K);
    launch_valu(o2, x16, w, N, K);
    CHK(hipDeviceSynchronize());
    u16* h1 = (u16*)malloc((long)M * N * 2);
    u16* h2 = (u16*)malloc((long)M * N * 2);
    CHK(hipMemcpy(h1, o1, (long)M * N * 2, hipMemcpyDeviceToHost));
    CHK(hipMemcpy(h2, o2, (long)M * N * 2, hipMemcpyDeviceToHost));
    float maxerr = 0.f;
    for (long i = 0; i < (long)M * N; ++i) {
      float d = fabsf(bf2f(h1[i]) - bf2f(h2[i]));
      float rel = d / (1.f + fabsf(bf2f(h2[i])));
      if (rel > maxerr) maxerr = rel;
    }
    launch_mfma2(o1, x16, w, N, K);
    CHK(hipDeviceSynchronize());
    CHK(hipMemcpy(h1, o1, (long)M * N * 2, hipMemcpyDeviceToHost));
    float maxerr2 = 0.f;
    for (long i = 0; i < (long)M * N; ++i) {
      float d = fabsf(bf2f(h1[i]) - bf2f(h2[i]));
      float rel = d / (1.f + fabsf(bf2f(h2[i])));
      if (rel > maxerr2) maxerr2 = rel;
    }
    launch_mfma3(o1, x16, w, N, K);
    CHK(hipDeviceSynchronize());
    CHK(hipMemcpy(h1, o1, (long)M * N * 2, hipMemcpyDeviceToHost));
    float maxerr3 = 0.f;
    for (long i = 0; i < (long)M * N; ++i) {
      float d = fabsf(bf2f(h1[i]) - bf2f(h2[i]));
      float rel = d / (1.f + fabsf(bf2f(h2[i])));
      if (rel > maxerr3) maxerr3 = rel;
    }
    const float us_m2 = bench(launch_mfma2, o1, x16, w, N, K);
    const float us_m3 = bench(launch_mfma3, o1, x16, w, N, K);
    const float us_v = bench(launch_valu, o2, x16, w, N, K);
    const double gb = (double)N * K * 2 / 1e9;
    printf("%-24s err %.4f/%.4f/%.4f  v2 %7.2f us (%.2f TB/s)  "
           "v3 %7.2f us (%.2f TB/s)  valu %7.2f us (%.2f TB/s)  "
           "v3-speedup %.2fx\n",
           s.name, maxerr, maxerr2, maxerr3, us_m2, gb / us_m2 * 1e3,
           us_m3, gb / us_m3 * 1e3, us_v, gb / us_v * 1e3, us_v / us_m3);
    (void)hipFree(x16); (void)hipFree(w); (void)hipFree(o1); (void)hipFree(o2);
    free(hx); free(hw); free(h1); free(h2);
  }
  return 0;
}
