// Decode-attention ablation (guide §5.4 rule 8): the split-K decode
// kernel shows ~19-20 us FIXED cost per call regardless of context
// (scripts/attn_probe.py: 20.45 us at 2k ctx / 8.4 MB KV, 24.3 us at 8k
// / 33.5 MB — the marginal bytes stream at 6.3 TB/s). This probe is a
// standalone copy of ops/csrc/fei_kernels.hip k_attn_decode<4, false>
// (the engine's G=4 8B shape, rope ablated separately by attn_probe)
// with template<int CUT> phase stubs, dead values kept live with empty
// asm (rule 17):
//   CUT 0 FULL          1 NOSCORE (skip K loads + dot)
//   2 NOSM (skip reduces/softmax/pl)   3 NOV (skip V loads + PV)
//   4 NOEPI (skip LDS-osh epilogue + partial writes; timing only)
//   5 STREAM (K+V loads only, keep-alive — the streaming floor)
// plus the separate combine kernel timed standalone.
// Llama-3-8B decode shape: B1 Hq32 Hkv8 D128, splits 32, n in {512, 8190}.
#include <hip/hip_runtime.h>
#include <cstdio>

typedef unsigned short u16;
typedef unsigned int u32;
typedef __attribute__((ext_vector_type(8))) short s16x8;

__device__ __forceinline__ float bf2f(u16 u) {
  union { float f; u32 i; } cv; cv.i = ((u32)u) << 16; return cv.f;
}
template <typename T>
__device__ __forceinline__ void keep(T& v) { asm volatile("" : "+v"(v)); }

__device__ __forceinline__ float dot8(s16x8 a, s16x8 b) {
  float acc = 0.f;
#pragma unroll
  for (int j = 0; j < 8; ++j) acc += bf2f((u16)a[j]) * bf2f((u16)b[j]);
  return acc;
}

constexpr int G = 4, D = 128, TILE = 256;

template <int OP>
__device__ __forceinline__ void red_vec(float v[G], float red[G][4]) {
  const int wid = threadIdx.x >> 6;
  const int nw = blockDim.x >> 6;
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

template <int CUT>
__global__ void __launch_bounds__(256)
k_ablate(const u16* __restrict__ q, const u16* __restrict__ kc,
         const u16* __restrict__ vc, float* __restrict__ part_o,
         float* __restrict__ part_ml, const int* __restrict__ pos,
         int Hq, int Hkv, int max_seq, int splits, float scale) {
  const int split = blockIdx.x;
  const int hkv = blockIdx.y;
  const int tid = threadIdx.x;
  __shared__ float qs[G][D];
  __shared__ float pl[G][TILE];
  __shared__ float red[G][4];
  __shared__ float osh[8][D / 2][2];

  const int n = pos[0] + 1;
  const int chunk = (n + splits - 1) / splits;
  const int start = split * chunk;
  const int end = min(start + chunk, n);
  if (start >= end) {
#pragma unroll
    for (int g = 0; g < G; ++g) {
      const int hq = hkv * G + g;
      float* po = part_o + (((long)hq) * splits + split) * D;
      float* pml = part_ml + (((long)hq) * splits + split) * 2;
      for (int d = tid; d < D; d += blockDim.x) po[d] = 0.f;
      if (tid == 0) { pml[0] = -1.0f / 0.0f; pml[1] = 0.f; }
    }
    return;
  }
  for (int i = tid; i < G * D; i += blockDim.x) {
    const int g = i / D, d = i % D;
    qs[g][d] = bf2f(q[(long)(hkv * G + g) * D + d]) * scale;
  }
  __syncthreads();

  const u16* kbase = kc + (long)hkv * max_seq * D;
  const u16* vbase = vc + (long)hkv * max_seq * D;
  const int dpairs = D / 2;
  const int kgroups = blockDim.x / dpairs;
  const int keys_per_group = TILE / kgroups;
  const int dp = tid % dpairs;
  const int kg = tid / dpairs;

  float m[G], l[G], sc[G], o0[G], o1[G];
#pragma unroll
  for (int g = 0; g < G; ++g) {
    m[g] = -1.0f / 0.0f; l[g] = 0.f; o0[g] = 0.f; o1[g] = 0.f;
  }

  for (int tile = start; tile < end; tile += TILE) {
    const int kk = tile + tid;
    if (CUT != 1 && CUT != 5) {
      if (kk < end) {
        const s16x8* krow = (const s16x8*)(kbase + (long)kk * D);
#pragma unroll
        for (int g = 0; g < G; ++g) sc[g] = 0.f;
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
    } else if (CUT == 5) {
      // streaming floor: touch every K byte, keep one value live
      float acc = 0.f;
      if (kk < end) {
        const s16x8* krow = (const s16x8*)(kbase + (long)kk * D);
        for (int i = 0; i < D / 8; ++i) {
          s16x8 kv8 = krow[i];
          acc += bf2f((u16)kv8[0]);
        }
      }
      keep(acc);
#pragma unroll
      for (int g = 0; g < G; ++g) { sc[g] = acc; keep(sc[g]); }
    } else {
#pragma unroll
      for (int g = 0; g < G; ++g) {
        sc[g] = (kk < end) ? 0.3f : -1.0f / 0.0f;
        keep(sc[g]);
      }
    }
    float alpha[G], pv[G];
    if (CUT != 2 && CUT != 5) {
      float tile_m[G];
#pragma unroll
      for (int g = 0; g < G; ++g) tile_m[g] = sc[g];
      red_vec<0>(tile_m, red);
#pragma unroll
      for (int g = 0; g < G; ++g) {
        const float m_new = fmaxf(m[g], tile_m[g]);
        alpha[g] = __expf(m[g] - m_new);
        pv[g] = (kk < end) ? __expf(sc[g] - m_new) : 0.f;
        pl[g][tid] = pv[g];
        m[g] = m_new;
      }
      float tile_sum[G];
#pragma unroll
      for (int g = 0; g < G; ++g) tile_sum[g] = pv[g];
      red_vec<1>(tile_sum, red);
#pragma unroll
      for (int g = 0; g < G; ++g) l[g] = l[g] * alpha[g] + tile_sum[g];
    } else {
#pragma unroll
      for (int g = 0; g < G; ++g) {
        alpha[g] = 1.f;
        pv[g] = sc[g]; keep(pv[g]);
        pl[g][tid] = pv[g];           // keep pl populated for the V pass
        l[g] += pv[g];
      }
      __syncthreads();                // pl visible (replaces reduce barrier)
    }
    if (CUT != 3 && CUT != 5) {
#pragma unroll
      for (int g = 0; g < G; ++g) { o0[g] *= alpha[g]; o1[g] *= alpha[g]; }
      const int kbase_local = kg * keys_per_group;
      const int kmax = min(TILE, end - tile);
      const int iters = min(keys_per_group, max(0, kmax - kbase_local));
#pragma unroll 4
      for (int j = 0; j < iters; ++j) {
        const int kl = kbase_local + j;
        const u16* vrow = vbase + (long)(tile + kl) * D + dp * 2;
        const u32 vpair = *(const u32*)vrow;
        const float v0 = bf2f((u16)(vpair & 0xffff));
        const float v1 = bf2f((u16)(vpair >> 16));
#pragma unroll
        for (int g = 0; g < G; ++g) {
          const float p = pl[g][kl];
          o0[g] = fmaf(p, v0, o0[g]);
          o1[g] = fmaf(p, v1, o1[g]);
        }
      }
      __syncthreads();
    } else if (CUT == 5) {
      // streaming floor: touch every V byte too
      const int kbase_local = kg * keys_per_group;
      const int kmax = min(TILE, end - tile);
      const int iters = min(keys_per_group, max(0, kmax - kbase_local));
      float acc = 0.f;
#pragma unroll 4
      for (int j = 0; j < iters; ++j) {
        const u16* vrow = vbase + (long)(tile + kbase_local + j) * D + dp * 2;
        acc += bf2f((u16)(*(const u32*)vrow & 0xffff));
      }
      keep(acc);
#pragma unroll
      for (int g = 0; g < G; ++g) { o0[g] += acc; o1[g] += acc; }
      __syncthreads();
    } else {
#pragma unroll
      for (int g = 0; g < G; ++g) {
        o0[g] += pl[g][tid & (TILE - 1)] * 1e-7f;
        o1[g] += 1e-7f;
        keep(o0[g]); keep(o1[g]);
      }
      __syncthreads();
    }
  }

  if (CUT != 4) {
#pragma unroll
    for (int g = 0; g < G; ++g) {
      const int hq = hkv * G + g;
      float* po = part_o + (((long)hq) * splits + split) * D;
      float* pml = part_ml + (((long)hq) * splits + split) * 2;
      osh[kg][dp][0] = o0[g];
      osh[kg][dp][1] = o1[g];
      __syncthreads();
      if (kg == 0) {
        float s0 = osh[0][dp][0], s1 = osh[0][dp][1];
        for (int gg = 1; gg < kgroups; ++gg) {
          s0 += osh[gg][dp][0];
          s1 += osh[gg][dp][1];
        }
        po[dp * 2] = s0;
        po[dp * 2 + 1] = s1;
      }
      if (tid == 0) { pml[0] = m[g]; pml[1] = l[g]; }
      __syncthreads();
    }
  } else {
#pragma unroll
    for (int g = 0; g < G; ++g) { keep(o0[g]); keep(o1[g]); keep(m[g]); keep(l[g]); }
    if (tid == 0 && o0[0] == 1e30f) part_o[0] = o0[0];  // never true
  }
}

#define CHK(x) do { hipError_t e = (x); if (e != hipSuccess) { \
  printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__); return 1; } } while (0)

template <int CUT>
float run(const u16* q, const u16* kc, const u16* vc, float* po, float* pml,
          const int* pos, int Hq, int Hkv, int MS, int splits) {
  dim3 grid(splits, Hkv, 1);
  for (int i = 0; i < 20; ++i)
    hipLaunchKernelGGL(k_ablate<CUT>, grid, dim3(256), 0, 0, q, kc, vc, po,
                       pml, pos, Hq, Hkv, MS, splits, 0.0883883f);
  hipEvent_t a, b;
  (void)hipEventCreate(&a);
  (void)hipEventCreate(&b);
  (void)hipEventRecord(a);
  const int REP = 300;
  for (int i = 0; i < REP; ++i)
    hipLaunchKernelGGL(k_ablate<CUT>, grid, dim3(256), 0, 0, q, kc, vc, po,
                       pml, pos, Hq, Hkv, MS, splits, 0.0883883f);
  (void)hipEventRecord(b);
  (void)hipEventSynchronize(b);
  float ms = 0.f;
  (void)hipEventElapsedTime(&ms, a, b);
  return ms * 1000.f / REP;                       // us per call
}

int main(int argc, char** argv) {
  const int Hq = 32, Hkv = 8, MS = 8192, splits = 32;
  u16 *q, *kc, *vc;
  float *po, *pml;
  int* pos;
  CHK(hipMalloc(&q, (long)Hq * D * 2));
  CHK(hipMalloc(&kc, (long)Hkv * MS * D * 2));
  CHK(hipMalloc(&vc, (long)Hkv * MS * D * 2));
  CHK(hipMalloc(&po, (long)Hq * splits * D * 4));
  CHK(hipMalloc(&pml, (long)Hq * splits * 2 * 4));
  CHK(hipMalloc(&pos, 4));
  CHK(hipMemset(q, 0x3c, (long)Hq * D * 2));      // ~0.0059 bf16 pattern
  CHK(hipMemset(kc, 0x3c, (long)Hkv * MS * D * 2));
  CHK(hipMemset(vc, 0x3c, (long)Hkv * MS * D * 2));
  const char* names[6] = {"FULL", "NOSCORE", "NOSM", "NOV", "NOEPI",
                          "STREAM"};
  for (int n : {512, 2048, 8190}) {
    int nh = n - 1;
    CHK(hipMemcpy(pos, &nh, 4, hipMemcpyHostToDevice));
    float us[6];
    us[0] = run<0>(q, kc, vc, po, pml, pos, Hq, Hkv, MS, splits);
    us[1] = run<1>(q, kc, vc, po, pml, pos, Hq, Hkv, MS, splits);
    us[2] = run<2>(q, kc, vc, po, pml, pos, Hq, Hkv, MS, splits);
    us[3] = run<3>(q, kc, vc, po, pml, pos, Hq, Hkv, MS, splits);
    us[4] = run<4>(q, kc, vc, po, pml, pos, Hq, Hkv, MS, splits);
    us[5] = run<5>(q, kc, vc, po, pml, pos, Hq, Hkv, MS, splits);
    printf("n=%4d: ", n);
    for (int c = 0; c < 6; ++c) printf("%s=%.2fus ", names[c], us[c]);
    printf("\n");
  }
  return 0;
}
