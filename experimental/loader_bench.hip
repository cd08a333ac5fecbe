// Loader-cadence microbench: ONE wave streams 16 KiB slots through an
// LDS ring with global_load_lds, under the stream engine's exact
// structure, sweeping the knobs the engine can turn:
//   depth  = slots allowed in flight before the vmcnt gate (1..4)
//   nt     = non-temporal aux on the weight stream
//   cons   = 3 consumer waves retiring slots with fdot2 dots (vs free)
// Prints GB/s per CU for each variant. 256 WGs x 256 threads, one per CU.
#include <hip/hip_runtime.h>
#include <cstdio>

typedef unsigned short u16;
typedef short s16x8 __attribute__((ext_vector_type(8)));
typedef short s16x2 __attribute__((ext_vector_type(2)));

constexpr int SLOT_BYTES = 16384;
constexpr int RING = 7;
constexpr int NSLOTS = 96;          // per WG per pass

__device__ __forceinline__ void glds16_nt(const void* gsrc, unsigned lds_dst,
                                          bool nt) {
  unsigned keep;
  if (nt)
    asm volatile("s_mov_b32 %0, m0\n\ts_mov_b32 m0, %2\n\ts_nop 0\n\t"
                 "global_load_lds_dwordx4 %1, off nt\n\ts_mov_b32 m0, %0"
                 : "=&s"(keep) : "v"(gsrc), "s"(lds_dst) : "memory");
  else
    asm volatile("s_mov_b32 %0, m0\n\ts_mov_b32 m0, %2\n\ts_nop 0\n\t"
                 "global_load_lds_dwordx4 %1, off\n\ts_mov_b32 m0, %0"
                 : "=&s"(keep) : "v"(gsrc), "s"(lds_dst) : "memory");
}

// all 16 pieces of a slot in ONE asm statement: 4 base pointers x 4
// offset immediates, s_add m0 walks the LDS ring (per-piece save/restore
// + per-statement compiler barriers measured 126 cyc/piece vs ~19 priced)
__device__ __forceinline__ void glds_slot_batched(const char* src,
                                                  unsigned lds_dst,
                                                  int lane, bool nt) {
  const char* s0 = src + lane * 16;
  const char* s1 = s0 + 4096;
  const char* s2 = s0 + 8192;
  const char* s3 = s0 + 12288;
  unsigned keep;
#define P(reg, off) \
  "global_load_lds_dwordx4 " reg ", off offset:" #off " nt\n\t" \
  "s_add_u32 m0, m0, 0x400\n\ts_nop 0\n\t"
  asm volatile(
      "s_mov_b32 %0, m0\n\t"
      "s_mov_b32 m0, %5\n\t"
      "s_nop 0\n\t"
      P("%1", 0) P("%1", 1024) P("%1", 2048) P("%1", 3072)
      P("%2", 0) P("%2", 1024) P("%2", 2048) P("%2", 3072)
      P("%3", 0) P("%3", 1024) P("%3", 2048) P("%3", 3072)
      P("%4", 0) P("%4", 1024) P("%4", 2048) P("%4", 3072)
      "s_mov_b32 m0, %0"
      : "=&s"(keep)
      : "v"(s0), "v"(s1), "v"(s2), "v"(s3), "s"(lds_dst)
      : "memory");
#undef P
  (void)nt;
}

template <int DEPTH, bool NT, bool CONS, bool BATCH = false>
__global__ void __launch_bounds__(256, 1)
k_loader_bench(const u16* __restrict__ w, float* __restrict__ sink,
               unsigned long long* __restrict__ cycles) {
  extern __shared__ __attribute__((aligned(16))) char lds[];
  volatile int* flags = (volatile int*)(lds + RING * SLOT_BYTES);
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wg = blockIdx.x;
  if (threadIdx.x < RING) flags[threadIdx.x] = 0;
  __syncthreads();
  const u16* base = w + (long)wg * NSLOTS * (SLOT_BYTES / 2);

  if (wave == 3) {
    const unsigned long long t0 = __builtin_amdgcn_s_memtime();
    int pend_slot[DEPTH];
    int pend_log[DEPTH];
#pragma unroll
    for (int q = 0; q < DEPTH; ++q) { pend_slot[q] = -1; pend_log[q] = -1; }
    for (int s = 0; s < NSLOTS; ++s) {
      const int slot = s % RING;
      if (s >= RING)
        while (flags[slot] != -(s - RING + 1)) __builtin_amdgcn_s_sleep(1);
      const unsigned rb = __builtin_amdgcn_readfirstlane(
          (unsigned)(unsigned long)(lds + slot * SLOT_BYTES));
      const char* src = (const char*)(base + (long)s * (SLOT_BYTES / 2));
      if (BATCH)
        glds_slot_batched(src, rb, lane, NT);
      else
        for (int j = 0; j < 16; ++j)
          glds16_nt(src + (long)j * 1024 + lane * 16, rb + j * 1024, NT);
      if (pend_slot[0] >= 0) {
        if (DEPTH == 1) asm volatile("s_waitcnt vmcnt(16)" ::: "memory");
        else if (DEPTH == 2) asm volatile("s_waitcnt vmcnt(32)" ::: "memory");
        else if (DEPTH == 3) asm volatile("s_waitcnt vmcnt(48)" ::: "memory");
        else asm volatile("s_waitcnt vmcnt(63)" ::: "memory");
        if (lane == 0) flags[pend_slot[0]] = pend_log[0] + 1;
      }
#pragma unroll
      for (int q = 0; q + 1 < DEPTH; ++q) {
        pend_slot[q] = pend_slot[q + 1]; pend_log[q] = pend_log[q + 1];
      }
      pend_slot[DEPTH - 1] = slot; pend_log[DEPTH - 1] = s;
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    if (lane == 0) {
#pragma unroll
      for (int q = 0; q < DEPTH; ++q)
        if (pend_slot[q] >= 0) flags[pend_slot[q]] = pend_log[q] + 1;
      cycles[wg] = __builtin_amdgcn_s_memtime() - t0;
    }
  } else {
    // consumers: wait READY, optionally dot the slot against itself,
    // mark consumed
    float acc = 0.f;
    for (int s = wave; s < NSLOTS; s += 3) {
      const int slot = s % RING;
      while (flags[slot] != s + 1) __builtin_amdgcn_s_sleep(1);
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      if (CONS) {
        const s16x8* p = (const s16x8*)(lds + slot * SLOT_BYTES);
        for (int i = lane; i < SLOT_BYTES / 16; i += 64) {
          const s16x8 v = p[i];
          const s16x2* vp = (const s16x2*)&v;
#pragma unroll
          for (int j = 0; j < 4; ++j)
            acc = __builtin_amdgcn_fdot2_f32_bf16(vp[j], vp[j], acc, false);
        }
      }
      if (lane == 0) flags[slot] = -(s + 1);
    }
    if (acc == 1.f / 0.f) sink[0] = acc;
  }
}

template <int DEPTH, bool NT, bool CONS, bool BATCH = false>
static void run_variant(const u16* w, float* sink, unsigned long long* cyc,
                        const char* name) {
  const int lds_bytes = RING * SLOT_BYTES + 256;
  for (int r = 0; r < 3; ++r)
    hipLaunchKernelGGL((k_loader_bench<DEPTH, NT, CONS, BATCH>), dim3(256),
                       dim3(256), lds_bytes, 0, w, sink, cyc);
  (void)hipDeviceSynchronize();
  unsigned long long h[256];
  (void)hipMemcpy(h, cyc, sizeof(h), hipMemcpyDeviceToHost);
  double mean = 0;
  for (int i = 0; i < 256; ++i) mean += (double)h[i];
  mean /= 256.0;
  const double us = mean / 2000.0;               // ~2 GHz shader clock
  const double gbs = (double)NSLOTS * SLOT_BYTES / (us * 1000.0);
  printf("%-28s %8.1f us  %6.2f us/slot  %6.1f GB/s/CU\n", name, us,
         us / NSLOTS, gbs);
}

int main() {
  u16* w;
  float* sink;
  unsigned long long* cyc;
  const long bytes = (long)256 * NSLOTS * SLOT_BYTES;
  if (hipMalloc(&w, bytes) != hipSuccess) return 1;
  (void)hipMemset(w, 0x3c, bytes);
  (void)hipMalloc(&sink, 4);
  (void)hipMalloc(&cyc, 256 * 8);
  run_variant<1, true, true>(w, sink, cyc, "depth1 nt cons");
  run_variant<2, true, true>(w, sink, cyc, "depth2 nt cons");
  run_variant<3, true, true>(w, sink, cyc, "depth3 nt cons");
  run_variant<4, true, true>(w, sink, cyc, "depth4(63) nt cons");
  run_variant<3, false, true>(w, sink, cyc, "depth3 DEFAULT cons");
  run_variant<3, true, false>(w, sink, cyc, "depth3 nt FREE(no dots)");
  run_variant<4, true, false>(w, sink, cyc, "depth4 nt FREE");
  // The BATCH variants are DISABLED: measured as a HANG on MI355X —
  // evidence that global_load_lds's offset: immediate does not combine
  // with an s_add-walked M0 the way a global-side offset would (the DMA
  // appears to land outside the ring and corrupt the flag words). Kept
  // compiled for the record; do not re-enable without an LDS-dest probe.
  if (0) {
    run_variant<3, true, true, true>(w, sink, cyc, "BATCH depth3 nt cons");
  }
  return 0;
}
