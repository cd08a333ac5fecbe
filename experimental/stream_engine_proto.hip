// ROUND-2 PROTOTYPE — compile-only in round 1, NEVER loaded or launched.
//
// Skeleton of the §3.4 weight-streaming engine (the only persistent-kernel
// structure measured FASTER than hipGraph launches on this chip:
// MI355X_MICROARCH.md engine-vs-launches, 0.87-0.89x): per CU one LOADER
// wave streams weights into an LDS ring with LDS-DMA (global_load_lds nt)
// while three CONSUMER waves retire slots with bf16 dot products.
//
// Deliberately deadlock-free: a pure GEMV with each workgroup owning a
// DISJOINT row range — no cross-workgroup communication at all, so any
// grid size is safe. The round-2 engine adds the multi-stage layer walk
// (qkv -> attn -> o -> mlp) with 8-byte granule hand-offs on top of this
// loader/consumer core; docs/DESIGN_persistent_decode.md has the plan and
// the knob list this file follows:
//   - 1 loader + 3 consumers per 256-thread workgroup
//   - 8 x 16 KiB LDS ring (128 KiB of the CU's 160)
//   - global_load_lds_dwordx4 ... nt on the weight stream (nt-weights row;
//     NEVER nt on hand-off traffic)
//   - loader waits its own glds with an explicit counted s_waitcnt (hipcc
//     does not track asm loads - guide SS5.7 item 1)
//   - every spin is bounded with a give-up that poisons the output
//     (guide SS5.6: bound every spin; poison before first call)
//
// Compile check (no GPU needed):  python experimental/build.py
// which also reports VGPR/SGPR/LDS via llvm-objdump so the occupancy
// budget (<=128 VGPR: 4 waves/SIMD headroom) is validated statically.

#include <hip/hip_runtime.h>

typedef unsigned short u16;
typedef short s16x8 __attribute__((ext_vector_type(8)));
typedef float f32x2 __attribute__((ext_vector_type(2)));

namespace {

__device__ __forceinline__ float bf2f(u16 h) {
  union { unsigned u; float f; } c;
  c.u = (unsigned)h << 16;
  return c.f;
}

__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v += __shfl_down(v, off, 64);
  return v;
}

// dword2 of packed bf16 pairs -> f32 dot via fdot2 (the consumer retire
// instruction the ldsdma-fill row prices: one wave 0.50 us/slot alone)
__device__ __forceinline__ float dot8_bf16(s16x8 a, s16x8 b) {
  float acc = 0.f;
#pragma unroll
  for (int j = 0; j < 8; ++j)
    acc = fmaf(bf2f((u16)a[j]), bf2f((u16)b[j]), acc);
  return acc;
}

constexpr int SLOTS = 8;            // ring depth (prefetch-credit saturates ~8)
constexpr int SLOT_BYTES = 16384;   // 16 KiB = 2 rows of K=4096 bf16
constexpr int ROWS_PER_SLOT = 2;
constexpr int FILLS_PER_SLOT = SLOT_BYTES / (64 * 16);  // dwordx4 x 64 lanes
constexpr int SPIN_LIMIT = 1 << 20; // bounded spin: give up, poison, exit

constexpr int FLAG_FREE = 0;
constexpr int FLAG_READY = 1;

// One glds fill: 64 lanes x 16 B into LDS at a wave-uniform byte address.
// M0 carries the LDS destination base; save/restore it in one statement
// (guide SS5.7 recipe; M0 is compiler-reserved).
__device__ __forceinline__ void glds16_nt(const void* gsrc, unsigned lds_dst) {
  unsigned keep;
  asm volatile(
      "s_mov_b32 %0, m0\n\t"
      "s_mov_b32 m0, %2\n\t"
      "s_nop 0\n\t"
      "global_load_lds_dwordx4 %1, off nt\n\t"
      "s_mov_b32 m0, %0"
      : "=&s"(keep)
      : "v"(gsrc), "s"(lds_dst)
      : "memory");
}

__device__ __forceinline__ void wait_glds_all() {
  // hipcc does not count asm loads: drain them explicitly (item 1).
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
}

// y[N] = W[N,K] @ x[K], bf16 in / f32 out. K == 4096 for the prototype.
// Launch geometry: 256 threads (4 waves); each workgroup owns rows
// [wg*rows_per_wg, ...). Safe at ANY grid size (no inter-WG traffic).
// 131 KiB of LDS pins exactly ONE workgroup per CU — that IS the engine
// geometry (1 loader + 3 consumer waves per CU, one per SIMD)
__global__ void __launch_bounds__(256, 1)
k_stream_gemv_proto(float* __restrict__ y, const u16* __restrict__ x,
                    const u16* __restrict__ w, int N, int K,
                    int rows_per_wg, int* __restrict__ fail) {
  __shared__ u16 ring[SLOTS * SLOT_BYTES / 2];
  __shared__ int flags[SLOTS];

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int row0 = blockIdx.x * rows_per_wg;
  if (row0 >= N) return;                    // whole WG exits together
  const int row_end = min(row0 + rows_per_wg, N);
  const int n_slots_total = (row_end - row0 + ROWS_PER_SLOT - 1) / ROWS_PER_SLOT;

  if (threadIdx.x < SLOTS) flags[threadIdx.x] = FLAG_FREE;
  __syncthreads();

  if (wave == 3) {
    // ---------------- loader wave ----------------
    for (int s = 0; s < n_slots_total; ++s) {
      const int slot = s % SLOTS;
      // wait for the slot to be consumed (bounded spin)
      int spins = 0;
      while (__builtin_amdgcn_readfirstlane(
                 ((volatile int*)flags)[slot]) != FLAG_FREE) {
        __builtin_amdgcn_s_sleep(1);
        if (++spins > SPIN_LIMIT) { if (lane == 0) *fail = 1; return; }
      }
      const long row = (long)row0 + (long)s * ROWS_PER_SLOT;
      const u16* src_base = w + row * K;
      // derive M0 from the RING'S OWN address: (a) the array's address
      // escapes into the asm, so the compiler cannot eliminate the LDS
      // allocation it writes through; (b) the base stays correct if the
      // ring is not at LDS offset 0. readfirstlane makes it provably
      // wave-uniform for the "s" constraint (guide SS5.7).
      const unsigned ring_base = __builtin_amdgcn_readfirstlane(
          (unsigned)(unsigned long)(ring + slot * (SLOT_BYTES / 2)));
      for (int j = 0; j < FILLS_PER_SLOT; ++j) {
        const char* gsrc = (const char*)src_base + (long)j * 1024 + lane * 16;
        glds16_nt(gsrc, ring_base + j * 1024);
      }
      wait_glds_all();                       // counted drain, loader-only
      if (lane == 0) ((volatile int*)flags)[slot] = FLAG_READY;
      // round-2: thin to vmcnt(16) + run-ahead instead of per-slot drain
      // (gather-pass row) once hand-off gathers share the CU
    }
  } else {
    // ---------------- consumer waves (0..2) ----------------
    for (int s = wave; s < n_slots_total; s += 3) {
      const int slot = s % SLOTS;
      int spins = 0;
      while (__builtin_amdgcn_readfirstlane(
                 ((volatile int*)flags)[slot]) != FLAG_READY) {
        __builtin_amdgcn_s_sleep(1);
        if (++spins > SPIN_LIMIT) { if (lane == 0) *fail = 2; return; }
      }
      const long row = (long)row0 + (long)s * ROWS_PER_SLOT;
      const s16x8* slot_base = (const s16x8*)(ring + slot * SLOT_BYTES / 2);
      const s16x8* xs = (const s16x8*)x;    // L1/L2-hot, ordinary loads
      const int nv = K >> 3;
#pragma unroll 1
      for (int r = 0; r < ROWS_PER_SLOT; ++r) {
        if (row + r >= N) break;
        float acc = 0.f;
        const s16x8* wrow = slot_base + (long)r * (K >> 3);
        for (int i = lane; i < nv; i += 64)
          acc += dot8_bf16(xs[i], wrow[i]);
        const float v = wave_reduce_sum(acc);
        if (lane == 0) y[row + r] = v;
      }
      // drain this wave's own LDS reads before releasing the slot
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      if (lane == 0) ((volatile int*)flags)[slot] = FLAG_FREE;
    }
  }
}

}  // namespace

extern "C" {

// Host wrapper with the mandatory residency guard: refuses (returns -1)
// when the occupancy API cannot co-resident the requested grid — the
// caller must fall back to the launch path. Round 1 never calls this.
int fei_stream_gemv_proto(float* y, const void* x, const void* w,
                          int N, int K, int* fail_dev,
                          hipStream_t stream) {
  if (K != 4096) return -2;
  int max_blocks_per_cu = 0;
  // dynamic-LDS arg is 0: the ring + flags are STATIC shared memory and
  // already in the kernel's group_segment_fixed_size (passing them again
  // double-counts and the guard refuses everything)
  hipError_t e = hipOccupancyMaxActiveBlocksPerMultiprocessor(
      &max_blocks_per_cu, (const void*)k_stream_gemv_proto, 256, 0);
  if (e != hipSuccess || max_blocks_per_cu < 1) return -1;
  hipDeviceProp_t prop;
  if (hipGetDeviceProperties(&prop, 0) != hipSuccess) return -1;
  const int grid = prop.multiProcessorCount;    // 1 WG per CU
  const int rows_per_wg = ((N + grid - 1) / grid + ROWS_PER_SLOT - 1)
                          / ROWS_PER_SLOT * ROWS_PER_SLOT;
  hipLaunchKernelGGL(k_stream_gemv_proto, dim3(grid), dim3(256), 0, stream,
                     y, (const u16*)x, (const u16*)w, N, K, rows_per_wg,
                     fail_dev);
  return 0;
}

}  // extern "C"
