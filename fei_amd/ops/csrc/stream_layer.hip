// Persistent weight-streaming decode layer (gfx950) — the §3.4 LDS-DMA
// engine structure (MI355X_MICROARCH.md rows: engine-vs-launches 0.87-0.89x,
// ldsdma-fill, prefetch-credit, gather-pass, nt-weights).
//
// ONE launch computes a whole Llama decode layer at batch 1:
//   S1 qkv = rmsnorm(h)*wa @ Wqkv^T          (norm prologue fused)
//   S2 split-K decode attention (in-kernel RoPE + KV append)
//   S2b per-head split combine
//   S3 h2 = h + att @ Wo^T                   (residual epilogue)
//   S4 act = swiglu(rmsnorm(h2)*wm @ Wgu^T)
//   S5 out = h2 + act @ Wdown^T
// replacing six kernel launches; the per-CU loader wave streams the next
// stage's weights (global_load_lds ... nt) ahead across every stage edge
// while consumers gather hand-offs, which is where the engine's win lives.
//
// Geometry: grid = 256 workgroups (one per CU; >80 KiB dynamic LDS pins
// residency), 256 threads = 4 waves: wave 3 = loader, waves 0-2 consumers.
// Wave 0 is the lead consumer: it gathers each stage's input from the
// granule buffers into LDS, preps the norm, runs this WG's attention
// split, then joins slot processing.
//
// Cross-WG hand-offs use the Guideline-16 R2 granule protocol: one
// naturally-aligned 8-byte {tag, payload} written by ONE relaxed
// agent-scope (sc1) store, swept by relaxed agent loads — no fences, no
// flags, placement-independent. tag = pos*1024 + layer*8 + stage, so tags
// are unique across the step's 32 layer launches and across decode steps
// (pos is device state, monotone within a decode run); the engine zeroes
// the workspace whenever pos can move backwards (prefill).
//
// Every spin is bounded: on timeout the WG stamps the fail word and
// RETURNS (no hang; the host checks the fail word at its next sync).
//
// Numerics mirror the launch-path kernels bit-for-bit where the operand
// order matches (same bf16 rounding points: gemv.hip k_gemv_norm /
// k_gemv_res, fei_kernels.hip k_attn_decode/_combine); reduction orders
// differ, so tests compare against the fp32 torch reference with the same
// tolerances as the launch path (tests/test_stream_gpu.py).
#include "fei_common.h"

namespace {

typedef unsigned long long u64g;
typedef __attribute__((address_space(1))) unsigned long long gu64;

constexpr int NWG = 256;          // one workgroup per CU
constexpr int SLOT_BYTES = 16384; // ring slot capacity (16 KiB)
constexpr int RING_SLOTS = 7;     // 7 x 16 KiB = 112 KiB of the 160
constexpr int SPIN_LIMIT = 1 << 21;
// ring flag protocol: 0 = never used; s+1 = logical slot s READY;
// -(s+1) = logical slot s consumed (physical slot free for s+RING_SLOTS).
// The logical id in the flag is what makes reuse unambiguous: a stale
// READY from an earlier occupant of the same physical slot can never
// satisfy a later logical slot's wait.

// dynamic-LDS carve offsets (bytes; all multiples of 16 — Guideline 17)
constexpr int LDS_RING = 0;
constexpr int LDS_X = RING_SLOTS * SLOT_BYTES;            // 8 KiB x-region
constexpr int LDS_ACT = LDS_X + 8192;                     // 28 KiB act
constexpr int LDS_CTRL = LDS_ACT + 28672;                 // control words
constexpr int LDS_TOTAL = LDS_CTRL + 256;

// control-word indices within the int32 control block
constexpr int C_RINGF = 0;                 // [RING_SLOTS] ring flags
constexpr int C_STAGE = C_RINGF + RING_SLOTS;  // stage-input-ready counter
constexpr int C_DONE1 = C_STAGE + 1;       // consumers done with S1 slots
constexpr int C_DONE3 = C_DONE1 + 1;
constexpr int C_DONE4 = C_DONE3 + 1;
constexpr int C_GATH3 = C_DONE4 + 1;       // 3-wave parallel-gather counters
constexpr int C_GATH4 = C_GATH3 + 1;
constexpr int C_GATH5 = C_GATH4 + 1;
constexpr int C_S2Q = C_GATH5 + 1;         // attention q/kn/vn staged
constexpr int C_S2C = C_S2Q + 1;           // attention wave partials done

#define wave_sum wave_reduce_sum
#define wave_max wave_reduce_max

typedef short s16x2 __attribute__((ext_vector_type(2)));

// packed-bf16 dot via v_dot2_f32_bf16 (2 MACs/instruction; the consumer
// retire instruction the ldsdma-fill row prices at 0.5 us/slot/wave) —
// ~6x fewer VALU ops than the fmaf+cvt chain
__device__ __forceinline__ float dot8(s16x8 a, s16x8 b) {
  float acc = 0.f;
  const s16x2* ap = (const s16x2*)&a;
  const s16x2* bp = (const s16x2*)&b;
#pragma unroll
  for (int j = 0; j < 4; ++j)
    acc = __builtin_amdgcn_fdot2_f32_bf16(ap[j], bp[j], acc, false);
  return acc;
}

__device__ __forceinline__ u32 f2u(float f) {
  union { float f; u32 i; } cv; cv.f = f; return cv.i;
}
__device__ __forceinline__ float u2f(u32 u) {
  union { float f; u32 i; } cv; cv.i = u; return cv.f;
}

// one 1-KiB LDS-DMA piece: 64 lanes x 16 B, nt (streamed-once weights;
// guide §5.7 recipe — M0 saved/restored in-statement, ring address passed
// through the asm so the allocation cannot be eliminated)
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

// R2 granule ops (relaxed agent scope = sc1, untorn 8-byte)
__device__ __forceinline__ void put_granule(u64g* g, unsigned tag,
                                            unsigned payload) {
  __hip_atomic_store((gu64*)g, ((u64g)tag << 32) | payload,
                     __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
}

__device__ __forceinline__ u64g get_granule(const u64g* g) {
  return __hip_atomic_load((const gu64*)g, __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_AGENT);
}

__device__ __forceinline__ unsigned pack_bf16(u16 lo, u16 hi) {
  return (unsigned)lo | ((unsigned)hi << 16);
}

struct StreamArgs {
  // activations / weights (bf16 unless noted)
  const u16* x_in;        // [C] layer input residual h
  u16* h_out;             // [C] layer output residual
  const u16* wqkv;        // [Nqkv, C]
  const u16* wo;          // [C, HqD]
  const u16* wgu;         // [2I, C]
  const u16* wdown;       // [C, I]
  const u16* norm_attn;   // [C]
  const u16* norm_mlp;    // [C]
  u16* k_cache;           // [Hkv, max_seq, D]
  u16* v_cache;
  const float* cos_sin;   // [max_seq, D/2, 2]
  const int* pos;         // [1]
  // granule workspace (u64 each)
  u64g* g_qkv;            // [Nqkv] f32 payload
  u64g* g_att;            // [Hq*D/2] 2xbf16 payload
  u64g* g_h2;             // [C/2] 2xbf16 payload
  u64g* g_act;            // [I] f32 payload
  u64g* g_done;           // [6, NWG] per-stage producer-done granules
  u64g* dbg;              // nullable: [NWG, 16] s_memtime phase stamps
  int* fail;
  // shape
  int C, Hq, Hkv, D, I, max_seq, layer;
  float eps, scale;
};

constexpr int NSPLIT = 32;

__device__ __forceinline__ void stamp(const StreamArgs& a, int wg, int slot,
                                      int lane) {
  if (a.dbg && lane == 0)
    a.dbg[wg * 16 + slot] = __builtin_amdgcn_s_memtime();
}

// one wave polls n done-granules until every tag matches. The done layer
// is what keeps the big data sweeps single-pass: polling re-reads 8 bytes
// per producer instead of the whole payload (multi-pass payload sweeps
// measured 2.4x the launch path in r2c3 — L2 poll traffic starved the
// weight stream). s_sleep(16) bounds the poll rate (polling-cost row).
__device__ __forceinline__ bool sweep_done(const u64g* base, int n,
                                           unsigned tag, int* fail,
                                           int code, int lane) {
  int spins = 0;
  for (;;) {
    bool ok = true;
    for (int i = lane; i < n; i += 64)
      ok &= (unsigned)(get_granule(&base[i]) >> 32) == tag;
    if (__all(ok)) return true;
    __builtin_amdgcn_s_sleep(16);
    if (++spins > (SPIN_LIMIT >> 8)) {
      if (lane == 0) atomicCAS(fail, 0, code);
      return false;
    }
  }
}

// bounded LDS-flag spin; returns false on timeout (fail stamped)
__device__ __forceinline__ bool lds_wait_eq(volatile int* w, int want,
                                            int* fail, int code) {
  int spins = 0;
  while (__builtin_amdgcn_readfirstlane(*w) != want) {
    __builtin_amdgcn_s_sleep(1);
    if (++spins > SPIN_LIMIT) {
      if ((threadIdx.x & 63) == 0) atomicCAS(fail, 0, code);
      return false;
    }
  }
  return true;
}

__device__ __forceinline__ bool lds_wait_ge(volatile int* w, int want,
                                            int* fail, int code) {
  int spins = 0;
  while (__builtin_amdgcn_readfirstlane(*w) < want) {
    __builtin_amdgcn_s_sleep(1);
    if (++spins > SPIN_LIMIT) {
      if ((threadIdx.x & 63) == 0) atomicCAS(fail, 0, code);
      return false;
    }
  }
  return true;
}

// ---------------------------------------------------------------------------
// LOADER (wave 3): streams this WG's weight slots in stage order through the
// LDS ring with nt LDS-DMA; one slot of run-ahead beyond the issuing slot
// (counted s_waitcnt vmcnt gate — the 'thinned' loader of the gather-pass
// row), ring-credit run-ahead across stage edges (prefetch-credit).
// Slot map for one WG (8B shapes; rows are the WG's contiguous blocks):
//   S1: 12 slots x 2 qkv rows (8 KiB row)
//   S3:  8 slots x 2 o rows
//   S4: 56 slots x {gate row i, up row i}
//   S5: 2*DR slots = half down-rows (K*2/2 bytes each)
// ---------------------------------------------------------------------------
__device__ void loader_wave(const StreamArgs& a, char* lds, volatile int* ctrl,
                            int wg, int lane) {
  const int C = a.C, I = a.I, D = a.D;
  const int Nqkv = (a.Hq + 2 * a.Hkv) * D;
  const int r1 = Nqkv / NWG;                  // qkv rows per WG (24)
  const int r3 = C / NWG;                     // o rows per WG (16)
  const int r4 = I / NWG;                     // act rows per WG (56)
  const int r5 = C / NWG;                     // down rows per WG (16)
  const int s1 = r1 / 2, s3 = r3 / 2, s4 = r4, s5 = 2 * r5;
  const int n_slots = s1 + s3 + s4 + s5;
  const long rowb = (long)C * 2;              // 8 KiB row bytes
  const long halfb = (long)I;                 // half down-row bytes (I*2/2)

  // three-deep publish pipeline: slots s, s-1, s-2 in flight; publish
  // s-3 once outstanding <= their piece sum. Depth-1 gating made the
  // cadence the LANDING latency (~1.2 us/slot measured) instead of the
  // issue rate (ldsdma-fill prices 0.64 us with overlap).
  int pend_slot[3] = {-1, -1, -1};
  int pend_log[3] = {-1, -1, -1};
  int pp1 = 0, pp2 = 0;                       // pieces of s-1, s-2
  for (int s = 0; s < n_slots; ++s) {
    const int slot = s % RING_SLOTS;
    if (s >= RING_SLOTS) {
      // previous occupant (logical s - RING_SLOTS) must be consumed
      if (!lds_wait_eq(&ctrl[C_RINGF + slot], -(s - RING_SLOTS + 1),
                       a.fail, 10)) return;
    }
    const unsigned ring_base = __builtin_amdgcn_readfirstlane(
        (unsigned)(unsigned long)(lds + LDS_RING + slot * SLOT_BYTES));
    int pieces;
    if (s < s1) {                             // S1: two qkv rows
      const long row = (long)wg * r1 + (long)(s) * 2;
      const char* src = (const char*)a.wqkv + row * rowb;
      pieces = (int)(2 * rowb) / 1024;
      for (int j = 0; j < pieces; ++j)
        glds16_nt(src + (long)j * 1024 + lane * 16, ring_base + j * 1024);
    } else if (s < s1 + s3) {                 // S3: two o rows
      const long row = (long)wg * r3 + (long)(s - s1) * 2;
      const char* src = (const char*)a.wo + row * rowb;
      pieces = (int)(2 * rowb) / 1024;
      for (int j = 0; j < pieces; ++j)
        glds16_nt(src + (long)j * 1024 + lane * 16, ring_base + j * 1024);
    } else if (s < s1 + s3 + s4) {            // S4: gate row + up row
      const long n = (long)wg * r4 + (s - s1 - s3);
      const char* gsrc = (const char*)a.wgu + n * rowb;
      const char* usrc = (const char*)a.wgu + ((long)I + n) * rowb;
      const int half = (int)rowb / 1024;      // 8 pieces each
      for (int j = 0; j < half; ++j)
        glds16_nt(gsrc + (long)j * 1024 + lane * 16, ring_base + j * 1024);
      for (int j = 0; j < half; ++j)
        glds16_nt(usrc + (long)j * 1024 + lane * 16,
                  ring_base + (half + j) * 1024);
      pieces = 2 * half;
    } else {                                  // S5: half a down row
      const int hs = s - s1 - s3 - s4;
      const long row = (long)wg * r5 + hs / 2;
      const char* src = (const char*)a.wdown + row * (halfb * 2)
                        + (hs & 1) * halfb;
      pieces = (int)halfb / 1024;             // 14 for I=14336
      for (int j = 0; j < pieces; ++j)
        glds16_nt(src + (long)j * 1024 + lane * 16, ring_base + j * 1024);
    }
    // counted gate (hipcc does not track asm loads — §5.7): allow the
    // newest THREE slots' pieces in flight; slot s-3 has landed, publish.
    if (pend_slot[0] >= 0) {
      const int allowed = pieces + pp1 + pp2;
      if (allowed >= 48)
        asm volatile("s_waitcnt vmcnt(48)" ::: "memory");
      else if (allowed >= 46)
        asm volatile("s_waitcnt vmcnt(46)" ::: "memory");
      else if (allowed >= 44)
        asm volatile("s_waitcnt vmcnt(44)" ::: "memory");
      else if (allowed >= 42)
        asm volatile("s_waitcnt vmcnt(42)" ::: "memory");
      else
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      if (lane == 0) ctrl[C_RINGF + pend_slot[0]] = pend_log[0] + 1;
    }
    pend_slot[0] = pend_slot[1]; pend_log[0] = pend_log[1];
    pend_slot[1] = pend_slot[2]; pend_log[1] = pend_log[2];
    pend_slot[2] = slot; pend_log[2] = s;
    pp2 = pp1; pp1 = pieces;
    // loader stage-issue stamps (s1/s3/s4/s5 ends -> dbg 10..13)
    if (a.dbg && lane == 0) {
      const int s1e = r1 / 2, s3e = s1e + r3 / 2, s4e = s3e + r4;
      if (s == s1e - 1) a.dbg[wg * 16 + 10] = __builtin_amdgcn_s_memtime();
      else if (s == s3e - 1) a.dbg[wg * 16 + 11] = __builtin_amdgcn_s_memtime();
      else if (s == s4e - 1) a.dbg[wg * 16 + 12] = __builtin_amdgcn_s_memtime();
      else if (s == n_slots - 1) a.dbg[wg * 16 + 13] = __builtin_amdgcn_s_memtime();
    }
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  if (lane == 0) {
#pragma unroll
    for (int q = 0; q < 3; ++q)
      if (pend_slot[q] >= 0) ctrl[C_RINGF + pend_slot[q]] = pend_log[q] + 1;
  }
}

// ---------------------------------------------------------------------------
// consumer slot processing helpers
// ---------------------------------------------------------------------------

// dot one weight row (LDS, base byte offset) against x (LDS bf16[C])
__device__ __forceinline__ float lds_row_dot(const char* lds, int row_off,
                                             const char* xoff, int C,
                                             int lane) {
  const s16x8* w = (const s16x8*)(lds + row_off);
  const s16x8* x = (const s16x8*)xoff;
  float acc = 0.f;
  const int nv = C >> 3;
  for (int i = lane; i < nv; i += 64) acc += dot8(w[i], x[i]);
  return wave_sum(acc);
}

// ---------------------------------------------------------------------------
// the kernel
// ---------------------------------------------------------------------------
template <int G>
__global__ void __launch_bounds__(256, 1)
k_stream_layer(StreamArgs a) {
  extern __shared__ __attribute__((aligned(16))) char lds[];
  volatile int* ctrl = (volatile int*)(lds + LDS_CTRL);
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wg = blockIdx.x;
  const int C = a.C, D = a.D, I = a.I, Hq = a.Hq, Hkv = a.Hkv;
  const int Nqkv = (Hq + 2 * Hkv) * D;
  const int r1 = Nqkv / NWG, r3 = C / NWG, r4 = I / NWG, r5 = C / NWG;
  const int s1 = r1 / 2, s3 = r3 / 2, s4 = r4, s5 = 2 * r5;
  const int pos_now = a.pos[0];
  const unsigned tagbase = (unsigned)pos_now * 1024u + (unsigned)a.layer * 8u;

  if (tid < 64) {                       // zero control words (one wave)
    for (int i = tid; i < 64; i += 64) ctrl[i] = 0;
  }
  __syncthreads();                      // before ANY glds is issued

  if (wave == 3) {
    stamp(a, wg, 14, lane);
    loader_wave(a, lds, ctrl, wg, lane);
    stamp(a, wg, 15, lane);
    return;
  }
  if (wave == 0) stamp(a, wg, 0, lane);

  // ------------------------------------------------------------------ S1 --
  // norm prologue from the layer input tensor, all three consumer waves
  // in parallel (plain loads: written by the previous launch, boundary-
  // synchronised). Sumsq partials combine through LDS.
  u16* xl = (u16*)(lds + LDS_X);        // stage-input LDS region (bf16[C])
  {
    float* ssred = (float*)(ctrl + 36);
    const s16x8* xr = (const s16x8*)a.x_in;
    const s16x8* wn = (const s16x8*)a.norm_attn;
    const int nv = C >> 3;
    const int per = (nv + 2) / 3;
    const int v_lo = wave * per, v_hi = min(v_lo + per, nv);
    float ss = 0.f;
    for (int i = v_lo + lane; i < v_hi; i += 64) {
      s16x8 v = xr[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float f = bf2f((u16)v[j]);
        ss = fmaf(f, f, ss);
      }
    }
    ss = wave_sum(ss);
    if (lane == 0) ssred[wave] = ss;
    __threadfence_block();
    if (lane == 0) atomicAdd((int*)&ctrl[C_STAGE], 1);
    if (!lds_wait_ge(&ctrl[C_STAGE], 3, a.fail, 20)) return;
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    const float inv = rsqrtf((ssred[0] + ssred[1] + ssred[2]) / (float)C
                             + a.eps);
    for (int i = v_lo + lane; i < v_hi; i += 64) {
      s16x8 v = xr[i], w = wn[i], o;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o[j] = (short)f2bf(bf2f((u16)v[j]) * inv * bf2f((u16)w[j]));
      ((s16x8*)xl)[i] = o;
    }
    __threadfence_block();
    if (lane == 0) atomicAdd((int*)&ctrl[C_STAGE], 1);
    if (!lds_wait_ge(&ctrl[C_STAGE], 6, a.fail, 20)) return;
    if (wave == 0) stamp(a, wg, 1, lane);
  }

  // S1 slots: 2 qkv rows each, slot s -> wave s%3; publish one granule
  // (f32 payload per output value, bf16-rounded like the launch GEMV)
  for (int s = wave; s < s1; s += 3) {
    const int slot = s % RING_SLOTS;
    if (!lds_wait_eq(&ctrl[C_RINGF + slot], s + 1, a.fail, 21)) return;
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    const float v0 = lds_row_dot(lds, LDS_RING + slot * SLOT_BYTES,
                                 (const char*)xl, C, lane);
    const float v1 = lds_row_dot(lds, LDS_RING + slot * SLOT_BYTES + C * 2,
                                 (const char*)xl, C, lane);
    if (lane == 0) {
      const long row = (long)wg * r1 + (long)s * 2;
      put_granule(&a.g_qkv[row], tagbase + 1, f2u(bf2f(f2bf(v0))));
      put_granule(&a.g_qkv[row + 1], tagbase + 1, f2u(bf2f(f2bf(v1))));
      ctrl[C_RINGF + slot] = -(s + 1);
    }
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");   // granule stores out
  if (lane == 0) {
    const int old = atomicAdd((int*)&ctrl[C_DONE1], 1);
    if (old == 2)        // last consumer wave: this WG's S1 outputs visible
      put_granule(&a.g_done[1 * NWG + wg], tagbase + 1, 0);
  }

  if (wave == 0) stamp(a, wg, 2, lane);          // S1 slots done

  // ------------------------------------------------------------------ S2 --
  // Attention v2: ONE workgroup per kv-head (wg < Hkv); its three
  // consumer waves split the keys, combine through LDS, and wave 0
  // publishes the head-group's att directly. v1 (32 splits x 8 kvh over
  // all 256 WGs + a cross-WG combine) spent ~46 us/layer in its three
  // global dependency hops (r2c6 probe); the WG-local form has ONE
  // producer hop and an 8-entry done poll.
  if (wg < Hkv) {
    const int kvh = wg;
    const int n = pos_now + 1;
    const int half = D / 2;
    const int p_new = n - 1;
    // scratch: qs [G][D] f32 + kn/vn bf16[D] in the x-region (dead after
    // S1: C_DONE1 == 3); per-wave partials in the ACT region (free until
    // S5): po3 [3][G][D] f32, ml3 [3][G][2] f32.
    float* qs = (float*)(lds + LDS_X);
    u16* kn = (u16*)(qs + G * D);
    u16* vn = kn + D;
    float* po3 = (float*)(lds + LDS_ACT);
    float* ml3 = po3 + 3 * G * D;
    const unsigned qtag = tagbase + 1;

    if (wave == 0) {
      if (!lds_wait_ge(&ctrl[C_DONE1], 3, a.fail, 22)) return;
      // poll only the WGs producing THIS kv-head's qkv slice
      {
        const int q_lo = (kvh * G * D) / r1;
        const int q_hi = (kvh * G * D + G * D - 1) / r1;
        const int k_lo = (Hq * D + kvh * D) / r1;
        const int k_hi = ((Hq + Hkv) * D + kvh * D + D - 1) / r1;
        if (!sweep_done(&a.g_done[1 * NWG + q_lo], q_hi - q_lo + 1,
                        tagbase + 1, a.fail, 40, lane)) return;
        if (!sweep_done(&a.g_done[1 * NWG + k_lo], k_hi - k_lo + 1,
                        tagbase + 1, a.fail, 40, lane)) return;
      }
      // gather + rope + scale q (sc1 sweeps, small and already-ready)
      {
        int spins = 0;
        for (;;) {
          bool ok = true;
          for (int i = lane; i < G * D; i += 64) {
            const int g = i / D, d = i % D;
            const u64g x = get_granule(&a.g_qkv[(long)(kvh * G + g) * D + d]);
            ok &= (unsigned)(x >> 32) == qtag;
            qs[g * D + d] = u2f((unsigned)x);
          }
          if (__all(ok)) break;
          __builtin_amdgcn_s_sleep(8);
          if (++spins > (SPIN_LIMIT >> 4)) {
            if (lane == 0) atomicCAS(a.fail, 0, 23);
            return;
          }
        }
        for (int i = lane; i < G * half; i += 64) {
          const int g = i / half, d = i % half;
          const float c = a.cos_sin[((long)p_new * half + d) * 2 + 0];
          const float sn = a.cos_sin[((long)p_new * half + d) * 2 + 1];
          const float x1 = qs[g * D + d], x2 = qs[g * D + d + half];
          qs[g * D + d] = (x1 * c - x2 * sn) * a.scale;
          qs[g * D + d + half] = (x2 * c + x1 * sn) * a.scale;
        }
      }
      // gather + rope the new token's k,v; append to the caches (readers
      // in THIS WG use the LDS copy; other WGs never touch key p_new)
      {
        int spins = 0;
        for (;;) {
          bool ok = true;
          for (int i = lane; i < 2 * D; i += 64) {
            const long idx = (i < D)
                ? (long)Hq * D + (long)kvh * D + i
                : (long)(Hq + Hkv) * D + (long)kvh * D + (i - D);
            const u64g x = get_granule(&a.g_qkv[idx]);
            ok &= (unsigned)(x >> 32) == qtag;
            const u16 b = f2bf(u2f((unsigned)x));
            if (i < D) kn[i] = b; else vn[i - D] = b;
          }
          if (__all(ok)) break;
          __builtin_amdgcn_s_sleep(8);
          if (++spins > (SPIN_LIMIT >> 4)) {
            if (lane == 0) atomicCAS(a.fail, 0, 24);
            return;
          }
        }
        for (int d = lane; d < half; d += 64) {
          const float c = a.cos_sin[((long)p_new * half + d) * 2 + 0];
          const float sn = a.cos_sin[((long)p_new * half + d) * 2 + 1];
          const float x1 = bf2f(kn[d]), x2 = bf2f(kn[d + half]);
          const u16 k0 = f2bf(x1 * c - x2 * sn);
          const u16 k1 = f2bf(x2 * c + x1 * sn);
          kn[d] = k0;
          kn[d + half] = k1;
          u16* kcp = a.k_cache + ((long)kvh * a.max_seq + p_new) * D;
          u16* vcp = a.v_cache + ((long)kvh * a.max_seq + p_new) * D;
          kcp[d] = k0; kcp[d + half] = k1;
          vcp[d] = vn[d]; vcp[d + half] = vn[d + half];
        }
      }
      __threadfence_block();
      if (lane == 0) atomicAdd((int*)&ctrl[C_S2Q], 1);
    } else {
      if (!lds_wait_ge(&ctrl[C_S2Q], 1, a.fail, 25)) return;
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    }

    // per-wave key range; online softmax (lane covers dims 2l, 2l+1)
    const int per_k = (n + 2) / 3;
    const int start = wave * per_k;
    const int end = min(start + per_k, n);
    const u16* kbase = a.k_cache + (long)kvh * a.max_seq * D;
    const u16* vbase = a.v_cache + (long)kvh * a.max_seq * D;
    float* pl = (float*)(ml3 + 3 * G * 2);         // [G][64] tile P scratch
    pl += wave * G * 64;                           // per-wave slice
    float m[G], l[G], o0[G], o1[G];
#pragma unroll
    for (int g = 0; g < G; ++g) {
      m[g] = -1.0f / 0.0f; l[g] = 0.f; o0[g] = 0.f; o1[g] = 0.f;
    }
    for (int tile = start; tile < end; tile += 64) {
      const int kk = tile + lane;
      float sc[G];
#pragma unroll
      for (int g = 0; g < G; ++g) sc[g] = -1.0f / 0.0f;
      if (kk < end) {
        // p_new's K row comes from the LDS copy; all other rows from the
        // cache (pointer select only — no flow divergence, loop unrolls)
        const s16x8* krow = (kk == p_new)
            ? (const s16x8*)kn : (const s16x8*)(kbase + (long)kk * D);
#pragma unroll
        for (int g = 0; g < G; ++g) sc[g] = 0.f;
#pragma unroll 4
        for (int i = 0; i < D / 8; ++i) {
          const s16x8 kv8 = krow[i];
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            const float kf = bf2f((u16)kv8[j]);
#pragma unroll
            for (int g = 0; g < G; ++g)
              sc[g] = fmaf(qs[g * D + i * 8 + j], kf, sc[g]);
          }
        }
      }
#pragma unroll
      for (int g = 0; g < G; ++g) {
        const float tile_m = wave_max(sc[g]);
        const float m_new = fmaxf(m[g], tile_m);
        const float alpha = __expf(m[g] - m_new);
        const float pv = (kk < end) ? __expf(sc[g] - m_new) : 0.f;
        pl[g * 64 + lane] = pv;
        m[g] = m_new;
        l[g] = l[g] * alpha + wave_sum(pv);
        o0[g] *= alpha; o1[g] *= alpha;
      }
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      // P*V. p_new is always the LAST key, so the cache-row loop below is
      // branch-free and unrolls — a data-dependent branch per key
      // serialised it into dependent L2 round trips (~39 us of S2 chain,
      // r2c8 probe; same lesson as the launch kernel's break-free loop).
      const int kmax = min(64, end - tile);
      const int kmax_c = (tile + kmax == n) ? kmax - 1 : kmax;  // cache keys
#pragma unroll 4
      for (int kl = 0; kl < kmax_c; ++kl) {
        const u16* vrow = vbase + (long)(tile + kl) * D + lane * 2;
        const float v0 = bf2f(vrow[0]);
        const float v1 = bf2f(vrow[1]);
#pragma unroll
        for (int g = 0; g < G; ++g) {
          const float pv = pl[g * 64 + kl];
          o0[g] = fmaf(pv, v0, o0[g]);
          o1[g] = fmaf(pv, v1, o1[g]);
        }
      }
      if (kmax_c < kmax) {                       // the new token's V (LDS)
        const float v0 = bf2f(vn[lane * 2]);
        const float v1 = bf2f(vn[lane * 2 + 1]);
#pragma unroll
        for (int g = 0; g < G; ++g) {
          const float pv = pl[g * 64 + kmax_c];
          o0[g] = fmaf(pv, v0, o0[g]);
          o1[g] = fmaf(pv, v1, o1[g]);
        }
      }
    }
    // stage this wave's partials in LDS; wave 0 combines
#pragma unroll
    for (int g = 0; g < G; ++g) {
      float* po = po3 + ((long)wave * G + g) * D;
      po[2 * lane] = o0[g];
      po[2 * lane + 1] = o1[g];
      if (lane == 0) {
        ml3[(wave * G + g) * 2] = (start < end) ? m[g] : -1.0f / 0.0f;
        ml3[(wave * G + g) * 2 + 1] = (start < end) ? l[g] : 0.f;
      }
    }
    __threadfence_block();
    if (lane == 0) atomicAdd((int*)&ctrl[C_S2C], 1);
    if (wave == 0) {
      if (!lds_wait_ge(&ctrl[C_S2C], 3, a.fail, 26)) return;
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
#pragma unroll
      for (int g = 0; g < G; ++g) {
        float mm = -1.0f / 0.0f;
#pragma unroll
        for (int wv = 0; wv < 3; ++wv)
          mm = fmaxf(mm, ml3[(wv * G + g) * 2]);
        float acc0 = 0.f, acc1 = 0.f, lt = 0.f;
#pragma unroll
        for (int wv = 0; wv < 3; ++wv) {
          const float w = __expf(ml3[(wv * G + g) * 2] - mm);
          lt = fmaf(w, ml3[(wv * G + g) * 2 + 1], lt);
          const float* po = po3 + ((long)wv * G + g) * D;
          acc0 = fmaf(w, po[2 * lane], acc0);
          acc1 = fmaf(w, po[2 * lane + 1], acc1);
        }
        const float inv_l = lt > 0.f ? 1.f / lt : 0.f;
        const int hq = kvh * G + g;
        put_granule(&a.g_att[(long)hq * (D / 2) + lane], tagbase + 3,
                    pack_bf16(f2bf(acc0 * inv_l), f2bf(acc1 * inv_l)));
      }
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      if (lane == 0) put_granule(&a.g_done[3 * NWG + kvh], tagbase + 3, 0);
      stamp(a, wg, 3, lane);                     // att published
    }
  }

  // ------------------------------------------------------------------ S3 --
  // att gather: done-poll (wave 0) -> ONE agent acquire (drops this CU's
  // stale L1 lines of the granule buffers) -> flag -> ALL THREE consumer
  // waves plain-load their third 16 B/lane wide (Guideline 16 R1 consumer
  // form; the producers stored sc1/write-through). A one-wave 8-byte sc1
  // sweep measured 87 us on the act edge (r2c5 probe).
  if (wave == 0) {
    if (!sweep_done(&a.g_done[3 * NWG], Hkv, tagbase + 3, a.fail, 42, lane))
      return;
    if (lane == 0)
      __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
    __threadfence_block();
    if (lane == 0) ctrl[C_STAGE] = 7;
  } else {
    if (!lds_wait_ge(&ctrl[C_STAGE], 7, a.fail, 28)) return;
  }
  {
    // granule PAIRS: 16 B = 2x{tag, 2xbf16}; each wave a contiguous third
    const int n_pair = Hq * D / 4;                   // 1024 pairs
    const int per = (n_pair + 2) / 3;
    const int p_lo = wave * per, p_hi = min(p_lo + per, n_pair);
    const ulonglong2* src2 = (const ulonglong2*)a.g_att;
    for (int i = p_lo + lane; i < p_hi; i += 64) {
      const ulonglong2 v = src2[i];
      ((unsigned*)xl)[2 * i] = (unsigned)v.x;        // 2 bf16
      ((unsigned*)xl)[2 * i + 1] = (unsigned)v.y;
    }
    __threadfence_block();
    if (lane == 0) atomicAdd((int*)&ctrl[C_GATH3], 1);
    if (!lds_wait_ge(&ctrl[C_GATH3], 3, a.fail, 27)) return;
    if (wave == 0) stamp(a, wg, 4, lane);            // att gathered
  }

  // S3 slots: 2 o rows; publish h2 granules (2 bf16) with residual add
  for (int s = wave; s < s3; s += 3) {
    const int slot = (s1 + s) % RING_SLOTS;
    if (!lds_wait_eq(&ctrl[C_RINGF + slot], s1 + s + 1, a.fail, 29)) return;
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    const float v0 = lds_row_dot(lds, LDS_RING + slot * SLOT_BYTES,
                                 (const char*)xl, Hq * D, lane);
    const float v1 = lds_row_dot(lds, LDS_RING + slot * SLOT_BYTES
                                 + Hq * D * 2, (const char*)xl, Hq * D, lane);
    if (lane == 0) {
      const long row = (long)wg * r3 + (long)s * 2;
      const u16 h0 = f2bf(bf2f(a.x_in[row]) + v0);
      const u16 h1 = f2bf(bf2f(a.x_in[row + 1]) + v1);
      put_granule(&a.g_h2[row / 2], tagbase + 4, pack_bf16(h0, h1));
      ctrl[C_RINGF + slot] = -(s1 + s + 1);
    }
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  if (lane == 0) {
    const int old = atomicAdd((int*)&ctrl[C_DONE3], 1);
    if (old == 2)
      put_granule(&a.g_done[4 * NWG + wg], tagbase + 4, 0);
  }

  if (wave == 0) stamp(a, wg, 5, lane);          // S3 slots done

  // ------------------------------------------------------------------ S4 --
  // wave 0: gather h2, norm-prologue into the x-region (xl reused — wait
  // for all consumers to leave S3)
  if (wave == 0) {
    if (!lds_wait_ge(&ctrl[C_DONE3], 3, a.fail, 30)) return;
    if (!sweep_done(&a.g_done[4 * NWG], NWG, tagbase + 4, a.fail, 43, lane))
      return;
    if (lane == 0)
      __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
    __threadfence_block();
    if (lane == 0) ctrl[C_STAGE] = 8;
  } else {
    if (!lds_wait_ge(&ctrl[C_STAGE], 8, a.fail, 32)) return;
  }
  {
    // parallel gather of raw h2 + per-wave sumsq partials, then every
    // wave normalizes its third (bf16-rounded like k_gemv_norm)
    float* ssred = (float*)(ctrl + 32);              // 3 f32 partials
    const int n_pair = C / 4;                        // 1024 pairs
    const int per = (n_pair + 2) / 3;
    const int p_lo = wave * per, p_hi = min(p_lo + per, n_pair);
    const ulonglong2* src2 = (const ulonglong2*)a.g_h2;
    float ss = 0.f;
    for (int i = p_lo + lane; i < p_hi; i += 64) {
      const ulonglong2 v = src2[i];
      const unsigned pa = (unsigned)v.x, pb = (unsigned)v.y;
      ((unsigned*)xl)[2 * i] = pa;
      ((unsigned*)xl)[2 * i + 1] = pb;
      const float f0 = bf2f((u16)(pa & 0xffff)), f1 = bf2f((u16)(pa >> 16));
      const float f2 = bf2f((u16)(pb & 0xffff)), f3 = bf2f((u16)(pb >> 16));
      ss = fmaf(f0, f0, fmaf(f1, f1, fmaf(f2, f2, fmaf(f3, f3, ss))));
    }
    ss = wave_sum(ss);
    if (lane == 0) ssred[wave] = ss;
    __threadfence_block();
    if (lane == 0) atomicAdd((int*)&ctrl[C_GATH4], 1);
    if (!lds_wait_ge(&ctrl[C_GATH4], 3, a.fail, 31)) return;
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    const float inv = rsqrtf((ssred[0] + ssred[1] + ssred[2]) / (float)C
                             + a.eps);
    const s16x8* wn = (const s16x8*)a.norm_mlp;
    const int v_lo = p_lo / 2, v_hi = p_hi / 2;      // vec8 range (C>>3)
    for (int i = v_lo + lane; i < v_hi; i += 64) {
      s16x8 v = ((s16x8*)xl)[i], w = wn[i], o;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o[j] = (short)f2bf(bf2f((u16)v[j]) * inv * bf2f((u16)w[j]));
      ((s16x8*)xl)[i] = o;
    }
    __threadfence_block();
    if (lane == 0) atomicAdd((int*)&ctrl[C_GATH4], 1);
    if (!lds_wait_ge(&ctrl[C_GATH4], 6, a.fail, 31)) return;
    if (wave == 0) stamp(a, wg, 6, lane);            // x2 ready
  }

  // S4 slots: {gate row, up row}; publish act granule (f32 payload)
  for (int s = wave; s < s4; s += 3) {
    const int slot = (s1 + s3 + s) % RING_SLOTS;
    if (!lds_wait_eq(&ctrl[C_RINGF + slot], s1 + s3 + s + 1, a.fail, 33))
      return;
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    const float g = lds_row_dot(lds, LDS_RING + slot * SLOT_BYTES,
                                (const char*)xl, C, lane);
    const float u = lds_row_dot(lds, LDS_RING + slot * SLOT_BYTES + C * 2,
                                (const char*)xl, C, lane);
    if (lane == 0) {
      const long n = (long)wg * r4 + s;
      const float silu = g / (1.f + __expf(-g));
      const float act = bf2f(f2bf(silu * u));        // launch-path rounding
      put_granule(&a.g_act[n], tagbase + 5, f2u(act));
      ctrl[C_RINGF + slot] = -(s1 + s3 + s + 1);
    }
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  if (lane == 0) {
    const int old = atomicAdd((int*)&ctrl[C_DONE4], 1);
    if (old == 2)
      put_granule(&a.g_done[5 * NWG + wg], tagbase + 5, 0);
  }

  if (wave == 0) stamp(a, wg, 7, lane);          // S4 slots done

  // ------------------------------------------------------------------ S5 --
  // wave 0 gathers act (bf16) into the ACT region
  u16* actl = (u16*)(lds + LDS_ACT);
  if (wave == 0) {
    if (!lds_wait_ge(&ctrl[C_DONE4], 3, a.fail, 34)) return;
    if (!sweep_done(&a.g_done[5 * NWG], NWG, tagbase + 5, a.fail, 44, lane))
      return;
    if (lane == 0)
      __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
    __threadfence_block();
    if (lane == 0) ctrl[C_STAGE] = 9;
  } else {
    if (!lds_wait_ge(&ctrl[C_STAGE], 9, a.fail, 36)) return;
  }
  {
    const int n_pair = I / 2;                        // f32-payload pairs
    const int per = (n_pair + 2) / 3;
    const int p_lo = wave * per, p_hi = min(p_lo + per, n_pair);
    const ulonglong2* src2 = (const ulonglong2*)a.g_act;
    for (int i = p_lo + lane; i < p_hi; i += 64) {
      const ulonglong2 v = src2[i];
      actl[2 * i] = f2bf(u2f((unsigned)v.x));
      actl[2 * i + 1] = f2bf(u2f((unsigned)v.y));
    }
    __threadfence_block();
    if (lane == 0) atomicAdd((int*)&ctrl[C_GATH5], 1);
    if (!lds_wait_ge(&ctrl[C_GATH5], 3, a.fail, 35)) return;
    if (wave == 0) stamp(a, wg, 8, lane);            // act gathered
  }

  // S5 slots: half down-rows; row r = slots (2r, 2r+1) -> wave r%3; the
  // wave accumulates across the pair and writes the layer output row.
  for (int r = wave; r < r5; r += 3) {
    float acc = 0.f;
    for (int hs = 0; hs < 2; ++hs) {
      const int s = 2 * r + hs;
      const int slot = (s1 + s3 + s4 + s) % RING_SLOTS;
      if (!lds_wait_eq(&ctrl[C_RINGF + slot], s1 + s3 + s4 + s + 1,
                       a.fail, 37)) return;
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      const s16x8* w = (const s16x8*)(lds + LDS_RING + slot * SLOT_BYTES);
      const s16x8* x = (const s16x8*)(actl + (long)hs * (I / 2));
      float part = 0.f;
      const int nv = (I / 2) >> 3;
      for (int i = lane; i < nv; i += 64) part += dot8(w[i], x[i]);
      acc += wave_sum(part);
      if (lane == 0) ctrl[C_RINGF + slot] = -(s1 + s3 + s4 + s + 1);
    }
    if (lane == 0 && r + 3 >= r5) stamp(a, wg, 9, lane);  // last S5 row
    if (lane == 0) {
      const long row = (long)wg * r5 + r;
      // residual: re-read own h2 granule (value published this launch)
      const u64g x = get_granule(&a.g_h2[row / 2]);
      const unsigned p = (unsigned)x;
      const float h2 = bf2f((u16)((row & 1) ? (p >> 16) : (p & 0xffff)));
      a.h_out[row] = f2bf(h2 + acc);
    }
  }
}

}  // namespace

extern "C" {

// residency + shape guard; returns 0 on OK (launchable), negative on
// refusal (caller falls back to the launch path)
int fei_stream_layer_check(int C, int Hq, int Hkv, int D, int I) {
  const int Nqkv = (Hq + 2 * Hkv) * D;
  if (D != 128 || Hq / Hkv > 8 || Hq % Hkv) return -2;
  if (Nqkv % (2 * NWG) || C % (2 * NWG) || I % NWG || (I / 2) % 1024)
    return -3;
  if (C * 2 > 8192 || I * 2 > 28672) return -4;   // LDS x/act regions
  if (I > SLOT_BYTES) return -7;                  // half down-row per slot
  if (Hq > 64 || NSPLIT != 32) return -5;
  int max_blocks = 0;
  hipError_t e = hipOccupancyMaxActiveBlocksPerMultiprocessor(
      &max_blocks, (const void*)k_stream_layer<4>, 256, LDS_TOTAL);
  if (e != hipSuccess || max_blocks < 1) return -1;
  hipDeviceProp_t prop;
  if (hipGetDeviceProperties(&prop, 0) != hipSuccess) return -1;
  if (prop.multiProcessorCount < NWG) return -6;
  return 0;
}

void fei_stream_layer(const void* x_in, void* h_out, const void* wqkv,
                      const void* wo, const void* wgu, const void* wdown,
                      const void* norm_attn, const void* norm_mlp,
                      void* k_cache, void* v_cache, const float* cos_sin,
                      const int* pos, void* g_qkv, void* g_att,
                      void* g_h2, void* g_act, void* g_done, void* dbg,
                      int* fail,
                      int C, int Hq, int Hkv, int D, int I, int max_seq,
                      int layer, float eps, float scale,
                      hipStream_t stream) {
  StreamArgs a;
  a.x_in = (const u16*)x_in; a.h_out = (u16*)h_out;
  a.wqkv = (const u16*)wqkv; a.wo = (const u16*)wo;
  a.wgu = (const u16*)wgu; a.wdown = (const u16*)wdown;
  a.norm_attn = (const u16*)norm_attn; a.norm_mlp = (const u16*)norm_mlp;
  a.k_cache = (u16*)k_cache; a.v_cache = (u16*)v_cache;
  a.cos_sin = cos_sin; a.pos = pos;
  a.g_qkv = (u64g*)g_qkv; a.g_att = (u64g*)g_att;
  a.g_h2 = (u64g*)g_h2; a.g_act = (u64g*)g_act;
  a.g_done = (u64g*)g_done; a.dbg = (u64g*)dbg; a.fail = fail;
  a.C = C; a.Hq = Hq; a.Hkv = Hkv; a.D = D; a.I = I; a.max_seq = max_seq;
  a.layer = layer; a.eps = eps; a.scale = scale;
  const int G = Hq / Hkv;
#define LSL(GV) hipLaunchKernelGGL((k_stream_layer<GV>), dim3(NWG), \
    dim3(256), LDS_TOTAL, stream, a)
  switch (G) {
    case 1: LSL(1); break;
    case 2: LSL(2); break;
    case 4: LSL(4); break;
    case 8: LSL(8); break;
    default: break;   // fei_stream_layer_check validates
  }
#undef LSL
}

}  // extern "C"
