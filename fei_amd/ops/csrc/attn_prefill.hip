// Flash-style causal GQA prefill attention for gfx950 (MFMA
// mfma_f32_16x16x32_bf16, LDS-staged K/V tiles, online softmax).
// Semantics: fei_amd/ops/reference.py::attn_prefill — q [B,S,Hq,D] (roped),
// caches [B,Hkv,max_seq,D] already hold the roped K and V for positions
// [0, pos0+S); q token s attends cache [0, pos0+s] (causal) or [0, kv_len)
// (bidirectional, used by the bge encoder path).
//
// Structure (v1, correctness-first; guide: cdna_hip_programming.md §5/§B):
//   grid (ceil(S/64), Hq, B); block 256 = 4 waves; wave w owns q rows
//   [qt*64 + 16w, +16). K/V tiles of 32 keys staged in LDS, shared by the
//   4 waves. Per tile: QK^T = 4 MFMAs per 16-key chunk (K-dim = D in steps
//   of 32), per-row online softmax in-register (16-lane shfl groups), P
//   staged through LDS to re-shape C-layout -> A-layout, PV = D/16 MFMAs.
//
// Fragment maps used (verified on hardware by k_mfma_probe / test_ops_gpu):
//   A[16x32]: lane l -> A[l&15][8*(l>>4)+j], j=0..7
//   B[32x16]: lane l -> B[8*(l>>4)+j][l&15]
//   C[16x16]: lane l, reg r -> C[4*(l>>4)+r][l&15]
#include "fei_common.h"

namespace {

template <int D>
__global__ void __launch_bounds__(256)
k_attn_prefill(const u16* __restrict__ q, const u16* __restrict__ kc,
               const u16* __restrict__ vc, u16* __restrict__ out,
               const int* __restrict__ pos0, const int* __restrict__ kv_len,
               int B, int S, int Hq, int Hkv, int max_seq, float scale,
               int causal, long q_ts) {
  constexpr int DC = D / 16;       // d-chunks (8 for D=128)
  constexpr int KS = D / 32;       // K-dim steps per QK^T mfma chain
  constexpr int KV = 64;           // keys per LDS tile (one barrier per 64)
  constexpr int NC = KV / 16;      // 16-key score chunks per tile
  constexpr int KA = KV / 32;      // P A-fragments per tile (K=32 each)
  const int qt = blockIdx.x;       // 64-row q tile
  const int hq = blockIdx.y;
  const int b = blockIdx.z;
  const int hkv = hq / (Hq / Hkv);
  const int tid = threadIdx.x;
  const int w = tid >> 6;          // wave 0..3
  const int lane = tid & 63;
  const int l15 = lane & 15;
  const int lg = lane >> 4;        // 16-lane group 0..3

  // K tile: row-major with a T2 XOR swizzle (guide Guideline 4): read
  // column-slice-wise with ds_read_b128, swizzling the 16-byte slot index
  // by (row&7) spreads each 16-lane group over 8 slots.
  // V tile: stored TRANSPOSED [col][key] in 8-key blocks so the PV
  // B-fragment (8 consecutive keys of one column) is ONE 16-byte vector
  // read — the r02 ablation (experimental/prefill_ablate.hip) measured
  // the per-element PV gather at 56% of the kernel. The 8-key block
  // index is XOR-swizzled by (col&7) to break the 128-byte column
  // stride's bank pattern (<=2-way). The transpose cost moves to the
  // staging scatter: 8 ds_write_b16 per global vec8 — 4x fewer LDS ops
  // than the per-element PV reads they replace.
  __shared__ u16 kt[64][D];
  __shared__ u16 vtT[D][64];
  __shared__ u16 p_lds[4][16][64];
#define SWZ16(row, col8) ((col8) ^ ((row) & 7))
#define VT_OFF(col, key) \
  ((col) * 64 + \
   ((((key) >> 3) ^ ((col) & 7) ^ (((col) >> 3) & 7)) << 3) + ((key) & 7))

  const int p0 = pos0[b];
  const int q_hi = min(qt * 64 + 64, S);           // exclusive rel row bound
  const int kv_end = causal ? (p0 + q_hi) : kv_len[b];

  // ---- load Q fragments (row = l&15 within this wave's 16-row block) ----
  const int qrow_rel = qt * 64 + w * 16 + l15;     // A-fragment row
  const int qrow_ld = min(qrow_rel, S - 1);        // clamp for tail
  s16x8 a_q[KS];
  {
    const u16* qp = q + ((long)b * S + qrow_ld) * q_ts + (long)hq * D + 8 * lg;
#pragma unroll
    for (int ks = 0; ks < KS; ++ks)
      a_q[ks] = *(const s16x8*)(qp + ks * 32);
  }

  // online-softmax state per C row r (replicated across the 16-lane group)
  float m_row[4], l_row[4];
  f32x4 o_acc[DC];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m_row[r] = -1.0f / 0.0f; l_row[r] = 0.f; }
#pragma unroll
  for (int dc = 0; dc < DC; ++dc) o_acc[dc] = f32x4{0.f, 0.f, 0.f, 0.f};

  const u16* kbase = kc + ((long)b * Hkv + hkv) * max_seq * D;
  const u16* vbase = vc + ((long)b * Hkv + hkv) * max_seq * D;

  const int ntiles = (kv_end + KV - 1) / KV;
  for (int t = 0; t < ntiles; ++t) {
    // ---- stage K/V tile (KV keys x D), zero-padded past kv_end ----------
    __syncthreads();                                // vt/kt reuse protection
    {
      const int nv8 = KV * D / 8;                   // vec8 slots in a tile
      for (int i = tid; i < nv8; i += 256) {
        const int key = i / (D / 8);
        const int col8 = i % (D / 8);
        const int kk = t * KV + key;
        const int dst = key * (D / 8) + SWZ16(key, col8);
        s16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
        s16x8 v8 = z;
        if (kk < kv_end) {
          ((s16x8*)kt)[dst] = *(const s16x8*)(kbase + (long)kk * D + col8 * 8);
          v8 = *(const s16x8*)(vbase + (long)kk * D + col8 * 8);
        } else {
          ((s16x8*)kt)[dst] = z;
        }
        // V transposed scatter (VT_OFF block swizzle)
#pragma unroll
        for (int j = 0; j < 8; ++j)
          ((u16*)vtT)[VT_OFF(col8 * 8 + j, key)] = (u16)v8[j];
      }
    }
    __syncthreads();

    // ---- QK^T: NC 16-key chunks ---------------------------------------
    f32x4 sfrag[NC];
#pragma unroll
    for (int c = 0; c < NC; ++c) {
      f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ks = 0; ks < KS; ++ks) {
        const int krow = c * 16 + l15;
        const int kcol8 = SWZ16(krow, ks * 4 + lg);
        const s16x8 b_k = *(const s16x8*)(&((s16x8*)kt)[krow * (D / 8) + kcol8]);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_q[ks], b_k, acc, 0, 0, 0);
      }
      sfrag[c] = acc;
    }

    // ---- mask + online softmax ----------------------------------------
    float p_val[NC][4];
    float alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qrow_abs = p0 + qt * 64 + w * 16 + 4 * lg + r;
      float sv[NC];
      float mx = -1.0f / 0.0f;
#pragma unroll
      for (int c = 0; c < NC; ++c) {
        sv[c] = sfrag[c][r] * scale;
        const int key = t * KV + c * 16 + l15;
        if (key >= kv_end || (causal && key > qrow_abs)) sv[c] = -1.0f / 0.0f;
        mx = fmaxf(mx, sv[c]);
      }
#pragma unroll
      for (int off = 1; off < 16; off <<= 1) mx = fmaxf(mx, __shfl_xor(mx, off));
      const float m_new = fmaxf(m_row[r], mx);
      alpha[r] = __expf(m_row[r] - m_new);          // 0 on first tile
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

    // ---- rescale O, stage P (C-layout -> A-layout via LDS) -------------
#pragma unroll
    for (int dc = 0; dc < DC; ++dc)
#pragma unroll
      for (int r = 0; r < 4; ++r) o_acc[dc][r] *= alpha[r];
#pragma unroll
    for (int c = 0; c < NC; ++c)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        p_lds[w][4 * lg + r][c * 16 + l15] = f2bf(p_val[c][r]);
    // same-wave LDS RAW: compiler inserts lgkmcnt waits; no barrier needed
    // (p_lds[w] is private to wave w).

    // ---- PV: KA P-fragments of K=32 each -------------------------------
    s16x8 a_p[KA];
#pragma unroll
    for (int ka = 0; ka < KA; ++ka)
      a_p[ka] = *(const s16x8*)(&p_lds[w][l15][ka * 32 + 8 * lg]);
#pragma unroll
    for (int dc = 0; dc < DC; ++dc) {
      const int col = dc * 16 + l15;
#pragma unroll
      for (int ka = 0; ka < KA; ++ka) {
        // 8 consecutive keys of this column = one 16-byte read
        const s16x8 b_v = *(const s16x8*)(
            &((u16*)vtT)[VT_OFF(col, ka * 32 + 8 * lg)]);
        o_acc[dc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_p[ka], b_v,
                                                            o_acc[dc], 0, 0, 0);
      }
    }
  }

  // ---- epilogue: O / l, store ------------------------------------------
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int srow = qt * 64 + w * 16 + 4 * lg + r;
    if (srow >= S) continue;
    const float inv_l = l_row[r] > 0.f ? 1.f / l_row[r] : 0.f;
    u16* orow = out + (((long)b * S + srow) * Hq + hq) * D;
#pragma unroll
    for (int dc = 0; dc < DC; ++dc)
      orow[dc * 16 + l15] = f2bf(o_acc[dc][r] * inv_l);
  }
}

}  // namespace

extern "C" {

void fei_attn_prefill(const void* q, const void* k_cache, const void* v_cache,
                      void* out, const int* pos0, const int* kv_len,
                      int B, int S, int Hq, int Hkv, int D, int max_seq,
                      float scale, int causal, long q_ts, hipStream_t stream) {
  dim3 grid((S + 63) / 64, Hq, B);
  if (D == 128) {
    hipLaunchKernelGGL(k_attn_prefill<128>, grid, dim3(256), 0, stream,
                       (const u16*)q, (const u16*)k_cache, (const u16*)v_cache,
                       (u16*)out, pos0, kv_len, B, S, Hq, Hkv, max_seq, scale,
                       causal, q_ts);
  } else if (D == 64) {
    hipLaunchKernelGGL(k_attn_prefill<64>, grid, dim3(256), 0, stream,
                       (const u16*)q, (const u16*)k_cache, (const u16*)v_cache,
                       (u16*)out, pos0, kv_len, B, S, Hq, Hkv, max_seq, scale,
                       causal, q_ts);
  }
}

// ---------------------------------------------------------------------------
// MFMA layout probe: C[16x16] = A[16x32] @ B[32x16] with the fragment maps
// above, all matrices row-major bf16 in global memory. Lets the GPU test
// falsify the assumed lane mappings with asymmetric inputs (guide §3).
// ---------------------------------------------------------------------------
__global__ void k_mfma_probe(const u16* __restrict__ A,
                             const u16* __restrict__ Bm,
                             float* __restrict__ C) {
  const int lane = threadIdx.x & 63;
  const int l15 = lane & 15;
  const int lg = lane >> 4;
  s16x8 a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = (short)A[l15 * 32 + 8 * lg + j];
    b[j] = (short)Bm[(8 * lg + j) * 16 + l15];
  }
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) C[(4 * lg + r) * 16 + l15] = acc[r];
}

void fei_mfma_probe(const void* A, const void* B, float* C,
                    hipStream_t stream) {
  hipLaunchKernelGGL(k_mfma_probe, dim3(1), dim3(64), 0, stream,
                     (const u16*)A, (const u16*)B, C);
}

}  // extern "C"
