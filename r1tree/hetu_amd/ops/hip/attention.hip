// Flash-attention 2 forward + backward, hand-written for gfx950 MFMA.
//
// Replaces the reference's vendored CUTLASS flash_attn
// (hetu/impl/kernel/FlashAttention.cu wrapping third_party/flash_attn) with
// a from-scratch CDNA4 kernel.
//
// Key MI355X design choices:
//  * mfma_f32_16x16x32_bf16 everywhere; all tiles are [rows][D] K-major so
//    A/B fragments are contiguous 16-B ds_read_b128 (no CUDA-style ldmatrix).
//  * Forward: S = Q K^T needs NO transposes (B-fragment of K^T reads K's
//    rows contiguously); P V stages V transposed (Vt) once per tile.
//  * Backward computes S^T = K Q^T and dP^T = V dO^T directly (again no
//    transposes), accumulates dK/dV per KV-tile in registers, and
//    scatter-adds dQ/dK/dV into fp32 buffers (GQA folds q-head groups).
//  * 4 waves/block, each owning 16 rows of the 64-row tile; online softmax
//    in fp32 with cross-lane (16-lane column group) shuffle reductions.
//  * LDS tiles XOR-swizzled (row&MASK)<<4 against the 16-lane ds_read_b128
//    bank conflict (guide §6 Guideline 4).
#include <torch/extension.h>
#include "ext_stream.h"
#include "common.h"

namespace {

using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int BM = 64;     // query-tile rows
constexpr int BN = 64;     // key-tile rows
constexpr int THREADS = 256;

DEV int swz(int row, int byte_in_row, int mask) {
  return byte_in_row ^ ((row & mask) << 4);
}

// Stage a row-major [ROWS][ROWB bytes] global tile into LDS via
// global_load_lds, inverse-swizzling the per-lane source (rule 21).
template <int ROWS, int ROWB, int MASK>
DEV void stage_rm(const bf16* __restrict__ gsrc, int64_t row_stride_elts,
                  char* lds, int tid, int max_row = 1 << 30) {
  constexpr int BYTES = ROWS * ROWB;
  constexpr int NINST = BYTES / 4096;
#pragma unroll
  for (int t = 0; t < NINST; ++t) {
    int lin = t * 4096 + tid * 16;
    int row = lin / ROWB;
    int cin = lin % ROWB;
    int sc = swz(row, cin, MASK);
    // clamp OOB rows to the last valid one: garbage bits can be NaN and
    // 0 * NaN = NaN inside the MFMAs even for masked entries
    const bf16* src = gsrc + (int64_t)min(row, max_row) * row_stride_elts
                    + sc / 2;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)src,
        (__attribute__((address_space(3))) void*)(lds + lin), 16, 0, 0);
  }
}

// Read one MFMA fragment (A or B; both are "row l&15, 8 contiguous
// K-elements at (l>>4)*8") from a swizzled row-major LDS tile.
template <int ROWB, int MASK>
DEV bf16x8 frag_rm(const char* lds, int row, int kbyte) {
  return *(const bf16x8*)(lds + row * ROWB + swz(row, kbyte, MASK));
}

// fragment C-layout helpers: col = lane&15, row = (lane>>4)*4 + reg
// row-stat reduction across the 16-lane column group:
DEV float colgroup_max(float x) {
  x = fmaxf(x, __shfl_xor(x, 1, WAVE));
  x = fmaxf(x, __shfl_xor(x, 2, WAVE));
  x = fmaxf(x, __shfl_xor(x, 4, WAVE));
  x = fmaxf(x, __shfl_xor(x, 8, WAVE));
  return x;
}
DEV float colgroup_sum(float x) {
  x += __shfl_xor(x, 1, WAVE);
  x += __shfl_xor(x, 2, WAVE);
  x += __shfl_xor(x, 4, WAVE);
  x += __shfl_xor(x, 8, WAVE);
  return x;
}

// ===========================================================================
// forward
// ===========================================================================
template <int D>
__global__ __launch_bounds__(THREADS) void fa_fwd_kernel(
    const bf16* __restrict__ Q, const bf16* __restrict__ K,
    const bf16* __restrict__ V, bf16* __restrict__ O,
    float* __restrict__ LSE, int B, int H, int Hkv, int S, int Skv,
    float scale, bool causal) {
  constexpr int ROWB = D * 2;            // Q/K tile row bytes
  constexpr int MASK = (D == 128) ? 15 : 7;
  constexpr int VT_ROWB = BN * 2;        // Vt row bytes (128)
  constexpr int VT_MASK = 7;
  constexpr int P_ROWB = BN * 2;
  constexpr int P_MASK = 7;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* q_lds = smem;                          // [BM][ROWB]
  char* k_lds = q_lds + BM * ROWB;             // [BN][ROWB]
  char* vt_lds = k_lds + BN * ROWB;            // [D][VT_ROWB]
  char* p_lds = vt_lds + D * VT_ROWB;          // [BM][P_ROWB]

  const int bh = blockIdx.y;                   // b*H + h
  const int h = bh % H;
  const int b = bh / H;
  const int hkv = h / (H / Hkv);
  const int q0 = blockIdx.x * BM;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wrow = wid * 16;                   // wave's rows within tile
  const int fr = lane & 15;
  const int kg = lane >> 4;

  const bf16* Qb = Q + ((int64_t)bh * S + q0) * D;
  const bf16* Kb = K + ((int64_t)(b * Hkv + hkv)) * Skv * D;
  const bf16* Vb = V + ((int64_t)(b * Hkv + hkv)) * Skv * D;

  // stage Q tile once
  stage_rm<BM, ROWB, MASK>(Qb, D, q_lds, tid, S - 1 - q0);

  f32x4 o_acc[D / 16];
#pragma unroll
  for (int c = 0; c < D / 16; ++c) o_acc[c] = {0.f, 0.f, 0.f, 0.f};
  float m_i[4], l_i[4];
#pragma unroll
  for (int j = 0; j < 4; ++j) { m_i[j] = -INFINITY; l_i[j] = 0.f; }

  const int n_tiles = causal
      ? min((q0 + BM + (Skv - S) + BN - 1) / BN, (Skv + BN - 1) / BN)
      : (Skv + BN - 1) / BN;
  // causal offset: query row r attends keys <= r + (Skv - S)
  const int diag_off = Skv - S;

  for (int kt = 0; kt < n_tiles; ++kt) {
    const int k0 = kt * BN;
    // ---- stage K tile; V transposed (scalar transpose) ----
    asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();            // prior P-reads/V-reads done
    stage_rm<BN, ROWB, MASK>(Kb + (int64_t)k0 * D, D, k_lds, tid,
                             Skv - 1 - k0);
    {
      // Vt[d][key]: each thread copies 8 contiguous d of one key row.
      constexpr int CHUNKS = BN * D / (THREADS * 8);
#pragma unroll
      for (int c = 0; c < CHUNKS; ++c) {
        int idx = (c * THREADS + tid) * 8;
        int key = idx / D;
        int d0 = idx % D;
        const bf16* src = Vb + (int64_t)min(k0 + key, Skv - 1) * D + d0;
        ushort8 u = *reinterpret_cast<const ushort8*>(src);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int row = d0 + j;
          *(unsigned short*)(vt_lds + row * VT_ROWB +
                             swz(row, key * 2, VT_MASK)) = u.v[j];
        }
      }
    }
    asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();

    // ---- S = Q K^T (this wave's 16 rows x 64 cols) ----
    f32x4 s_acc[4];
#pragma unroll
    for (int n = 0; n < 4; ++n) s_acc[n] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int kk = 0; kk < D / 32; ++kk) {
      bf16x8 qf = frag_rm<ROWB, MASK>(q_lds, wrow + fr,
                                      (kk * 32 + kg * 8) * 2);
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        bf16x8 kf = frag_rm<ROWB, MASK>(k_lds, n * 16 + fr,
                                        (kk * 32 + kg * 8) * 2);
        s_acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf, kf,
                                                           s_acc[n], 0, 0, 0);
      }
    }
    // ---- online softmax ----
    float pmax[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) pmax[j] = -INFINITY;
#pragma unroll
    for (int n = 0; n < 4; ++n) {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float sv = s_acc[n][j] * scale;
        if (causal) {
          int qrow = q0 + wrow + kg * 4 + j;
          int kcol = k0 + n * 16 + fr;
          if (kcol > qrow + diag_off) sv = -INFINITY;
        }
        if (k0 + n * 16 + fr >= Skv) sv = -INFINITY;
        s_acc[n][j] = sv;
        pmax[j] = fmaxf(pmax[j], sv);
      }
    }
#pragma unroll
    for (int j = 0; j < 4; ++j) pmax[j] = colgroup_max(pmax[j]);
    float alpha[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float mn = fmaxf(m_i[j], pmax[j]);
      alpha[j] = (m_i[j] == -INFINITY) ? 0.f : __expf(m_i[j] - mn);
      m_i[j] = mn;
    }
    // rescale O
#pragma unroll
    for (int c = 0; c < D / 16; ++c)
#pragma unroll
      for (int j = 0; j < 4; ++j) o_acc[c][j] *= alpha[j];
    // P = exp(S - m), write to this wave's private P rows (bf16, swizzled)
    float psum[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int n = 0; n < 4; ++n) {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float p = (s_acc[n][j] == -INFINITY) ? 0.f
                  : __expf(s_acc[n][j] - m_i[j]);
        psum[j] += p;
        int row = wrow + kg * 4 + j;
        int colb = (n * 16 + fr) * 2;
        *(unsigned short*)(p_lds + row * P_ROWB + swz(row, colb, P_MASK)) =
            f2bf(p);
      }
    }
#pragma unroll
    for (int j = 0; j < 4; ++j)
      l_i[j] = l_i[j] * alpha[j] + colgroup_sum(psum[j]);

    // ---- O += P @ V (reads this wave's own P rows; in-wave LDS ordering,
    // but fence so the compiler cannot reorder the typed reads) ----
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
#pragma unroll
    for (int kk = 0; kk < BN / 32; ++kk) {
      bf16x8 pf = frag_rm<P_ROWB, P_MASK>(p_lds, wrow + fr,
                                          (kk * 32 + kg * 8) * 2);
#pragma unroll
      for (int c = 0; c < D / 16; ++c) {
        bf16x8 vf = frag_rm<VT_ROWB, VT_MASK>(vt_lds, c * 16 + fr,
                                              (kk * 32 + kg * 8) * 2);
        o_acc[c] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf, vf,
                                                           o_acc[c], 0, 0, 0);
      }
    }
  }

  // ---- epilogue: O /= l; write O and LSE ----
  bf16* Ob = O + ((int64_t)bh * S + q0) * D;
  float* Lb = LSE + (int64_t)bh * S + q0;
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    int row = wrow + kg * 4 + j;
    if (q0 + row < S) {
      float inv = (l_i[j] > 0.f) ? 1.f / l_i[j] : 0.f;
#pragma unroll
      for (int c = 0; c < D / 16; ++c) {
        int col = c * 16 + fr;
        Ob[(int64_t)row * D + col] = (bf16)(o_acc[c][j] * inv);
      }
      if (fr == 0)
        Lb[row] = (l_i[j] > 0.f) ? m_i[j] + __logf(l_i[j]) : -INFINITY;
    }
  }
}

// ===========================================================================
// backward: delta = rowsum(dO * O)
// ===========================================================================
__global__ void fa_bwd_delta_kernel(const bf16* __restrict__ dO,
                                    const bf16* __restrict__ O,
                                    float* __restrict__ delta,
                                    int64_t rows, int D) {
  __shared__ float smem[16];
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const bf16* a = dO + row * D;
    const bf16* b = O + row * D;
    float s = 0.f;
    for (int i = threadIdx.x * 8; i + 8 <= D; i += 256 * 8) {
      float av[8], bv[8];
      VecIO<bf16>::load(a + i, av);
      VecIO<bf16>::load(b + i, bv);
#pragma unroll
      for (int j = 0; j < 8; ++j) s += av[j] * bv[j];
    }
    s = block_sum(s, smem);
    if (threadIdx.x == 0) delta[row] = s;
  }
}

// ===========================================================================
// backward main: one workgroup per (bh, kv-tile); accumulates dK/dV in
// registers, atomically adds dQ (fp32) and dK/dV (fp32, GQA-folded).
// ===========================================================================
template <int D>
__global__ __launch_bounds__(THREADS) void fa_bwd_kernel(
    const bf16* __restrict__ Q, const bf16* __restrict__ K,
    const bf16* __restrict__ V, const bf16* __restrict__ dO,
    const float* __restrict__ LSE, const float* __restrict__ DELTA,
    float* __restrict__ dQ, float* __restrict__ dK, float* __restrict__ dV,
    int B, int H, int Hkv, int S, int Skv, float scale, bool causal) {
  constexpr int ROWB = D * 2;
  constexpr int MASK = (D == 128) ? 15 : 7;
  constexpr int T_ROWB = BM * 2;   // transposed-tile row bytes (q-major)
  constexpr int T_MASK = 7;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* k_lds = smem;                       // [BN][ROWB]
  char* v_lds = k_lds + BN * ROWB;          // [BN][ROWB]
  char* q_lds = v_lds + BN * ROWB;          // [BM][ROWB]
  char* do_lds = q_lds + BM * ROWB;         // [BM][ROWB]
  char* qt_lds = do_lds + BM * ROWB;        // [D][T_ROWB]
  char* dot_lds = qt_lds + D * T_ROWB;      // [D][T_ROWB]
  char* kt_lds = dot_lds + D * T_ROWB;      // [D][BN*2]
  char* pt_lds = kt_lds + D * BN * 2;       // [BN][T_ROWB]
  char* dst_lds = pt_lds + BN * T_ROWB;     // [BN][T_ROWB] (dS^T)
  char* ds_lds = dst_lds + BN * T_ROWB;     // [BM][BN*2]  (dS)

  const int bh = blockIdx.y;
  const int h = bh % H;
  const int b = bh / H;
  const int hkv = h / (H / Hkv);
  const int k0 = blockIdx.x * BN;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wrow = wid * 16;     // wave's 16 keys within tile
  const int fr = lane & 15;
  const int kg = lane >> 4;

  const bf16* Kb = K + ((int64_t)(b * Hkv + hkv) * Skv + k0) * D;
  const bf16* Vb = V + ((int64_t)(b * Hkv + hkv) * Skv + k0) * D;
  const bf16* Qh = Q + (int64_t)bh * S * D;
  const bf16* dOh = dO + (int64_t)bh * S * D;
  const float* lse_h = LSE + (int64_t)bh * S;
  const float* del_h = DELTA + (int64_t)bh * S;

  // stage K, V, Kt once per workgroup
  stage_rm<BN, ROWB, MASK>(Kb, D, k_lds, tid, Skv - 1 - k0);
  stage_rm<BN, ROWB, MASK>(Vb, D, v_lds, tid, Skv - 1 - k0);
  {
    constexpr int CHUNKS = BN * D / (THREADS * 8);
#pragma unroll
    for (int c = 0; c < CHUNKS; ++c) {
      int idx = (c * THREADS + tid) * 8;
      int key = idx / D;
      int d0 = idx % D;
      ushort8 u = *reinterpret_cast<const ushort8*>(
          Kb + (int64_t)min(key, Skv - 1 - k0) * D + d0);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int row = d0 + j;
        *(unsigned short*)(kt_lds + row * BN * 2 +
                           swz(row, key * 2, T_MASK)) = u.v[j];
      }
    }
  }

  f32x4 dk_acc[D / 16], dv_acc[D / 16];
#pragma unroll
  for (int c = 0; c < D / 16; ++c) {
    dk_acc[c] = {0.f, 0.f, 0.f, 0.f};
    dv_acc[c] = {0.f, 0.f, 0.f, 0.f};
  }

  const int diag_off = Skv - S;
  // causal: keys k only see queries q with q + diag_off >= k
  int qt_start = causal ? max(0, (k0 - diag_off) / BM) : 0;

  for (int qt = qt_start; qt * BM < S; ++qt) {
    const int q0 = qt * BM;
    asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    stage_rm<BM, ROWB, MASK>(Qh + (int64_t)q0 * D, D, q_lds, tid,
                             S - 1 - q0);
    stage_rm<BM, ROWB, MASK>(dOh + (int64_t)q0 * D, D, do_lds, tid,
                             S - 1 - q0);
    {
      constexpr int CHUNKS = BM * D / (THREADS * 8);
#pragma unroll
      for (int c = 0; c < CHUNKS; ++c) {
        int idx = (c * THREADS + tid) * 8;
        int qr = idx / D;
        int d0 = idx % D;
        ushort8 uq = *reinterpret_cast<const ushort8*>(
            Qh + (int64_t)min(q0 + qr, S - 1) * D + d0);
        ushort8 ud = *reinterpret_cast<const ushort8*>(
            dOh + (int64_t)min(q0 + qr, S - 1) * D + d0);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int row = d0 + j;
          *(unsigned short*)(qt_lds + row * T_ROWB +
                             swz(row, qr * 2, T_MASK)) = uq.v[j];
          *(unsigned short*)(dot_lds + row * T_ROWB +
                             swz(row, qr * 2, T_MASK)) = ud.v[j];
        }
      }
    }
    asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();

    // ---- S^T = K Q^T: rows=this wave's 16 keys, cols=64 queries ----
    f32x4 st_acc[4];
#pragma unroll
    for (int n = 0; n < 4; ++n) st_acc[n] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int kk = 0; kk < D / 32; ++kk) {
      bf16x8 kf = frag_rm<ROWB, MASK>(k_lds, wrow + fr,
                                      (kk * 32 + kg * 8) * 2);
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        // B-frag of Q^T: lane l -> Q[q = n*16 + (l&15)][d contiguous]
        bf16x8 qf = frag_rm<ROWB, MASK>(q_lds, n * 16 + fr,
                                        (kk * 32 + kg * 8) * 2);
        st_acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            kf, qf, st_acc[n], 0, 0, 0);
      }
    }
    // ---- dP^T = V dO^T ----
    f32x4 dpt_acc[4];
#pragma unroll
    for (int n = 0; n < 4; ++n) dpt_acc[n] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int kk = 0; kk < D / 32; ++kk) {
      bf16x8 vf = frag_rm<ROWB, MASK>(v_lds, wrow + fr,
                                      (kk * 32 + kg * 8) * 2);
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        bf16x8 df = frag_rm<ROWB, MASK>(do_lds, n * 16 + fr,
                                        (kk * 32 + kg * 8) * 2);
        dpt_acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            vf, df, dpt_acc[n], 0, 0, 0);
      }
    }

    // ---- P^T = exp(S^T*scale - lse[q]); dS^T = P^T*(dP^T - delta[q])*scale
    // C-layout: row = key = wrow + kg*4 + j, col = q = n*16 + fr.
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      int qcol = q0 + n * 16 + fr;
      float lse_q = (qcol < S) ? lse_h[qcol] : INFINITY;
      float del_q = (qcol < S) ? del_h[qcol] : 0.f;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int krow = k0 + wrow + kg * 4 + j;
        float sv = st_acc[n][j] * scale;
        bool masked = (qcol >= S) || (krow >= Skv) ||
                      (causal && krow > qcol + diag_off);
        float p = masked ? 0.f : __expf(sv - lse_q);
        // guard: OOB-staged rows can hold NaN bit patterns; 0*NaN = NaN
        float ds = masked ? 0.f : p * (dpt_acc[n][j] - del_q) * scale;
        st_acc[n][j] = p;      // now holds P^T
        dpt_acc[n][j] = ds;    // now holds dS^T
        int row = wrow + kg * 4 + j;
        int colb = (n * 16 + fr) * 2;
        *(unsigned short*)(pt_lds + row * T_ROWB + swz(row, colb, T_MASK)) =
            f2bf(p);
        *(unsigned short*)(dst_lds + row * T_ROWB + swz(row, colb, T_MASK)) =
            f2bf(ds);
        // dS (q-major) for the dQ pass — needs all waves' keys: shared
        int dsrow = n * 16 + fr;        // q row
        int dscolb = (wrow + kg * 4 + j) * 2;   // key col
        *(unsigned short*)(ds_lds + dsrow * (BN * 2) +
                           swz(dsrow, dscolb, T_MASK)) = f2bf(ds);
      }
    }

    // ---- dV += P^T dO ; dK += dS^T Q  (own rows; fence vs reorder) ----
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
#pragma unroll
    for (int kk = 0; kk < BM / 32; ++kk) {
      bf16x8 ptf = frag_rm<T_ROWB, T_MASK>(pt_lds, wrow + fr,
                                           (kk * 32 + kg * 8) * 2);
      bf16x8 dstf = frag_rm<T_ROWB, T_MASK>(dst_lds, wrow + fr,
                                            (kk * 32 + kg * 8) * 2);
#pragma unroll
      for (int c = 0; c < D / 16; ++c) {
        // B-frag of dO: lane l -> dOt[d = c*16+(l&15)][q contiguous]
        bf16x8 dof = frag_rm<T_ROWB, T_MASK>(dot_lds, c * 16 + fr,
                                             (kk * 32 + kg * 8) * 2);
        dv_acc[c] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            ptf, dof, dv_acc[c], 0, 0, 0);
        bf16x8 qtf = frag_rm<T_ROWB, T_MASK>(qt_lds, c * 16 + fr,
                                             (kk * 32 + kg * 8) * 2);
        dk_acc[c] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            dstf, qtf, dk_acc[c], 0, 0, 0);
      }
    }

    // ---- dQ_part = dS K : wave handles its 16 q rows over all 64 keys ----
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();   // ds_lds written by all waves
    {
      f32x4 dq_acc[D / 16];
#pragma unroll
      for (int c = 0; c < D / 16; ++c) dq_acc[c] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < BN / 32; ++kk) {
        bf16x8 dsf = frag_rm<BN * 2, T_MASK>(ds_lds, wrow + fr,
                                             (kk * 32 + kg * 8) * 2);
#pragma unroll
        for (int c = 0; c < D / 16; ++c) {
          bf16x8 ktf = frag_rm<BN * 2, T_MASK>(kt_lds, c * 16 + fr,
                                               (kk * 32 + kg * 8) * 2);
          dq_acc[c] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              dsf, ktf, dq_acc[c], 0, 0, 0);
        }
      }
      float* dQb = dQ + (int64_t)bh * S * D;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int qrow = q0 + wrow + kg * 4 + j;
        if (qrow < S) {
#pragma unroll
          for (int c = 0; c < D / 16; ++c)
            atomicAdd(dQb + (int64_t)qrow * D + c * 16 + fr, dq_acc[c][j]);
        }
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();   // before next iteration overwrites LDS
  }

  // ---- write dK/dV (fp32 atomic: GQA head groups fold together) ----
  float* dKb = dK + ((int64_t)(b * Hkv + hkv)) * Skv * D;
  float* dVb = dV + ((int64_t)(b * Hkv + hkv)) * Skv * D;
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    int krow = k0 + wrow + kg * 4 + j;
    if (krow < Skv) {
#pragma unroll
      for (int c = 0; c < D / 16; ++c) {
        atomicAdd(dKb + (int64_t)krow * D + c * 16 + fr, dk_acc[c][j]);
        atomicAdd(dVb + (int64_t)krow * D + c * 16 + fr, dv_acc[c][j]);
      }
    }
  }
}

}  // namespace

bool fa2_fwd_supported(int D, int S);
void fa2_fwd_launch(const void* q, const void* k, const void* v, void* o,
                    float* lse, int B, int H, int Hkv, int S, int Skv,
                    float scale, bool causal, hipStream_t stream,
                    FaStrides sq, FaStrides skv, FaStrides so);

std::vector<torch::Tensor> flash_attn_fwd(torch::Tensor q, torch::Tensor k,
                                          torch::Tensor v, bool causal,
                                          double scale) {
  TORCH_CHECK(q.dim() == 4, "q must be [B,H,S,D]");
  TORCH_CHECK(q.scalar_type() == at::kBFloat16, "flash_attn: bf16 only");
  const int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  const int Hkv = k.size(1), Skv = k.size(2);
  TORCH_CHECK(D == 64 || D == 128, "flash_attn: head dim 64/128");
  TORCH_CHECK(H % Hkv == 0, "GQA requires H % Hkv == 0");
  auto o = torch::empty_like(q);
  auto lse = torch::empty({B, H, S}, q.options().dtype(at::kFloat));
  auto stream = hetu_current_stream();
  static const bool force_v1 = [] {
    const char* e = getenv("HETU_AMD_FA1");
    return e && e[0] == '1';
  }();
  if (!force_v1 && fa2_fwd_supported(D, S)) {
    FaStrides sq{(long long)H * S * D, (long long)S * D, (long long)D};
    FaStrides skv{(long long)Hkv * Skv * D, (long long)Skv * D,
                  (long long)D};
    fa2_fwd_launch(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                   lse.data_ptr<float>(), B, H, Hkv, S, Skv, (float)scale,
                   causal, stream, sq, skv, sq);
    return {o, lse};
  }
  dim3 grid((S + BM - 1) / BM, B * H);
  size_t lds = (size_t)BM * D * 2 + BN * D * 2 + D * BN * 2 + BM * BN * 2;
  if (D == 128) {
    hipLaunchKernelGGL(fa_fwd_kernel<128>, grid, dim3(THREADS), lds, stream,
                       (const bf16*)q.data_ptr(), (const bf16*)k.data_ptr(),
                       (const bf16*)v.data_ptr(), (bf16*)o.data_ptr(),
                       lse.data_ptr<float>(), B, H, Hkv, S, Skv,
                       (float)scale, causal);
  } else {
    hipLaunchKernelGGL(fa_fwd_kernel<64>, grid, dim3(THREADS), lds, stream,
                       (const bf16*)q.data_ptr(), (const bf16*)k.data_ptr(),
                       (const bf16*)v.data_ptr(), (bf16*)o.data_ptr(),
                       lse.data_ptr<float>(), B, H, Hkv, S, Skv,
                       (float)scale, causal);
  }
  return {o, lse};
}

bool fa2_bwd_supported(int D, int S);
std::vector<torch::Tensor> fa2_bwd_launch(
    torch::Tensor dout, torch::Tensor q, torch::Tensor k, torch::Tensor v,
    torch::Tensor out, torch::Tensor lse, bool causal, double scale);

std::vector<torch::Tensor> flash_attn_bwd(torch::Tensor dout,
                                          torch::Tensor q, torch::Tensor k,
                                          torch::Tensor v, torch::Tensor out,
                                          torch::Tensor lse, bool causal,
                                          double scale) {
  const int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  const int Hkv = k.size(1), Skv = k.size(2);
  static const bool bwd_force_v1 = [] {
    const char* e = getenv("HETU_AMD_FA1");
    return e && e[0] == '1';
  }();
  if (!bwd_force_v1 && fa2_bwd_supported(D, S)) {
    return fa2_bwd_launch(dout, q, k, v, out, lse, causal, scale);
  }
  auto stream = hetu_current_stream();
  auto delta = torch::empty({B, H, S}, q.options().dtype(at::kFloat));
  {
    int64_t rows = (int64_t)B * H * S;
    int grid = (int)std::min<int64_t>(rows, 8192);
    hipLaunchKernelGGL(fa_bwd_delta_kernel, dim3(grid), dim3(256), 0, stream,
                       (const bf16*)dout.data_ptr(),
                       (const bf16*)out.data_ptr(), delta.data_ptr<float>(),
                       rows, D);
  }
  auto dq32 = torch::zeros_like(q, q.options().dtype(at::kFloat));
  auto dk32 = torch::zeros_like(k, k.options().dtype(at::kFloat));
  auto dv32 = torch::zeros_like(v, v.options().dtype(at::kFloat));
  dim3 grid((Skv + BN - 1) / BN, B * H);
  size_t lds = (size_t)4 * BM * D * 2   // k,v,q,do (BM==BN)
             + 3 * D * BM * 2           // qt, dot, kt
             + 3 * BM * BN * 2;         // pt, dst, ds
  if (D == 128) {
    hipLaunchKernelGGL(fa_bwd_kernel<128>, grid, dim3(THREADS), lds, stream,
                       (const bf16*)q.data_ptr(), (const bf16*)k.data_ptr(),
                       (const bf16*)v.data_ptr(),
                       (const bf16*)dout.data_ptr(), lse.data_ptr<float>(),
                       delta.data_ptr<float>(), dq32.data_ptr<float>(),
                       dk32.data_ptr<float>(), dv32.data_ptr<float>(),
                       B, H, Hkv, S, Skv, (float)scale, causal);
  } else {
    hipLaunchKernelGGL(fa_bwd_kernel<64>, grid, dim3(THREADS), lds, stream,
                       (const bf16*)q.data_ptr(), (const bf16*)k.data_ptr(),
                       (const bf16*)v.data_ptr(),
                       (const bf16*)dout.data_ptr(), lse.data_ptr<float>(),
                       delta.data_ptr<float>(), dq32.data_ptr<float>(),
                       dk32.data_ptr<float>(), dv32.data_ptr<float>(),
                       B, H, Hkv, S, Skv, (float)scale, causal);
  }
  return {dq32.to(q.scalar_type()), dk32.to(k.scalar_type()),
          dv32.to(v.scalar_type())};
}
