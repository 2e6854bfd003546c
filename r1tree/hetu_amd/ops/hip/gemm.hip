// Hand-written bf16 MFMA GEMM for gfx950: C[M,N] = A[M,K] @ W[N,K]^T.
//
// "TN" form: both operands K-major, so every MFMA fragment load is a
// contiguous 16-byte ds_read_b128 (A-frag: lane l -> A[row l&15][k
// (l>>4)*8..+8]; B-frag: lane l -> W[col l&15][k (l>>4)*8..+8]) — the
// layout the transformer forward linears (y = x @ W^T, torch convention
// W[N,K]) use natively.
//
// Structure (CDNA HIP guide §5 ladder step 3): 128x128 tile, BK=64, 4 waves
// (2x2), double-buffered LDS staged with __builtin_amdgcn_global_load_lds
// width 16, per-lane-source XOR swizzle (rule 21: linear LDS dest,
// inverse-swizzled source, swizzled read), mfma_f32_16x16x32_bf16,
// XCD-aware bijective blockIdx remap (T1).
// Replaces the reference's cuBLAS MatMul path (hetu/impl/cuda/CUDABlas.cc)
// with a native CDNA4 kernel for the hot-path shapes.
#include <torch/extension.h>
#include "ext_stream.h"
#include "common.h"

namespace {

using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int BM = 128, BN = 128, BK = 64;
constexpr int THREADS = 256;

// XOR swizzle on the byte offset within a 128-B LDS row (bits 4..6):
// spreads the 16-lane ds_read_b128 group over 8 slots (<=2-way conflict).
DEV int swz(int row, int byte_in_row) {
  return byte_in_row ^ ((row & 7) << 4);
}

// Stage a [ROWS][BK] bf16 K-major tile from global to LDS with
// global_load_lds (lane-linear dest), applying the inverse swizzle on the
// per-lane SOURCE address. Each instruction moves 256 lanes x 16 B = 4 KiB.
// ROWS*BK*2 bytes total -> ROWS*BK*2/4096 instructions.
template <int ROWS>
DEV void stage_tile(const bf16* __restrict__ gsrc, int64_t ld,  // elements
                    char* lds /*byte base of tile*/, int tid) {
  constexpr int BYTES = ROWS * BK * 2;
  constexpr int NINST = BYTES / 4096;
#pragma unroll
  for (int t = 0; t < NINST; ++t) {
    int lin = t * 4096 + tid * 16;          // linear LDS byte offset
    int row = lin / (BK * 2);
    int cin = lin % (BK * 2);
    int src_c = swz(row, cin);              // inverse == forward (involution)
    const bf16* src = gsrc + (int64_t)row * ld + src_c / 2;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)src,
        (__attribute__((address_space(3))) void*)(lds + lin),
        16, 0, 0);
  }
}

__global__ __launch_bounds__(THREADS) void gemm_tn_bf16_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ W,
    bf16* __restrict__ C, int M, int N, int K, int nwg_m, int nwg_n) {
  // dynamic LDS: [2 buffers][A 128*64 | W 128*64] bf16
  extern __shared__ __attribute__((aligned(16))) char smem[];
  constexpr u32 ASZ = BM * BK * 2;          // 16 KiB
  constexpr u32 WSZ = BN * BK * 2;
  constexpr u32 BUF = ASZ + WSZ;            // 32 KiB per buffer

  // ---- XCD-aware bijective remap (T1) ----
  int nwg = nwg_m * nwg_n;
  int orig = blockIdx.x;
  int q = nwg / 8, r = nwg % 8;
  int xcd = orig % 8, slot = orig / 8;
  int wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + slot;
  int bm = (wgid / nwg_n) * BM;
  int bn = (wgid % nwg_n) * BN;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;                 // 4 waves: 2x2
  const int wr = (wid >> 1) * 64;           // wave row offset in tile
  const int wc = (wid & 1) * 64;            // wave col offset

  const int frag_row = lane & 15;           // fragment row/col within 16
  const int kgrp = lane >> 4;               // 0..3 -> k-offset *8

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const bf16* Ab = A + (int64_t)bm * K;
  const bf16* Wb = W + (int64_t)bn * K;

  // prologue: stage tile 0 into buffer 0
  stage_tile<BM>(Ab, K, smem, tid);
  stage_tile<BN>(Wb, K, smem + ASZ, tid);

  const int ktiles = K / BK;
  int cur = 0;
  for (int kt = 0; kt < ktiles; ++kt) {
    // issue next-tile staging into the other buffer before computing
    if (kt + 1 < ktiles) {
      const bf16* An = Ab + (kt + 1) * BK;
      const bf16* Wn = Wb + (kt + 1) * BK;
      stage_tile<BM>(An, K, smem + (cur ^ 1) * BUF, tid);
      stage_tile<BN>(Wn, K, smem + (cur ^ 1) * BUF + ASZ, tid);
    }
    // wait for CURRENT buffer: all its loads were issued before the next
    // tile's, so vmcnt counts: next-tile has 16 glds in flight.
    if (kt + 1 < ktiles)
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();

    const char* abuf = smem + cur * BUF;
    const char* wbuf = smem + cur * BUF + ASZ;
    // 2 MFMA K-steps of 32 within BK=64
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 afrag[4], wfrag[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int arow = wr + i * 16 + frag_row;
        int abyte = swz(arow, (kk * 32 + kgrp * 8) * 2);
        afrag[i] = *(const bf16x8*)(abuf + arow * (BK * 2) + abyte);
      }
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int wrow = wc + j * 16 + frag_row;
        int wbyte = swz(wrow, (kk * 32 + kgrp * 8) * 2);
        wfrag[j] = *(const bf16x8*)(wbuf + wrow * (BK * 2) + wbyte);
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[i], wfrag[j], acc[i][j], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    __builtin_amdgcn_s_barrier();
    cur ^= 1;
  }

  // epilogue: C[row][col], row = bm+wr+i*16+(lane>>4)*4+reg,
  // col = bn+wc+j*16+(lane&15)
  const int c_col0 = bn + wc + frag_row;
  const int c_row0 = bm + wr + kgrp * 4;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
#pragma unroll
      for (int rg = 0; rg < 4; ++rg) {
        int row = c_row0 + i * 16 + rg;
        int col = c_col0 + j * 16;
        if (row < M && col < N)
          C[(int64_t)row * N + col] = (bf16)acc[i][j][rg];
      }
    }
  }
}

}  // namespace

torch::Tensor gemm_bf16(torch::Tensor x, torch::Tensor w, bool trans_w) {
  TORCH_CHECK(trans_w, "gemm_bf16: only TN (w [N,K]) supported");
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 &&
              w.scalar_type() == at::kBFloat16, "gemm_bf16: bf16 only");
  const int M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K, "shape mismatch");
  TORCH_CHECK(M % BM == 0 && N % BN == 0 && K % BK == 0,
              "gemm_bf16 requires M%128==0, N%128==0, K%64==0 "
              "(caller falls back to library GEMM otherwise)");
  auto y = torch::empty({M, N}, x.options());
  auto stream = hetu_current_stream();
  int nwg_m = M / BM, nwg_n = N / BN;
  size_t lds = 2 * (BM + BN) * BK * 2;
  hipLaunchKernelGGL(gemm_tn_bf16_kernel, dim3(nwg_m * nwg_n), dim3(THREADS),
                     lds, stream, (const bf16*)x.data_ptr(),
                     (const bf16*)w.data_ptr(), (bf16*)y.data_ptr(),
                     M, N, K, nwg_m, nwg_n);
  return y;
}
