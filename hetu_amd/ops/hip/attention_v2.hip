// Flash-attention 2 forward, 8-wave 32x32-MFMA structure for gfx950.
//
// Replaces the reference's vendored CUTLASS flash-attn fwd
// (hetu/impl/kernel/FlashAttention.cu) with the CDNA4-native design from the
// MI355X kernel playbook (cdna_hip_programming.md Appendix B):
//  * 8 waves x 32 q-rows (BM=256), KV tiles of 64, mfma_f32_32x32x16_bf16.
//  * swapped QK^T: S^T = mfma(K, Q) puts a full P-row in each lane's
//    registers (col = lane&31 = q) -> softmax entirely in-register, the
//    only cross-lane step is a lane<->lane+32 half exchange.
//  * P -> bf16 A-fragments via v_cvt_pk_bf16_f32 + v_permlane32_swap_b32
//    (no P round-trip through LDS).
//  * PV swapped too: O^T += mfma(V^T, P) keeps the softmax statistics
//    lane-local for the O rescale.
//  * K in LDS row-major XOR-swizzled (conflict-free ds_read_b128);
//    V in [32key][16d] subtiles consumed by ds_read_b64_tr_b16 (the 4x4
//    hardware transpose read).
//  * register staging (T14): issue next tile's global loads before compute,
//    ds_write after the barrier; double-buffered LDS.
#include <torch/extension.h>
#include "ext_stream.h"
#include "common.h"

namespace {

using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x16 = __attribute__((ext_vector_type(16))) float;
using u32x2 = __attribute__((ext_vector_type(2))) unsigned int;

constexpr int BM = 256;       // q rows per workgroup (8 waves x 32)
constexpr int BN = 64;        // kv tile
constexpr int THREADS = BM * 2;   // one wave per 32 q rows
constexpr int NGLDS = 64 * 256 / (THREADS * 16);  // glds per lane per image
constexpr int NBUF = 3;       // KV ring depth (counted-vmcnt pipelining)

// K tile LDS image: row-major [64][256B], byte-in-row ^ ((row&15)<<4)
DEV int kswz(int row, int byte_in_row) {
  return row * 256 + (byte_in_row ^ ((row & 15) << 4));
}

// V tile LDS image: [kb][db] subtiles of [32 keys][16 d] bf16 (1 KiB each,
// row stride 32 B): key k, dim d -> subtile (k>>5, d>>4)
DEV int voff(int key, int d) {
  return ((key >> 5) * 8 + (d >> 4)) * 1024 + (key & 31) * 32 + (d & 15) * 2;
}

template <int D>  // D = head dim (128)
// 96 KiB LDS -> one workgroup per CU regardless; a 2-block register cap
// would only force tighter allocation for a block that cannot schedule
__global__ __launch_bounds__(THREADS, 1) void fa2_fwd_kernel(
    const bf16* __restrict__ Q, const bf16* __restrict__ K,
    const bf16* __restrict__ V, bf16* __restrict__ O,
    float* __restrict__ LSE, int B, int H, int Hkv, int S, int Skv,
    float scale, bool causal, FaStrides sq, FaStrides skv, FaStrides so,
    const int* __restrict__ cu, int lse_T) {
  static_assert(D == 128, "fa2 fwd: D=128 only");
  constexpr int KBYTES = BN * D * 2;          // 16 KiB
  extern __shared__ __attribute__((aligned(16))) char smem[];
  auto k_lds = [&](int b) -> char* { return smem + b * KBYTES; };
  auto v_lds = [&](int b) -> char* { return smem + (NBUF + b) * KBYTES; };

  // grid is (bh, q-block): consecutive blockIdx.x (-> XCD b%8 placement)
  // walk bh, so every CU samples ALL causal depths — with (q-block, bh)
  // each CU aliased onto ONE depth and causal ran as slow as non-causal
  const int bh = blockIdx.x;
  const int h = bh % H;
  const int b = bh / H;
  const int hkv = h / (H / Hkv);
  const int q0 = blockIdx.y * BM;
  // packed-varlen mode (reference mha_varlen_fwd): blockIdx "batch" is a
  // SEGMENT of the packed [T, H, D] layout; segment bounds come from the
  // device-side cu_seqlens — one launch covers every ragged segment, no
  // host loop, no host sync.
  int row0 = 0;
  if (cu != nullptr) {
    row0 = cu[b];
    S = cu[b + 1] - row0;
    Skv = S;
    if (q0 >= S) return;        // ragged tail tile of a shorter segment
  }

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int iq = lane & 31;      // this lane's q row within the wave block
  const int hi = lane >> 5;

  const int64_t qb_off = cu ? (int64_t)row0 * sq.rs : (int64_t)b * sq.bs;
  const int64_t kb_off = cu ? (int64_t)row0 * skv.rs
                            : (int64_t)b * skv.bs;
  const bf16* Qb = Q + qb_off + (int64_t)h * sq.hs;
  const bf16* Kb = K + kb_off + (int64_t)hkv * skv.hs;
  const bf16* Vb = V + kb_off + (int64_t)hkv * skv.hs;

  const int my_q = q0 + wid * 32 + iq;          // global q row (this lane)
  const int diag = Skv - S;                     // causal offset

  // ---- Q -> registers, prescaled by softmax scale --------------------
  // lane holds Q[my_q][16*kk + 8*hi .. +8) for kk = 0..7
  bf16x8 qreg[8];
  {
    const bf16* qrow = Qb + (int64_t)min(my_q, S - 1) * sq.rs;
#pragma unroll
    for (int kk = 0; kk < 8; ++kk) {
      ushort8 u = *reinterpret_cast<const ushort8*>(qrow + kk * 16 + hi * 8);
      union { ushort8 us; bf16x8 v; } cvt;
#pragma unroll
      for (int j = 0; j < 8; ++j) cvt.us.v[j] = f2bf(bf2f(u.v[j]) * scale);
      qreg[kk] = cvt.v;
    }
  }

  // ---- staging: global_load_lds DMA, swizzle folded into the SOURCE
  // address (rule 21: glds writes lane-linear, dest = base + lane*16).
  // Per tile: 16 KiB K + 16 KiB V = 16+16 glds instructions; each wave
  // issues 2 K + 2 V (1 KiB each).  For linear byte position `pos` in the
  // tile image, the source element is the inverse of the LDS layout:
  //   K (row-major [64][256B], XOR swizzle): row = pos>>8,
  //     bytecol = (pos&255) ^ ((row&15)<<4), d = bytecol>>1
  //   V ([kb][db] subtiles of [32key][16d]): subtile = pos>>10,
  //     key = (subtile>>3)*32 + ((pos>>5)&31),
  //     d = (subtile&7)*16 + ((pos>>4)&1)*8
  const int wlane16 = lane * 16;
#define ISSUE_GLDS(k0, buf)                                                 \
  do {                                                                      \
    _Pragma("unroll")                                                       \
    for (int i = 0; i < NGLDS; ++i) {                                       \
      int pos = (wid * NGLDS + i) * 1024 + wlane16;                             \
      int krow = pos >> 8;                                                  \
      int kd = ((pos & 255) ^ ((krow & 15) << 4)) >> 1;                     \
      const bf16* ksrc =                                                    \
          Kb + (int64_t)min((k0) + krow, Skv - 1) * skv.rs + kd;            \
      __builtin_amdgcn_global_load_lds(                                     \
          (const __attribute__((address_space(1))) void*)ksrc,              \
          (__attribute__((address_space(3))) void*)(k_lds(buf) + pos),      \
          16, 0, 0);                                                        \
      int st = pos >> 10;                                                   \
      int vkey = (st >> 3) * 32 + ((pos >> 5) & 31);                        \
      int vd = (st & 7) * 16 + ((pos >> 4) & 1) * 8;                        \
      const bf16* vsrc =                                                    \
          Vb + (int64_t)min((k0) + vkey, Skv - 1) * skv.rs + vd;            \
      __builtin_amdgcn_global_load_lds(                                     \
          (const __attribute__((address_space(1))) void*)vsrc,              \
          (__attribute__((address_space(3))) void*)(v_lds(buf) + pos),      \
          16, 0, 0);                                                        \
    }                                                                       \
  } while (0)

  // ---- accumulators ---------------------------------------------------
  f32x16 o_acc[4];                 // O^T[32*dt + crow(reg,hi)][q = iq]
#pragma unroll
  for (int dt = 0; dt < 4; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r) o_acc[dt][r] = 0.f;
  float m_i = -INFINITY, l_i = 0.f;

  const int wave_qmax = q0 + wid * 32 + 31;
  int n_tiles = (Skv + BN - 1) / BN;
  if (causal) {
    int lim = (q0 + BM + diag + BN - 1) / BN;   // block-level causal bound
    n_tiles = min(n_tiles, max(lim, 1));
  }

  // prologue: tiles 0 and 1 in flight (3-deep ring, counted vmcnt:
  // per wave each tile is 2*NGLDS glds; waiting vmcnt(2*NGLDS) at the
  // loop head means "tile t landed, tile t+1 still flying" — the DMA for
  // t+2 then issues after the barrier and lands under two tiles of MFMAs)
  ISSUE_GLDS(0, 0);
  // the counted wait below assumes a full tile's loads may trail behind
  // the newest ISSUE; with a single tile, issue a harmless duplicate so
  // tile 0 is guaranteed landed when its compute starts
  if (n_tiles > 1) ISSUE_GLDS(BN, 1);
  else ISSUE_GLDS(0, 1);

  for (int t = 0; t < n_tiles; ++t) {
    const int k0 = t * BN;
    const int cur = t % NBUF;
    asm volatile("s_waitcnt vmcnt(%0)" :: "i"(2 * NGLDS) : "memory");
    __builtin_amdgcn_s_barrier();
    if (t + 2 < n_tiles) ISSUE_GLDS(k0 + 2 * BN, (t + 2) % NBUF);

    // does this wave have any unmasked key in this tile?
    const bool active = !causal || (k0 <= wave_qmax + diag);

    float p[2][16];                // P^T rows: p[ct][r] = P[q][32ct+crow]
    if (active) {
      // ---- S^T = K Q^T ------------------------------------------------
#pragma unroll
      for (int ct = 0; ct < 2; ++ct) {
        f32x16 acc;
#pragma unroll
        for (int r = 0; r < 16; ++r) acc[r] = 0.f;
#pragma unroll
        for (int kk = 0; kk < 8; ++kk) {
          // A-frag: K rows 32*ct + (lane&31), d slice 16kk + 8hi..+8
          bf16x8 kf = *reinterpret_cast<const bf16x8*>(
              k_lds(cur) + kswz(32 * ct + iq, kk * 32 + hi * 16));
          acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qreg[kk], acc,
                                                        0, 0, 0);
        }
#pragma unroll
        for (int r = 0; r < 16; ++r) p[ct][r] = acc[r];
      }

      // ---- mask + online softmax (all in-register; q = iq lane-local) --
      float pmax = -INFINITY;
#pragma unroll
      for (int ct = 0; ct < 2; ++ct)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int key = k0 + 32 * ct + (r & 3) + 8 * (r >> 2) + 4 * hi;
          bool masked = (key >= Skv) || (my_q >= S) ||
                        (causal && key > my_q + diag);
          if (masked) p[ct][r] = -INFINITY;
          pmax = fmaxf(pmax, p[ct][r]);
        }
      pmax = fmaxf(pmax, __shfl_xor(pmax, 32, 64));
      float mn = fmaxf(m_i, pmax);
      float alpha = (m_i == -INFINITY || mn == -INFINITY)
                        ? ((m_i == -INFINITY) ? 0.f : 1.f)
                        : __expf(m_i - mn);
      m_i = mn;
      float psum = 0.f;
#pragma unroll
      for (int ct = 0; ct < 2; ++ct)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          float pv = (p[ct][r] == -INFINITY || mn == -INFINITY)
                         ? 0.f : __expf(p[ct][r] - mn);
          p[ct][r] = pv;
          psum += pv;
        }
      psum += __shfl_xor(psum, 32, 64);
      l_i = l_i * alpha + psum;
      // rescale O^T
#pragma unroll
      for (int dt = 0; dt < 4; ++dt)
#pragma unroll
        for (int r = 0; r < 16; ++r) o_acc[dt][r] *= alpha;

      // ---- P -> bf16 A-fragments (cvt_pk + permlane32_swap), fused with
      // PV so only ONE pfrag is live at a time (register pressure).
      // frag[ks] (ks = key slot of 16): lane needs P[q][16ks + 8hi + j],
      // j = 0..7, as 4 u32 words.  O^T += mfma(Vfrag, Pfrag):
      // Vfrag(dt, ks): lane row d = 32dt + iq, k = key 16ks + 8hi + j via
      // ds_read_b64_tr_b16: each 4-lane cluster reads a [4 key][4 d] tile;
      // lane (4c + j) supplies &V[key_base + j][d_base(cluster)] (8 B,
      // 4 contiguous d), lane (4c + k) receives column k = its own d.
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        bf16x8 pfrag;
        {
          const int ct = ks >> 1;
          const int rb = (ks & 1) * 8;   // regs rb..rb+7 -> keys 16(ks&1)..
          unsigned w0, w1, w2, w3;
          asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2"
                       : "=v"(w0) : "v"(p[ct][rb + 0]), "v"(p[ct][rb + 1]));
          asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2"
                       : "=v"(w2) : "v"(p[ct][rb + 4]), "v"(p[ct][rb + 5]));
          asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2"
                       : "=v"(w1) : "v"(p[ct][rb + 2]), "v"(p[ct][rb + 3]));
          asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2"
                       : "=v"(w3) : "v"(p[ct][rb + 6]), "v"(p[ct][rb + 7]));
          // half exchange: w0 <- keys {+0,+1}, w2 <- keys {+4,+5} (lo half
          // keeps own, hi half receives partner's; header derivation)
          asm volatile("s_nop 1\n\tv_permlane32_swap_b32 %0, %1"
                       : "+v"(w0), "+v"(w2));
          asm volatile("s_nop 1\n\tv_permlane32_swap_b32 %0, %1"
                       : "+v"(w1), "+v"(w3));
          union { unsigned u[4]; bf16x8 v; } pk;
          pk.u[0] = w0; pk.u[1] = w1; pk.u[2] = w2; pk.u[3] = w3;
          pfrag = pk.v;
        }
        // Measured tr_b16 semantics (scripts/probe_tr16): lane 16g+4r+c
        // receives elem j from the address of lane 16g+4j+r, at +2c
        // bytes.  So lane a supplies the key selected by (a>>2)&3 and the
        // d-block selected by a&3; its own received d = dbase + (a&31).
        // voff is linear in the d-subtile index, so all 4 dt fragments
        // sit at static offsets (+2048*dt) from ONE address — issue all
        // 8 tr reads back-to-back under a single lgkmcnt wait instead of
        // 4 serialized round trips (the per-pair wait was the fwd
        // kernel's biggest stall).
        const int keyb = 16 * ks + 8 * hi + ((lane >> 2) & 3);
        const int dbase0 = 16 * ((lane >> 4) & 1) + 4 * (lane & 3);
        // ds ops take a 32-bit LDS offset, not a generic 64-bit pointer
        unsigned a1 = (unsigned)(uintptr_t)(
            (__attribute__((address_space(3))) char*)(v_lds(cur) +
                                                      voff(keyb, dbase0)));
        u32x2 r1[4], r2[4];
        asm volatile(
            "ds_read_b64_tr_b16 %0, %8\n\t"
            "ds_read_b64_tr_b16 %1, %8 offset:128\n\t"
            "ds_read_b64_tr_b16 %2, %8 offset:2048\n\t"
            "ds_read_b64_tr_b16 %3, %8 offset:2176\n\t"
            "ds_read_b64_tr_b16 %4, %8 offset:4096\n\t"
            "ds_read_b64_tr_b16 %5, %8 offset:4224\n\t"
            "ds_read_b64_tr_b16 %6, %8 offset:6144\n\t"
            "ds_read_b64_tr_b16 %7, %8 offset:6272\n\t"
            "s_waitcnt lgkmcnt(0)"
            : "=&v"(r1[0]), "=&v"(r2[0]), "=&v"(r1[1]), "=&v"(r2[1]),
              "=&v"(r1[2]), "=&v"(r2[2]), "=&v"(r1[3]), "=&v"(r2[3])
            : "v"(a1));
        __builtin_amdgcn_sched_barrier(0);
#pragma unroll
        for (int dt = 0; dt < 4; ++dt) {
          union { u32x2 u[2]; bf16x8 v; } vf;
          vf.u[0] = r1[dt]; vf.u[1] = r2[dt];
          o_acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              vf.v, pfrag, o_acc[dt], 0, 0, 0);
        }
      }
    }

  }

  // ---- epilogue -------------------------------------------------------
  if (my_q < S) {
    float inv = (l_i > 0.f) ? 1.f / l_i : 0.f;
    bf16* orow = O + (cu ? (int64_t)row0 * so.rs : (int64_t)b * so.bs)
                 + (int64_t)h * so.hs + (int64_t)my_q * so.rs;
#pragma unroll
    for (int dt = 0; dt < 4; ++dt) {
#pragma unroll
      for (int g = 0; g < 4; ++g) {           // 4 runs of 4 contiguous d
        int d0 = 32 * dt + 8 * g + 4 * hi;    // crow runs: (r&3) + 8*(r>>2)
        unsigned lo, hsp;
        float f0 = o_acc[dt][4 * g + 0] * inv;
        float f1 = o_acc[dt][4 * g + 1] * inv;
        float f2 = o_acc[dt][4 * g + 2] * inv;
        float f3 = o_acc[dt][4 * g + 3] * inv;
        asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2"
                     : "=v"(lo) : "v"(f0), "v"(f1));
        asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2"
                     : "=v"(hsp) : "v"(f2), "v"(f3));
        union { unsigned u[2]; uint2 v; } st;
        st.u[0] = lo; st.u[1] = hsp;
        *reinterpret_cast<uint2*>(orow + d0) = st.v;
      }
    }
    if (hi == 0) {
      int64_t li = cu ? ((int64_t)h * lse_T + row0 + my_q)
                      : ((int64_t)bh * S + my_q);
      LSE[li] = (l_i > 0.f) ? m_i + __logf(l_i) : -INFINITY;
    }
  }
}

}  // namespace

bool fa2_fwd_supported(int D, int S) { return D == 128; }

void fa2_fwd_launch(const void* q, const void* k, const void* v, void* o,
                    float* lse, int B, int H, int Hkv, int S, int Skv,
                    float scale, bool causal, hipStream_t stream,
                    FaStrides sq, FaStrides skv, FaStrides so) {
  dim3 grid(B * H, (S + BM - 1) / BM);
  size_t lds = 2 * NBUF * (size_t)BN * 128 * 2;   // 96 KiB
  hipLaunchKernelGGL(fa2_fwd_kernel<128>, grid, dim3(THREADS), lds, stream,
                     (const bf16*)q, (const bf16*)k, (const bf16*)v,
                     (bf16*)o, lse, B, H, Hkv, S, Skv, scale, causal,
                     sq, skv, so, (const int*)nullptr, 0);
}

// Packed-varlen entry (reference FlashAttention.cu mha_varlen_fwd): ONE
// kernel launch over all ragged segments; q/k/v [T, H, D] packed, cu
// int32 [nseg+1] on DEVICE.  Returns (o [T, H, D], lse [H, T] fp32).
std::vector<torch::Tensor> flash_attn_varlen_fwd(
    torch::Tensor q, torch::Tensor k, torch::Tensor v, torch::Tensor cu,
    int64_t max_seqlen, bool causal, double scale) {
  TORCH_CHECK(q.dim() == 3 && q.is_contiguous() && k.is_contiguous() &&
              v.is_contiguous(), "varlen: q/k/v [T, H, D] contiguous");
  TORCH_CHECK(q.scalar_type() == at::kBFloat16, "varlen: bf16 only");
  TORCH_CHECK(cu.scalar_type() == at::kInt && cu.is_cuda() &&
              cu.is_contiguous(), "varlen: cu int32 on device");
  const int T = q.size(0), H = q.size(1), D = q.size(2);
  const int Hkv = k.size(1);
  TORCH_CHECK(D == 128, "fa2 varlen: D=128 only");
  const int nseg = cu.numel() - 1;
  auto o = torch::empty_like(q);
  auto lse = torch::empty({H, (int64_t)T},
                          q.options().dtype(at::kFloat));
  FaStrides sq{0, (int64_t)D, (int64_t)H * D};
  FaStrides skv{0, (int64_t)D, (int64_t)Hkv * D};
  auto stream = hetu_current_stream();
  dim3 grid(nseg * H, ((int)max_seqlen + BM - 1) / BM);
  size_t lds = 2 * NBUF * (size_t)BN * 128 * 2;
  hipLaunchKernelGGL(fa2_fwd_kernel<128>, grid, dim3(THREADS), lds, stream,
                     (const bf16*)q.data_ptr(), (const bf16*)k.data_ptr(),
                     (const bf16*)v.data_ptr(), (bf16*)o.data_ptr(),
                     lse.data_ptr<float>(), nseg, H, Hkv, 0, 0,
                     (float)scale, causal, sq, skv, sq,
                     cu.data_ptr<int>(), T);
  return {o, lse};
}

// Fused-QKV entry: qkv [B, S, (H+2*Hkv)*D] straight from the column-
// parallel GEMM (q|k|v interleaved per token); no slice/transpose copies.
// Returns (o [B, S, H*D], lse [B, H, S]).
std::vector<torch::Tensor> flash_attn_fwd_qkv(torch::Tensor qkv, int64_t H,
                                              int64_t Hkv, int64_t D,
                                              bool causal, double scale) {
  TORCH_CHECK(qkv.dim() == 3 && qkv.is_contiguous());
  TORCH_CHECK(qkv.scalar_type() == at::kBFloat16, "fa2 qkv: bf16 only");
  TORCH_CHECK(D == 128, "fa2 qkv: D=128 only");
  const int B = qkv.size(0), S = qkv.size(1);
  const int64_t C = (H + 2 * Hkv) * D;
  TORCH_CHECK(qkv.size(2) == C, "qkv width mismatch");
  auto o = torch::empty({B, (int64_t)S, H * D}, qkv.options());
  auto lse = torch::empty({B, H, S}, qkv.options().dtype(at::kFloat));
  auto stream = hetu_current_stream();
  const bf16* base = (const bf16*)qkv.data_ptr();
  FaStrides sqkv{(long long)S * C, (long long)D, (long long)C};
  FaStrides so{(long long)S * H * D, (long long)D, (long long)H * D};
  fa2_fwd_launch(base, base + H * D, base + (H + Hkv) * D, o.data_ptr(),
                 lse.data_ptr<float>(), B, H, Hkv, S, S, (float)scale,
                 causal, stream, sqkv, sqkv, so);
  return {o, lse};
}
