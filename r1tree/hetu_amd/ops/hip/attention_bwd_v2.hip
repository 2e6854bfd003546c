// Flash-attention 2 backward, 8-wave 32x32-MFMA structure for gfx950.
//
// Two atomics-free kernels (the standard FA2 split; reference relied on
// vendored CUTLASS kernels — this is a from-scratch CDNA4 design):
//  * dq kernel: grid over q-blocks (8 waves x 32 q); K/V tiles stream
//    through LDS exactly like the fa2 forward; dS is rebuilt from the saved
//    lse and packed in-register (cvt_pk + permlane32_swap) into MFMA
//    B-fragments; dQ^T accumulates in registers, stores like the fwd O.
//  * dkv kernel: grid over kv-blocks (8 waves x 32 keys, keys exclusive per
//    wave); Q/dO tiles stream through LDS; P^T/dS^T round-trip through tiny
//    per-wave 72B-row LDS images consumed by ds_read_b64_tr_b16; dK/dV
//    accumulate in registers; plain stores (atomicAdd only under GQA).
// Every layout primitive (kswz glds staging, tr_b16 address formulas, the
// permlane pack, mfma fragment maps) is validated by scripts/probe_fa2.hip.
#include <torch/extension.h>
#include "ext_stream.h"
#include "common.h"

namespace {

using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x16 = __attribute__((ext_vector_type(16))) float;
using u32x2 = __attribute__((ext_vector_type(2))) unsigned int;

constexpr int THREADS = 512;

DEV int kswz_row(int row, int byte_in_row) {
  return row * 256 + (byte_in_row ^ ((row & 15) << 4));
}

// pack p[16] (C layout: col=q lane-local, rows=key crow) into a bf16
// A/B-fragment with own-index q, k = 8 keys starting at 8*(half) within the
// 16-key slice given by register quads rb..rb+7 (see fa2 fwd derivation).
#define PACK_FRAG(dst, p, rb)                                              \
  do {                                                                     \
    unsigned w0, w1, w2, w3;                                               \
    asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2"                            \
                 : "=v"(w0) : "v"(p[(rb) + 0]), "v"(p[(rb) + 1]));         \
    asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2"                            \
                 : "=v"(w2) : "v"(p[(rb) + 4]), "v"(p[(rb) + 5]));         \
    asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2"                            \
                 : "=v"(w1) : "v"(p[(rb) + 2]), "v"(p[(rb) + 3]));         \
    asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2"                            \
                 : "=v"(w3) : "v"(p[(rb) + 6]), "v"(p[(rb) + 7]));         \
    asm volatile("s_nop 1\n\tv_permlane32_swap_b32 %0, %1"                 \
                 : "+v"(w0), "+v"(w2));                                    \
    asm volatile("s_nop 1\n\tv_permlane32_swap_b32 %0, %1"                 \
                 : "+v"(w1), "+v"(w3));                                    \
    union { unsigned u[4]; bf16x8 v; } pk_;                                \
    pk_.u[0] = w0; pk_.u[1] = w1; pk_.u[2] = w2; pk_.u[3] = w3;            \
    dst = pk_.v;                                                           \
  } while (0)

// tr_b16 read of an A/B fragment (own = column dim, k = 8 rows at
// rowbase + 8*hi) from a swizzled row-major [rows][256 B] image.
#define TR_RM(dst, base_lds, rowb, colbyte)                                \
  do {                                                                     \
    int r1_ = (rowb) + ((lane >> 2) & 3);                                  \
    int r2_ = r1_ + 4;                                                     \
    unsigned a1_ = (unsigned)(uintptr_t)(                                  \
        (__attribute__((address_space(3))) char*)(                         \
            (base_lds) + r1_ * 256 + ((colbyte) ^ ((r1_ & 15) << 4))));    \
    unsigned a2_ = (unsigned)(uintptr_t)(                                  \
        (__attribute__((address_space(3))) char*)(                         \
            (base_lds) + r2_ * 256 + ((colbyte) ^ ((r2_ & 15) << 4))));    \
    u32x2 x1_, x2_;                                                        \
    asm volatile("ds_read_b64_tr_b16 %0, %2\n\t"                           \
                 "ds_read_b64_tr_b16 %1, %3\n\t"                           \
                 "s_waitcnt lgkmcnt(0)"                                    \
                 : "=&v"(x1_), "=&v"(x2_) : "v"(a1_), "v"(a2_));           \
    __builtin_amdgcn_sched_barrier(0);                                     \
    union { u32x2 u[2]; bf16x8 v; } f_;                                    \
    f_.u[0] = x1_; f_.u[1] = x2_;                                          \
    dst = f_.v;                                                            \
  } while (0)

// batched TR_RM: all 4 dt fragments (colbyte cb0 + 64*dt) of the same
// row pair in ONE asm block — 8 tr reads issued back-to-back under a
// single lgkmcnt wait instead of 4 serialized round trips.
#define TR_RM4(d0, d1, d2, d3, base_lds, rowb, cb0)                        \
  do {                                                                     \
    int r1_ = (rowb) + ((lane >> 2) & 3);                                  \
    int r2_ = r1_ + 4;                                                     \
    unsigned b1_ = (unsigned)(uintptr_t)(                                  \
        (__attribute__((address_space(3))) char*)((base_lds) + r1_ * 256));\
    unsigned b2_ = (unsigned)(uintptr_t)(                                  \
        (__attribute__((address_space(3))) char*)((base_lds) + r2_ * 256));\
    const int m1_ = (r1_ & 15) << 4, m2_ = (r2_ & 15) << 4;                \
    u32x2 x_[8];                                                           \
    asm volatile("ds_read_b64_tr_b16 %0, %8\n\t"                           \
                 "ds_read_b64_tr_b16 %1, %9\n\t"                           \
                 "ds_read_b64_tr_b16 %2, %10\n\t"                          \
                 "ds_read_b64_tr_b16 %3, %11\n\t"                          \
                 "ds_read_b64_tr_b16 %4, %12\n\t"                          \
                 "ds_read_b64_tr_b16 %5, %13\n\t"                          \
                 "ds_read_b64_tr_b16 %6, %14\n\t"                          \
                 "ds_read_b64_tr_b16 %7, %15\n\t"                          \
                 "s_waitcnt lgkmcnt(0)"                                    \
                 : "=&v"(x_[0]), "=&v"(x_[1]), "=&v"(x_[2]), "=&v"(x_[3]), \
                   "=&v"(x_[4]), "=&v"(x_[5]), "=&v"(x_[6]), "=&v"(x_[7])  \
                 : "v"(b1_ + (((cb0) + 0) ^ m1_)),                         \
                   "v"(b2_ + (((cb0) + 0) ^ m2_)),                         \
                   "v"(b1_ + (((cb0) + 64) ^ m1_)),                        \
                   "v"(b2_ + (((cb0) + 64) ^ m2_)),                        \
                   "v"(b1_ + (((cb0) + 128) ^ m1_)),                       \
                   "v"(b2_ + (((cb0) + 128) ^ m2_)),                       \
                   "v"(b1_ + (((cb0) + 192) ^ m1_)),                       \
                   "v"(b2_ + (((cb0) + 192) ^ m2_)));                      \
    __builtin_amdgcn_sched_barrier(0);                                     \
    union { u32x2 u[2]; bf16x8 v; } f_;                                    \
    f_.u[0] = x_[0]; f_.u[1] = x_[1]; d0 = f_.v;                           \
    f_.u[0] = x_[2]; f_.u[1] = x_[3]; d1 = f_.v;                           \
    f_.u[0] = x_[4]; f_.u[1] = x_[5]; d2 = f_.v;                           \
    f_.u[0] = x_[6]; f_.u[1] = x_[7]; d3 = f_.v;                           \
  } while (0)

// tr_b16 read from a 72B-row [32 q][32 key] P'/dS' image (own=key, k=q).
#define TR_P(dst, base_lds, qb, keybyte)                                   \
  do {                                                                     \
    int r1_ = (qb) + ((lane >> 2) & 3);                                    \
    unsigned a1_ = (unsigned)(uintptr_t)(                                  \
        (__attribute__((address_space(3))) char*)((base_lds) + r1_ * 72 +  \
                                                  (keybyte)));             \
    u32x2 x1_, x2_;                                                        \
    asm volatile("ds_read_b64_tr_b16 %0, %2\n\t"                           \
                 "ds_read_b64_tr_b16 %1, %2 offset:288\n\t"                \
                 "s_waitcnt lgkmcnt(0)"                                    \
                 : "=&v"(x1_), "=&v"(x2_) : "v"(a1_));                     \
    __builtin_amdgcn_sched_barrier(0);                                     \
    union { u32x2 u[2]; bf16x8 v; } f_;                                    \
    f_.u[0] = x1_; f_.u[1] = x2_;                                          \
    dst = f_.v;                                                            \
  } while (0)

// ===========================================================================
// dq kernel: 8 waves x 32 q rows; loops over 64-key KV tiles.
// ===========================================================================
template <int D>
__global__ __launch_bounds__(THREADS, 2) void fa2_bwd_dq_kernel(
    const bf16* __restrict__ Q, const bf16* __restrict__ K,
    const bf16* __restrict__ V, const bf16* __restrict__ dO,
    const float* __restrict__ LSE, const float* __restrict__ DELTA,
    bf16* __restrict__ dQ, int B, int H, int Hkv, int S, int Skv,
    float scale, bool causal, FaStrides sq, FaStrides skv, FaStrides sdo,
    FaStrides sdq) {
  static_assert(D == 128);
  constexpr int KB = 64 * 256;     // 16 KiB per rm image
  extern __shared__ __attribute__((aligned(16))) char smem[];
  auto k_lds = [&](int b) -> char* { return smem + b * KB; };
  auto v_lds = [&](int b) -> char* { return smem + (2 + b) * KB; };

  // (bh, q-block) grid: spreads causal depths across CUs (see fwd note)
  const int bh = blockIdx.x;
  const int h = bh % H;
  const int b = bh / H;
  const int hkv = h / (H / Hkv);
  const int q0 = blockIdx.y * 256;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int iq = lane & 31;
  const int hi = lane >> 5;
  const int g1 = (lane >> 4) & 1;

  const bf16* Qb = Q + (int64_t)b * sq.bs + (int64_t)h * sq.hs;
  const bf16* dOb = dO + (int64_t)b * sdo.bs + (int64_t)h * sdo.hs;
  const bf16* Kb = K + (int64_t)b * skv.bs + (int64_t)hkv * skv.hs;
  const bf16* Vb = V + (int64_t)b * skv.bs + (int64_t)hkv * skv.hs;

  const int my_q = q0 + wid * 32 + iq;
  const int diag = Skv - S;
  const float lse_q = LSE[(int64_t)bh * S + min(my_q, S - 1)];
  const float del_q = DELTA[(int64_t)bh * S + min(my_q, S - 1)];

  // Q (scaled) and dO in registers
  bf16x8 qreg[8], doreg[8];
  {
    const bf16* qrow = Qb + (int64_t)min(my_q, S - 1) * sq.rs;
    const bf16* drow = dOb + (int64_t)min(my_q, S - 1) * sdo.rs;
#pragma unroll
    for (int kk = 0; kk < 8; ++kk) {
      ushort8 uq = *reinterpret_cast<const ushort8*>(qrow + kk * 16 + hi * 8);
      ushort8 ud = *reinterpret_cast<const ushort8*>(drow + kk * 16 + hi * 8);
      union { ushort8 us; bf16x8 v; } cq, cd;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        cq.us.v[j] = f2bf(bf2f(uq.v[j]) * scale);
        cd.us.v[j] = ud.v[j];
      }
      qreg[kk] = cq.v;
      doreg[kk] = cd.v;
    }
  }

  const int wlane16 = lane * 16;
#define DQ_GLDS(k0, buf)                                                    \
  do {                                                                      \
    _Pragma("unroll")                                                       \
    for (int i = 0; i < 2; ++i) {                                           \
      int pos = (wid * 2 + i) * 1024 + wlane16;                             \
      int krow = pos >> 8;                                                  \
      int kd = ((pos & 255) ^ ((krow & 15) << 4)) >> 1;                     \
      int64_t roff = (int64_t)min((k0) + krow, Skv - 1) * skv.rs + kd;      \
      __builtin_amdgcn_global_load_lds(                                     \
          (const __attribute__((address_space(1))) void*)(Kb + roff),       \
          (__attribute__((address_space(3))) void*)(k_lds(buf) + pos),      \
          16, 0, 0);                                                        \
      __builtin_amdgcn_global_load_lds(                                     \
          (const __attribute__((address_space(1))) void*)(Vb + roff),       \
          (__attribute__((address_space(3))) void*)(v_lds(buf) + pos),      \
          16, 0, 0);                                                        \
    }                                                                       \
  } while (0)

  f32x16 dq_acc[4];
#pragma unroll
  for (int dt = 0; dt < 4; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r) dq_acc[dt][r] = 0.f;

  const int wave_qmax = q0 + wid * 32 + 31;
  int n_tiles = (Skv + 63) / 64;
  if (causal) n_tiles = min(n_tiles, max((q0 + 256 + diag + 63) / 64, 1));

  DQ_GLDS(0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  for (int t = 0; t < n_tiles; ++t) {
    const int k0 = t * 64;
    const int cur = t & 1;
    const bool have_next = (t + 1) < n_tiles;
    if (have_next) DQ_GLDS(k0 + 64, cur ^ 1);

    const bool active = !causal || (k0 <= wave_qmax + diag);
    if (active) {
      float ds[2][16];
      // ---- S^T and dP^T ----------------------------------------------
#pragma unroll
      for (int ct = 0; ct < 2; ++ct) {
        f32x16 sa, da;
#pragma unroll
        for (int r = 0; r < 16; ++r) { sa[r] = 0.f; da[r] = 0.f; }
#pragma unroll
        for (int kk = 0; kk < 8; ++kk) {
          bf16x8 kf = *reinterpret_cast<const bf16x8*>(
              k_lds(cur) + kswz_row(32 * ct + iq, kk * 32 + hi * 16));
          bf16x8 vf = *reinterpret_cast<const bf16x8*>(
              v_lds(cur) + kswz_row(32 * ct + iq, kk * 32 + hi * 16));
          sa = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qreg[kk], sa,
                                                       0, 0, 0);
          da = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, doreg[kk], da,
                                                       0, 0, 0);
        }
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int key = k0 + 32 * ct + (r & 3) + 8 * (r >> 2) + 4 * hi;
          bool masked = (key >= Skv) || (my_q >= S) ||
                        (causal && key > my_q + diag);
          float pv = masked ? 0.f : __expf(sa[r] - lse_q);
          ds[ct][r] = masked ? 0.f : pv * (da[r] - del_q);
        }
      }
      // ---- dQ^T += K^T dS --------------------------------------------
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        bf16x8 dsf;
        PACK_FRAG(dsf, ds[ks >> 1], (ks & 1) * 8);
        bf16x8 kt0, kt1, kt2, kt3;
        TR_RM4(kt0, kt1, kt2, kt3, k_lds(cur), 16 * ks + 8 * hi,
               (16 * g1 + 4 * (lane & 3)) * 2);
        dq_acc[0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            kt0, dsf, dq_acc[0], 0, 0, 0);
        dq_acc[1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            kt1, dsf, dq_acc[1], 0, 0, 0);
        dq_acc[2] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            kt2, dsf, dq_acc[2], 0, 0, 0);
        dq_acc[3] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            kt3, dsf, dq_acc[3], 0, 0, 0);
      }
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  // ---- epilogue: dQ[my_q][d] = scale * dq^T[d][my_q] -------------------
  if (my_q < S) {
    bf16* qrow = dQ + (int64_t)b * sdq.bs + (int64_t)h * sdq.hs
               + (int64_t)my_q * sdq.rs;
#pragma unroll
    for (int dt = 0; dt < 4; ++dt) {
#pragma unroll
      for (int gq = 0; gq < 4; ++gq) {
        int d0 = 32 * dt + 8 * gq + 4 * hi;
        unsigned lo, hs;
        float f0 = dq_acc[dt][4 * gq + 0] * scale;
        float f1 = dq_acc[dt][4 * gq + 1] * scale;
        float f2 = dq_acc[dt][4 * gq + 2] * scale;
        float f3 = dq_acc[dt][4 * gq + 3] * scale;
        asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2"
                     : "=v"(lo) : "v"(f0), "v"(f1));
        asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2"
                     : "=v"(hs) : "v"(f2), "v"(f3));
        union { unsigned u[2]; uint2 v; } st;
        st.u[0] = lo; st.u[1] = hs;
        *reinterpret_cast<uint2*>(qrow + d0) = st.v;
      }
    }
  }
#undef DQ_GLDS
}

// ===========================================================================
// dkv kernel: 8 waves x 32 keys (exclusive); loops over 32-q tiles.
// ===========================================================================
template <int D>
// 132 KiB LDS -> occupancy 1 by LDS; see the fwd note on launch bounds
__global__ __launch_bounds__(THREADS, 1) void fa2_bwd_dkv_kernel(
    const bf16* __restrict__ Q, const bf16* __restrict__ K,
    const bf16* __restrict__ V, const bf16* __restrict__ dO,
    const float* __restrict__ LSE, const float* __restrict__ DELTA,
    float* __restrict__ dK32, float* __restrict__ dV32,
    bf16* __restrict__ dK16, bf16* __restrict__ dV16,
    int B, int H, int Hkv, int S, int Skv, float scale, bool causal,
    FaStrides sq, FaStrides skv, FaStrides sdo, FaStrides sdkv) {
  static_assert(D == 128);
  constexpr int QB = 32 * 256;     // 8 KiB per rm image
  extern __shared__ __attribute__((aligned(16))) char smem[];
  auto q_lds = [&](int b) -> char* { return smem + b * QB; };
  auto do_lds = [&](int b) -> char* { return smem + (2 + b) * QB; };
  char* vw_base = smem + 4 * QB;   // per-wave V image, 8 KiB each
  char* pw_base = smem + 4 * QB + 8 * 8192;   // per-wave P'/dS' (2 x 2304 B)

  // (bh, kv-block) grid: spreads causal depths across CUs (see fwd note)
  const int bh = blockIdx.x;
  const int h = bh % H;
  const int b = bh / H;
  const int hkv = h / (H / Hkv);
  const int kb0 = blockIdx.y * 256;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int iq = lane & 31;        // here: own q-col index AND own key row
  const int hi = lane >> 5;
  const int g1 = (lane >> 4) & 1;

  const bf16* Qb = Q + (int64_t)b * sq.bs + (int64_t)h * sq.hs;
  const bf16* dOb = dO + (int64_t)b * sdo.bs + (int64_t)h * sdo.hs;
  const bf16* Kb = K + (int64_t)b * skv.bs + (int64_t)hkv * skv.hs;
  const bf16* Vb = V + (int64_t)b * skv.bs + (int64_t)hkv * skv.hs;
  const float* lse_b = LSE + (int64_t)bh * S;
  const float* del_b = DELTA + (int64_t)bh * S;

  const int my_key = kb0 + wid * 32 + iq;      // lane's exclusive key row
  const int diag = Skv - S;

  char* vw_lds = vw_base + wid * 8192;
  char* p_lds = pw_base + wid * 4608;
  char* ds_lds = p_lds + 2304;

  // K (scaled) in registers; V in a per-wave swizzled row-major LDS image
  // (keeps the VGPR budget at 2 waves/SIMD without spills)
  bf16x8 kreg[8];
  {
    const bf16* krow = Kb + (int64_t)min(my_key, Skv - 1) * skv.rs;
#pragma unroll
    for (int kk = 0; kk < 8; ++kk) {
      ushort8 uk = *reinterpret_cast<const ushort8*>(krow + kk * 16 + hi * 8);
      union { ushort8 us; bf16x8 v; } ck;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        ck.us.v[j] = f2bf(bf2f(uk.v[j]) * scale);
      kreg[kk] = ck.v;
    }
  }
#pragma unroll
  for (int i = 0; i < 8; ++i) {     // stage wave's V rows (8 KiB, glds)
    int pos = (i * 64 + lane) * 16;
    int vrow = pos >> 8;
    int vd = ((pos & 255) ^ ((vrow & 15) << 4)) >> 1;
    const bf16* vsrc = Vb
        + (int64_t)min(kb0 + wid * 32 + vrow, Skv - 1) * skv.rs + vd;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)vsrc,
        (__attribute__((address_space(3))) void*)(vw_lds + pos), 16, 0, 0);
  }

  const int wlane16 = lane * 16;
  // per tile: Qrm 8 KiB + dOrm 8 KiB; 512 lanes x 16 B covers 8 KiB, so one
  // glds per lane per image.
#define DKV_GLDS(qt0, buf)                                                  \
  do {                                                                      \
    int pos = tid * 16;                                                     \
    int qrow = pos >> 8;                                                    \
    int qd = ((pos & 255) ^ ((qrow & 15) << 4)) >> 1;                       \
    int qr_ = min((qt0) + qrow, S - 1);                                     \
    __builtin_amdgcn_global_load_lds(                                       \
        (const __attribute__((address_space(1))) void*)(                    \
            Qb + (int64_t)qr_ * sq.rs + qd),                                \
        (__attribute__((address_space(3))) void*)(q_lds(buf) + pos),        \
        16, 0, 0);                                                          \
    __builtin_amdgcn_global_load_lds(                                       \
        (const __attribute__((address_space(1))) void*)(                    \
            dOb + (int64_t)qr_ * sdo.rs + qd),                              \
        (__attribute__((address_space(3))) void*)(do_lds(buf) + pos),       \
        16, 0, 0);                                                          \
  } while (0)

  f32x16 dk_acc[4], dv_acc[4];
#pragma unroll
  for (int dt = 0; dt < 4; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r) { dk_acc[dt][r] = 0.f; dv_acc[dt][r] = 0.f; }

  const int n_q_tiles = (S + 31) / 32;
  int t_start = 0;
  if (causal) t_start = max(0, (kb0 - diag) / 32);
  const int wave_kmin = kb0 + wid * 32;        // first key this wave owns

  DKV_GLDS(t_start * 32, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  for (int t = t_start; t < n_q_tiles; ++t) {
    const int qt0 = t * 32;
    const int cur = (t - t_start) & 1;
    const bool have_next = (t + 1) < n_q_tiles;
    if (have_next) DKV_GLDS(qt0 + 32, cur ^ 1);

    // wave active if any of its keys can see this q tile
    const bool active = !causal || (qt0 + 31 >= wave_kmin - diag);
    if (active) {
      const int my_q = qt0 + iq;               // lane's q column
      const float lse_q = lse_b[min(my_q, S - 1)];
      const float del_q = del_b[min(my_q, S - 1)];
      // ---- S^T = K_s Q^T ; dP^T = V dO^T ------------------------------
      f32x16 sa, da;
#pragma unroll
      for (int r = 0; r < 16; ++r) { sa[r] = 0.f; da[r] = 0.f; }
#pragma unroll
      for (int kk = 0; kk < 8; ++kk) {
        bf16x8 qf = *reinterpret_cast<const bf16x8*>(
            q_lds(cur) + kswz_row(iq, kk * 32 + hi * 16));
        bf16x8 df = *reinterpret_cast<const bf16x8*>(
            do_lds(cur) + kswz_row(iq, kk * 32 + hi * 16));
        bf16x8 vf = *reinterpret_cast<const bf16x8*>(
            vw_lds + kswz_row(iq, kk * 32 + hi * 16));
        sa = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kreg[kk], qf, sa,
                                                     0, 0, 0);
        da = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, df, da,
                                                     0, 0, 0);
      }
      // ---- P^T, dS^T; write the per-wave 72B-row images ---------------
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int key = kb0 + wid * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        bool masked = (key >= Skv) || (my_q >= S) ||
                      (causal && key > my_q + diag);
        float pv = masked ? 0.f : __expf(sa[r] - lse_q);
        sa[r] = pv;                             // P^T
        da[r] = masked ? 0.f : pv * (da[r] - del_q);   // dS^T (no scale)
      }
#pragma unroll
      for (int rq = 0; rq < 4; ++rq) {          // 4 key-quads
        int keyb = (8 * rq + 4 * hi) & 31;      // local key of quad start
        unsigned pw0, pw1, dw0, dw1;
        asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2"
                     : "=v"(pw0) : "v"(sa[4 * rq + 0]), "v"(sa[4 * rq + 1]));
        asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2"
                     : "=v"(pw1) : "v"(sa[4 * rq + 2]), "v"(sa[4 * rq + 3]));
        asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2"
                     : "=v"(dw0) : "v"(da[4 * rq + 0]), "v"(da[4 * rq + 1]));
        asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2"
                     : "=v"(dw1) : "v"(da[4 * rq + 2]), "v"(da[4 * rq + 3]));
        union { unsigned u[2]; uint2 v; } sp, sd;
        sp.u[0] = pw0; sp.u[1] = pw1;
        sd.u[0] = dw0; sd.u[1] = dw1;
        // row = q (iq), cols = keys keyb..keyb+3 (8 B)
        *reinterpret_cast<uint2*>(p_lds + iq * 72 + keyb * 2) = sp.v;
        *reinterpret_cast<uint2*>(ds_lds + iq * 72 + keyb * 2) = sd.v;
      }
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

      // ---- dV += P^T dO ; dK += dS^T Q --------------------------------
#pragma unroll
      for (int s = 0; s < 2; ++s) {             // k = q slices of 16
        bf16x8 ptf, dstf;
        TR_P(ptf, p_lds, 16 * s + 8 * hi, (16 * g1 + 4 * (lane & 3)) * 2);
        TR_P(dstf, ds_lds, 16 * s + 8 * hi, (16 * g1 + 4 * (lane & 3)) * 2);
        bf16x8 do0, do1, do2, do3, qa0, qa1, qa2, qa3;
        TR_RM4(do0, do1, do2, do3, do_lds(cur), 16 * s + 8 * hi,
               (16 * g1 + 4 * (lane & 3)) * 2);
        TR_RM4(qa0, qa1, qa2, qa3, q_lds(cur), 16 * s + 8 * hi,
               (16 * g1 + 4 * (lane & 3)) * 2);
        dv_acc[0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            ptf, do0, dv_acc[0], 0, 0, 0);
        dk_acc[0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            dstf, qa0, dk_acc[0], 0, 0, 0);
        dv_acc[1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            ptf, do1, dv_acc[1], 0, 0, 0);
        dk_acc[1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            dstf, qa1, dk_acc[1], 0, 0, 0);
        dv_acc[2] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            ptf, do2, dv_acc[2], 0, 0, 0);
        dk_acc[2] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            dstf, qa2, dk_acc[2], 0, 0, 0);
        dv_acc[3] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            ptf, do3, dv_acc[3], 0, 0, 0);
        dk_acc[3] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            dstf, qa3, dk_acc[3], 0, 0, 0);
      }
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  // ---- epilogue: dK = scale * acc; plain stores unless GQA -------------
  const bool gqa = (H != Hkv);
#pragma unroll
  for (int dt = 0; dt < 4; ++dt) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int key = kb0 + wid * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
      if (key < Skv) {
        float dkv_ = dk_acc[dt][r] * scale;
        float dvv_ = dv_acc[dt][r];
        if (gqa) {
          // f32 accumulation buffers stay BHSD-contiguous
          int64_t off32 = ((int64_t)(b * Hkv + hkv) * Skv + key) * D
                        + 32 * dt + iq;
          atomicAdd(dK32 + off32, dkv_);
          atomicAdd(dV32 + off32, dvv_);
        } else {
          int64_t off = (int64_t)b * sdkv.bs + (int64_t)hkv * sdkv.hs
                      + (int64_t)key * sdkv.rs + 32 * dt + iq;
          dK16[off] = (bf16)dkv_;
          dV16[off] = (bf16)dvv_;
        }
      }
    }
  }
#undef DKV_GLDS
}

// delta = rowsum(dO * O) — reuse pattern from v1
__global__ void fa2_delta_kernel(const bf16* __restrict__ dO,
                                 const bf16* __restrict__ O,
                                 float* __restrict__ delta,
                                 int64_t rows, int D, int H, int S,
                                 FaStrides so) {
  // delta rows are LSE-ordered (b, h, s); dO/O may be BHSD or BS[HD]
  // D == 128 fast path: 4 lanes per row (32 elems each, 16B vector
  // loads), quad shuffle reduction — pure bandwidth (the old one-block-
  // per-row version left 240/256 lanes idle: 580us vs the ~65us bound).
  const int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (D == 128) {
    int64_t row = tid >> 2;
    const int q = tid & 3;
    const int64_t rstride = ((int64_t)gridDim.x * blockDim.x) >> 2;
    for (; row < rows; row += rstride) {
      const int64_t b = row / ((int64_t)H * S);
      const int64_t hh = (row / S) % H;
      const int64_t ss = row % S;
      const int64_t base = b * so.bs + hh * so.hs + ss * so.rs + q * 32;
      const bf16* a = dO + base;
      const bf16* bb = O + base;
      float s = 0.f;
#pragma unroll
      for (int i = 0; i < 32; i += 8) {
        float av[8], bv[8];
        VecIO<bf16>::load(a + i, av);
        VecIO<bf16>::load(bb + i, bv);
#pragma unroll
        for (int j = 0; j < 8; ++j) s += av[j] * bv[j];
      }
      s += __shfl_xor(s, 1);
      s += __shfl_xor(s, 2);
      if (q == 0) delta[row] = s;
    }
    return;
  }
  // generic fallback: one lane per row, scalar loads
  for (int64_t row = tid; row < rows;
       row += (int64_t)gridDim.x * blockDim.x) {
    const int64_t b = row / ((int64_t)H * S);
    const int64_t hh = (row / S) % H;
    const int64_t ss = row % S;
    const int64_t base = b * so.bs + hh * so.hs + ss * so.rs;
    const bf16* a = dO + base;
    const bf16* bb = O + base;
    float s = 0.f;
    for (int i = 0; i < D; ++i) s += (float)a[i] * (float)bb[i];
    delta[row] = s;
  }
}

}  // namespace

bool fa2_bwd_supported(int D, int S) { return D == 128; }

namespace {
// shared core over arbitrary layouts
std::vector<torch::Tensor> fa2_bwd_core(
    const bf16* doutp, const bf16* qp, const bf16* kp, const bf16* vp,
    const bf16* outp, torch::Tensor lse, int B, int H, int Hkv, int S,
    int Skv, int D, bool causal, float scale, FaStrides sq, FaStrides skv,
    FaStrides sdo, bf16* dqp, bf16* dk16p, bf16* dv16p, float* dk32p,
    float* dv32p, FaStrides sdq, FaStrides sdkv, bool gqa) {
  auto stream = hetu_current_stream();
  auto delta = torch::empty({B, H, S}, lse.options());
  {
    int64_t rows = (int64_t)B * H * S;
    int grid = (int)std::min<int64_t>((rows * 4 + 255) / 256, 16384);
    hipLaunchKernelGGL(fa2_delta_kernel, dim3(grid), dim3(256), 0, stream,
                       doutp, outp, delta.data_ptr<float>(), rows, D, H, S,
                       sdo);
  }
  {
    dim3 grid(B * H, (S + 255) / 256);
    size_t lds = 4 * (size_t)64 * 256;          // 64 KiB
    hipLaunchKernelGGL(fa2_bwd_dq_kernel<128>, grid, dim3(THREADS), lds,
                       stream, qp, kp, vp, doutp, lse.data_ptr<float>(),
                       delta.data_ptr<float>(), dqp,
                       B, H, Hkv, S, Skv, scale, causal, sq, skv, sdo, sdq);
  }
  {
    dim3 grid(B * H, (Skv + 255) / 256);
    size_t lds = 4 * (size_t)32 * 256 + 8 * 8192 + 8 * 4608;
    hipLaunchKernelGGL(fa2_bwd_dkv_kernel<128>, grid, dim3(THREADS), lds,
                       stream, qp, kp, vp, doutp, lse.data_ptr<float>(),
                       delta.data_ptr<float>(), dk32p, dv32p, dk16p, dv16p,
                       B, H, Hkv, S, Skv, scale, causal, sq, skv, sdo,
                       sdkv);
  }
  return {};
}
}  // namespace

std::vector<torch::Tensor> fa2_bwd_launch(
    torch::Tensor dout, torch::Tensor q, torch::Tensor k, torch::Tensor v,
    torch::Tensor out, torch::Tensor lse, bool causal, double scale) {
  const int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  const int Hkv = k.size(1), Skv = k.size(2);
  auto dq = torch::empty_like(q);
  const bool gqa = (H != Hkv);
  torch::Tensor dk16, dv16, dk32, dv32;
  if (gqa) {
    dk32 = torch::zeros_like(k, k.options().dtype(at::kFloat));
    dv32 = torch::zeros_like(v, v.options().dtype(at::kFloat));
  } else {
    dk16 = torch::empty_like(k);
    dv16 = torch::empty_like(v);
  }
  FaStrides sq{(long long)H * S * D, (long long)S * D, (long long)D};
  FaStrides skv{(long long)Hkv * Skv * D, (long long)Skv * D,
                (long long)D};
  fa2_bwd_core((const bf16*)dout.data_ptr(), (const bf16*)q.data_ptr(),
               (const bf16*)k.data_ptr(), (const bf16*)v.data_ptr(),
               (const bf16*)out.data_ptr(), lse, B, H, Hkv, S, Skv, D,
               causal, (float)scale, sq, skv, sq,
               (bf16*)dq.data_ptr(),
               gqa ? nullptr : (bf16*)dk16.data_ptr(),
               gqa ? nullptr : (bf16*)dv16.data_ptr(),
               gqa ? dk32.data_ptr<float>() : nullptr,
               gqa ? dv32.data_ptr<float>() : nullptr, sq, skv, gqa);
  if (gqa) {
    return {dq, dk32.to(k.scalar_type()), dv32.to(v.scalar_type())};
  }
  return {dq, dk16, dv16};
}

// Fused-QKV backward: dout/out [B,S,H*D], qkv [B,S,(H+2Hkv)*D] ->
// dqkv [B,S,(H+2Hkv)*D] written in place by the kernels (no slice or
// transpose copies).
torch::Tensor flash_attn_bwd_qkv(torch::Tensor dout, torch::Tensor qkv,
                                 torch::Tensor out, torch::Tensor lse,
                                 int64_t H, int64_t Hkv, int64_t D,
                                 bool causal, double scale) {
  TORCH_CHECK(qkv.dim() == 3 && qkv.is_contiguous());
  TORCH_CHECK(dout.is_contiguous() && out.is_contiguous());
  const int B = qkv.size(0), S = qkv.size(1);
  const int64_t C = (H + 2 * Hkv) * D;
  const bool gqa = (H != Hkv);
  auto dqkv = torch::empty_like(qkv);
  torch::Tensor dk32, dv32;
  float *dk32p = nullptr, *dv32p = nullptr;
  if (gqa) {
    dk32 = torch::zeros({B, Hkv, (int64_t)S, D},
                        qkv.options().dtype(at::kFloat));
    dv32 = torch::zeros_like(dk32);
    dk32p = dk32.data_ptr<float>();
    dv32p = dv32.data_ptr<float>();
  }
  FaStrides sqkv{(long long)S * C, (long long)D, (long long)C};
  FaStrides so{(long long)S * H * D, (long long)D, (long long)H * D};
  const bf16* base = (const bf16*)qkv.data_ptr();
  bf16* dbase = (bf16*)dqkv.data_ptr();
  fa2_bwd_core((const bf16*)dout.data_ptr(), base, base + H * D,
               base + (H + Hkv) * D, (const bf16*)out.data_ptr(), lse,
               B, H, Hkv, S, S, D, causal, (float)scale, sqkv, sqkv, so,
               dbase,
               gqa ? nullptr : dbase + H * D,
               gqa ? nullptr : dbase + (H + Hkv) * D,
               dk32p, dv32p, sqkv, sqkv, gqa);
  if (gqa) {
    // scatter the fp32 accumulators into the k/v sections of dqkv
    auto dkv_view = dqkv.view({B, (int64_t)S, H + 2 * Hkv, D});
    dkv_view.narrow(2, H, Hkv).copy_(
        dk32.permute({0, 2, 1, 3}).to(qkv.scalar_type()));
    dkv_view.narrow(2, H + Hkv, Hkv).copy_(
        dv32.permute({0, 2, 1, 3}).to(qkv.scalar_type()));
  }
  return dqkv;
}
