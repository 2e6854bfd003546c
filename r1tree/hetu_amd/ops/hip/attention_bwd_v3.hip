// Flash-attention backward v3 (EXPERIMENTAL, round-2 candidate — compiled
// but not wired; A/B via the flash_attn_bwd_v3 binding).
//
// Same math/layouts as attention_bwd_v2.hip with the forward's proven
// pipeline upgrades applied mechanically:
//   B1. 3-deep staging ring with counted s_waitcnt vmcnt(N) at the loop
//       head (v2 drains vmcnt(0) every tile — the same change took the
//       causal fwd 325 -> 494 TF/s together with the grid order).
//   B2. walking-pointer glds source addressing with a clamped fallback
//       for the ragged tail (v2 recomputes min()*stride per tile).
//   B3. launch_bounds occupancy 1: at 96/148 KiB LDS only one block fits
//       per CU anyway, so the 2-block register budget of v2 only forces
//       spills (v2's own dkv could take the same change — round 2).
// Non-GQA (H == Hkv), BHSD contiguous, D=128 only.
#include <torch/extension.h>
#include "ext_stream.h"
#include "common.h"

namespace {

using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x16 = __attribute__((ext_vector_type(16))) float;
using u32x2 = __attribute__((ext_vector_type(2))) unsigned int;

constexpr int THREADS = 512;
constexpr int NBUF = 3;

DEV int kswz_row3(int row, int byte_in_row) {
  return row * 256 + (byte_in_row ^ ((row & 15) << 4));
}

#define PACK_FRAG3(dst, p, rb)                                             \
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

#define TR_RM4_3(d0, d1, d2, d3, base_lds, rowb, cb0)                      \
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

#define TR_P3(dst, base_lds, qb, keybyte)                                  \
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
// dq kernel v3: 8 waves x 32 q rows; 3-ring KV staging, counted vmcnt.
// ===========================================================================
template <int D>
__global__ __launch_bounds__(THREADS, 1) void fa3_bwd_dq_kernel(
    const bf16* __restrict__ Q, const bf16* __restrict__ K,
    const bf16* __restrict__ V, const bf16* __restrict__ dO,
    const float* __restrict__ LSE, const float* __restrict__ DELTA,
    bf16* __restrict__ dQ, int B, int H, int S, int Skv,
    float scale, bool causal) {
  static_assert(D == 128);
  constexpr int KB = 64 * 256;     // 16 KiB per rm image
  extern __shared__ __attribute__((aligned(16))) char smem[];
  auto k_lds = [&](int bb) -> char* { return smem + bb * KB; };
  auto v_lds = [&](int bb) -> char* { return smem + (NBUF + bb) * KB; };

  const int bh = blockIdx.x;
  const int q0 = blockIdx.y * 256;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int iq = lane & 31;
  const int hi = lane >> 5;
  const int g1 = (lane >> 4) & 1;

  const bf16* Qb = Q + (int64_t)bh * S * D;
  const bf16* dOb = dO + (int64_t)bh * S * D;
  const bf16* Kb = K + (int64_t)bh * Skv * D;
  const bf16* Vb = V + (int64_t)bh * Skv * D;

  const int my_q = q0 + wid * 32 + iq;
  const int diag = Skv - S;
  const float lse_q = LSE[(int64_t)bh * S + min(my_q, S - 1)];
  const float del_q = DELTA[(int64_t)bh * S + min(my_q, S - 1)];

  bf16x8 qreg[8], doreg[8];
  {
    const bf16* qrow = Qb + (int64_t)min(my_q, S - 1) * D;
    const bf16* drow = dOb + (int64_t)min(my_q, S - 1) * D;
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

  // B2: walking pointers (2 slots per lane, K and V share layout)
  const int wlane16 = lane * 16;
  int pos_c[2], krow_c[2], kd_c[2];
  const bf16* kp[2];
  const bf16* vp[2];
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    int pos = (wid * 2 + i) * 1024 + wlane16;
    pos_c[i] = pos;
    krow_c[i] = pos >> 8;
    kd_c[i] = ((pos & 255) ^ ((krow_c[i] & 15) << 4)) >> 1;
    kp[i] = Kb + (int64_t)krow_c[i] * D + kd_c[i];
    vp[i] = Vb + (int64_t)krow_c[i] * D + kd_c[i];
  }
  constexpr int64_t STEP = 64 * D;

#define DQ3_FAST(buf)                                                       \
  do {                                                                      \
    _Pragma("unroll")                                                       \
    for (int i = 0; i < 2; ++i) {                                           \
      __builtin_amdgcn_global_load_lds(                                     \
          (const __attribute__((address_space(1))) void*)kp[i],             \
          (__attribute__((address_space(3))) void*)(k_lds(buf) + pos_c[i]), \
          16, 0, 0);                                                        \
      __builtin_amdgcn_global_load_lds(                                     \
          (const __attribute__((address_space(1))) void*)vp[i],             \
          (__attribute__((address_space(3))) void*)(v_lds(buf) + pos_c[i]), \
          16, 0, 0);                                                        \
      kp[i] += STEP;                                                        \
      vp[i] += STEP;                                                        \
    }                                                                       \
  } while (0)

#define DQ3_CLAMPED(k0v, buf)                                               \
  do {                                                                      \
    _Pragma("unroll")                                                       \
    for (int i = 0; i < 2; ++i) {                                           \
      int64_t roff = (int64_t)min((k0v) + krow_c[i], Skv - 1) * D + kd_c[i];\
      __builtin_amdgcn_global_load_lds(                                     \
          (const __attribute__((address_space(1))) void*)(Kb + roff),       \
          (__attribute__((address_space(3))) void*)(k_lds(buf) + pos_c[i]), \
          16, 0, 0);                                                        \
      __builtin_amdgcn_global_load_lds(                                     \
          (const __attribute__((address_space(1))) void*)(Vb + roff),       \
          (__attribute__((address_space(3))) void*)(v_lds(buf) + pos_c[i]), \
          16, 0, 0);                                                        \
      kp[i] += STEP;                                                        \
      vp[i] += STEP;                                                        \
    }                                                                       \
  } while (0)

#define DQ3_ISSUE(k0v, buf)                                                 \
  do {                                                                      \
    if ((k0v) + 64 <= Skv) DQ3_FAST(buf);                                   \
    else DQ3_CLAMPED(k0v, buf);                                             \
  } while (0)

  f32x16 dq_acc[4];
#pragma unroll
  for (int dt = 0; dt < 4; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r) dq_acc[dt][r] = 0.f;

  const int wave_qmax = q0 + wid * 32 + 31;
  int n_tiles = (Skv + 63) / 64;
  if (causal) n_tiles = min(n_tiles, max((q0 + 256 + diag + 63) / 64, 1));

  DQ3_ISSUE(0, 0);
  if (n_tiles > 1) DQ3_ISSUE(64, 1);
  else DQ3_CLAMPED(0, 1);     // duplicate: keeps the counted wait sound

  for (int t = 0; t < n_tiles; ++t) {
    const int k0 = t * 64;
    const int cur = t % NBUF;
    // B1: allow the newest tile's 4 loads to stay in flight
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    if (t + 2 < n_tiles) DQ3_ISSUE(k0 + 128, (t + 2) % NBUF);

    const bool active = !causal || (k0 <= wave_qmax + diag);
    if (active) {
      float ds[2][16];
#pragma unroll
      for (int ct = 0; ct < 2; ++ct) {
        f32x16 sa, da;
#pragma unroll
        for (int r = 0; r < 16; ++r) { sa[r] = 0.f; da[r] = 0.f; }
#pragma unroll
        for (int kk = 0; kk < 8; ++kk) {
          bf16x8 kf = *reinterpret_cast<const bf16x8*>(
              k_lds(cur) + kswz_row3(32 * ct + iq, kk * 32 + hi * 16));
          bf16x8 vf = *reinterpret_cast<const bf16x8*>(
              v_lds(cur) + kswz_row3(32 * ct + iq, kk * 32 + hi * 16));
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
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        bf16x8 dsf;
        PACK_FRAG3(dsf, ds[ks >> 1], (ks & 1) * 8);
        bf16x8 kt0, kt1, kt2, kt3;
        TR_RM4_3(kt0, kt1, kt2, kt3, k_lds(cur), 16 * ks + 8 * hi,
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
  }
#undef DQ3_ISSUE
#undef DQ3_FAST
#undef DQ3_CLAMPED

  if (my_q < S) {
    bf16* qrow = dQ + ((int64_t)bh * S + my_q) * D;
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
        union { unsigned u[2]; uint2 v; } stv;
        stv.u[0] = lo; stv.u[1] = hs;
        *reinterpret_cast<uint2*>(qrow + d0) = stv.v;
      }
    }
  }
}

// ===========================================================================
// dkv kernel v3: 8 waves x 32 keys; 3-ring Q/dO staging, counted vmcnt.
// ===========================================================================
template <int D>
__global__ __launch_bounds__(THREADS, 1) void fa3_bwd_dkv_kernel(
    const bf16* __restrict__ Q, const bf16* __restrict__ K,
    const bf16* __restrict__ V, const bf16* __restrict__ dO,
    const float* __restrict__ LSE, const float* __restrict__ DELTA,
    bf16* __restrict__ dK16, bf16* __restrict__ dV16,
    int B, int H, int S, int Skv, float scale, bool causal) {
  static_assert(D == 128);
  constexpr int QB = 32 * 256;     // 8 KiB per rm image
  extern __shared__ __attribute__((aligned(16))) char smem[];
  // dkv uses a 2-buffer ring with issue-at-end + counted vmcnt (the 3rd
  // buffer's live state spilled past the 2-waves/SIMD VGPR budget)
  auto q_lds = [&](int bb) -> char* { return smem + bb * QB; };
  auto do_lds = [&](int bb) -> char* { return smem + (2 + bb) * QB; };
  char* vw_base = smem + 4 * QB;
  char* pw_base = vw_base + 8 * 8192;

  const int bh = blockIdx.x;
  const int kb0 = blockIdx.y * 256;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int iq = lane & 31;
  const int hi = lane >> 5;
  const int g1 = (lane >> 4) & 1;

  const bf16* Qb = Q + (int64_t)bh * S * D;
  const bf16* dOb = dO + (int64_t)bh * S * D;
  const bf16* Kb = K + (int64_t)bh * Skv * D;
  const bf16* Vb = V + (int64_t)bh * Skv * D;
  const float* lse_b = LSE + (int64_t)bh * S;
  const float* del_b = DELTA + (int64_t)bh * S;

  const int my_key = kb0 + wid * 32 + iq;
  const int diag = Skv - S;

  char* vw_lds = vw_base + wid * 8192;
  char* p_lds = pw_base + wid * 4608;
  char* ds_lds = p_lds + 2304;

  bf16x8 kreg[8];
  {
    const bf16* krow = Kb + (int64_t)min(my_key, Skv - 1) * D;
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
  for (int i = 0; i < 8; ++i) {     // stage the wave's V rows (8 KiB)
    int pos = (i * 64 + lane) * 16;
    int vrow = pos >> 8;
    int vd = ((pos & 255) ^ ((vrow & 15) << 4)) >> 1;
    const bf16* vsrc = Vb
        + (int64_t)min(kb0 + wid * 32 + vrow, Skv - 1) * D + vd;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)vsrc,
        (__attribute__((address_space(3))) void*)(vw_lds + pos), 16, 0, 0);
  }

  // walking pointers: one (q, do) pair per lane per tile
  const int pos0 = tid * 16;
  const int qrow_c = pos0 >> 8;
  const int qd_c = ((pos0 & 255) ^ ((qrow_c & 15) << 4)) >> 1;

  const int n_q_tiles = (S + 31) / 32;
  int t_start = 0;
  if (causal) t_start = max(0, (kb0 - diag) / 32);
  const int wave_kmin = kb0 + wid * 32;

  // dkv keeps v2-style clamped addressing (1 slot/lane, the address math
  // is cheap); the v3 win here is the 3-ring counted-vmcnt pipeline —
  // the kernel sits exactly at the 2-waves/SIMD VGPR budget, so no
  // walking pointers
#define DKV3_ISSUE(qt0, buf)                                                \
  do {                                                                      \
    int qr_ = min((qt0) + qrow_c, S - 1);                                   \
    __builtin_amdgcn_global_load_lds(                                       \
        (const __attribute__((address_space(1))) void*)(                    \
            Qb + (int64_t)qr_ * D + qd_c),                                  \
        (__attribute__((address_space(3))) void*)(q_lds(buf) + pos0),       \
        16, 0, 0);                                                          \
    __builtin_amdgcn_global_load_lds(                                       \
        (const __attribute__((address_space(1))) void*)(                    \
            dOb + (int64_t)qr_ * D + qd_c),                                 \
        (__attribute__((address_space(3))) void*)(do_lds(buf) + pos0),      \
        16, 0, 0);                                                          \
  } while (0)

  f32x16 dk_acc[4], dv_acc[4];
#pragma unroll
  for (int dt = 0; dt < 4; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r) { dk_acc[dt][r] = 0.f; dv_acc[dt][r] = 0.f; }

  DKV3_ISSUE(t_start * 32, 0);
  if (t_start + 1 < n_q_tiles) DKV3_ISSUE(t_start * 32 + 32, 1);
  else DKV3_ISSUE(t_start * 32, 1);   // duplicate keeps the wait sound

  for (int t = t_start; t < n_q_tiles; ++t) {
    const int qt0 = t * 32;
    const int cur = (t - t_start) & 1;
    // allow the newest issue's 2 loads to stay in flight; the V image
    // and tile t have landed once this retires
    asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
    __builtin_amdgcn_s_barrier();

    const bool active = !causal || (qt0 + 31 >= wave_kmin - diag);
    if (active) {
      const int my_q = qt0 + iq;
      const float lse_q = lse_b[min(my_q, S - 1)];
      const float del_q = del_b[min(my_q, S - 1)];
      f32x16 sa, da;
#pragma unroll
      for (int r = 0; r < 16; ++r) { sa[r] = 0.f; da[r] = 0.f; }
#pragma unroll
      for (int kk = 0; kk < 8; ++kk) {
        bf16x8 qf = *reinterpret_cast<const bf16x8*>(
            q_lds(cur) + kswz_row3(iq, kk * 32 + hi * 16));
        bf16x8 df = *reinterpret_cast<const bf16x8*>(
            do_lds(cur) + kswz_row3(iq, kk * 32 + hi * 16));
        bf16x8 vf = *reinterpret_cast<const bf16x8*>(
            vw_lds + kswz_row3(iq, kk * 32 + hi * 16));
        sa = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kreg[kk], qf, sa,
                                                     0, 0, 0);
        da = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, df, da,
                                                     0, 0, 0);
      }
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int key = kb0 + wid * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        bool masked = (key >= Skv) || (my_q >= S) ||
                      (causal && key > my_q + diag);
        float pv = masked ? 0.f : __expf(sa[r] - lse_q);
        sa[r] = pv;
        da[r] = masked ? 0.f : pv * (da[r] - del_q);
      }
#pragma unroll
      for (int rq = 0; rq < 4; ++rq) {
        int keyb = (8 * rq + 4 * hi) & 31;
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
        *reinterpret_cast<uint2*>(p_lds + iq * 72 + keyb * 2) = sp.v;
        *reinterpret_cast<uint2*>(ds_lds + iq * 72 + keyb * 2) = sd.v;
      }
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

#pragma unroll
      for (int s = 0; s < 2; ++s) {
        bf16x8 ptf, dstf;
        TR_P3(ptf, p_lds, 16 * s + 8 * hi, (16 * g1 + 4 * (lane & 3)) * 2);
        TR_P3(dstf, ds_lds, 16 * s + 8 * hi,
              (16 * g1 + 4 * (lane & 3)) * 2);
        bf16x8 do0, do1, do2, do3, qa0, qa1, qa2, qa3;
        TR_RM4_3(do0, do1, do2, do3, do_lds(cur), 16 * s + 8 * hi,
                 (16 * g1 + 4 * (lane & 3)) * 2);
        TR_RM4_3(qa0, qa1, qa2, qa3, q_lds(cur), 16 * s + 8 * hi,
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
    // all waves done reading buffer `cur` -> refill it with tile t+2
    // (the DMA then lands under tile t+1's compute)
    __builtin_amdgcn_s_barrier();
    if (t + 2 < n_q_tiles) DKV3_ISSUE(qt0 + 64, cur);
  }
#undef DKV3_ISSUE

#pragma unroll
  for (int dt = 0; dt < 4; ++dt) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int key = kb0 + wid * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
      if (key < Skv) {
        int64_t off = ((int64_t)bh * Skv + key) * D + 32 * dt + iq;
        dK16[off] = (bf16)(dk_acc[dt][r] * scale);
        dV16[off] = (bf16)(dv_acc[dt][r]);
      }
    }
  }
}

// delta = rowsum(dO * O), BHSD contiguous (same as v2's fast path)
__global__ void fa3_delta_kernel(const bf16* __restrict__ dO,
                                 const bf16* __restrict__ O,
                                 float* __restrict__ delta, int64_t rows) {
  const int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t row = tid >> 2;
  const int q = tid & 3;
  const int64_t rstride = ((int64_t)gridDim.x * blockDim.x) >> 2;
  for (; row < rows; row += rstride) {
    const bf16* a = dO + row * 128 + q * 32;
    const bf16* bb = O + row * 128 + q * 32;
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
}

}  // namespace

std::vector<torch::Tensor> flash_attn_bwd_v3(torch::Tensor dout,
                                             torch::Tensor q,
                                             torch::Tensor k,
                                             torch::Tensor v,
                                             torch::Tensor out,
                                             torch::Tensor lse,
                                             bool causal, double scale) {
  const int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  const int Hkv = k.size(1), Skv = k.size(2);
  TORCH_CHECK(D == 128 && H == Hkv, "fa3 bwd: D=128, non-GQA only");
  auto stream = hetu_current_stream();
  auto delta = torch::empty({B, H, S}, q.options().dtype(at::kFloat));
  {
    int64_t rows = (int64_t)B * H * S;
    int grid = (int)std::min<int64_t>((rows * 4 + 255) / 256, 16384);
    hipLaunchKernelGGL(fa3_delta_kernel, dim3(grid), dim3(256), 0, stream,
                       (const bf16*)dout.data_ptr(),
                       (const bf16*)out.data_ptr(),
                       delta.data_ptr<float>(), rows);
  }
  auto dq = torch::empty_like(q);
  auto dk = torch::empty_like(k);
  auto dv = torch::empty_like(v);
  {
    dim3 grid(B * H, (S + 255) / 256);
    size_t lds = 2 * NBUF * (size_t)64 * 256;   // 96 KiB
    hipLaunchKernelGGL(fa3_bwd_dq_kernel<128>, grid, dim3(THREADS), lds,
                       stream, (const bf16*)q.data_ptr(),
                       (const bf16*)k.data_ptr(), (const bf16*)v.data_ptr(),
                       (const bf16*)dout.data_ptr(), lse.data_ptr<float>(),
                       delta.data_ptr<float>(), (bf16*)dq.data_ptr(),
                       B, H, S, Skv, (float)scale, causal);
  }
  {
    dim3 grid(B * H, (Skv + 255) / 256);
    size_t lds = 4 * (size_t)32 * 256 + 8 * 8192 + 8 * 4608;
    hipLaunchKernelGGL(fa3_bwd_dkv_kernel<128>, grid, dim3(THREADS), lds,
                       stream, (const bf16*)q.data_ptr(),
                       (const bf16*)k.data_ptr(), (const bf16*)v.data_ptr(),
                       (const bf16*)dout.data_ptr(), lse.data_ptr<float>(),
                       delta.data_ptr<float>(), (bf16*)dk.data_ptr(),
                       (bf16*)dv.data_ptr(), B, H, S, Skv, (float)scale,
                       causal);
  }
  return {dq, dk, dv};
}
