// Flash-attention forward v3 (EXPERIMENTAL, round-2 candidate — compiled
// but not wired into the default dispatch; select with the
// flash_attn_fwd_v3 binding / HETU_AMD_FA_V3=1).
//
// Same math and layouts as attention_v2.hip; addresses the measured
// VALU-bound profile (PMC: 17 VALU/MFMA, disassembly: ~350 address-math
// VALU per tile):
//   A1. glds source addresses: per-slot byte pointers precomputed once,
//       advanced by one uniform offset per tile; the min-clamp slow path
//       only runs for the ragged tail tile.
//   A2. LDS read addresses (K fragments, V transpose base) precomputed as
//       lane constants; per tile a single add of the ring-buffer base.
//   A3. defer-max: the o_acc rescale (64 v_mul) is skipped when no lane's
//       running max changed this tile (wave vote).
#include <torch/extension.h>
#include "ext_stream.h"
#include "common.h"

namespace {

using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x16 = __attribute__((ext_vector_type(16))) float;
using u32x2 = __attribute__((ext_vector_type(2))) unsigned int;

constexpr int BM = 256;
constexpr int BN = 64;
constexpr int THREADS = BM * 2;
constexpr int NGLDS = 64 * 256 / (THREADS * 16);
constexpr int NBUF = 3;

DEV int kswz3(int row, int byte_in_row) {
  return row * 256 + (byte_in_row ^ ((row & 15) << 4));
}

DEV int voff3(int key, int d) {
  return ((key >> 5) * 8 + (d >> 4)) * 1024 + (key & 31) * 32 + (d & 15) * 2;
}

template <int D>
__global__ __launch_bounds__(THREADS, 1) void fa3_fwd_kernel(
    const bf16* __restrict__ Q, const bf16* __restrict__ K,
    const bf16* __restrict__ V, bf16* __restrict__ O,
    float* __restrict__ LSE, int B, int H, int Hkv, int S, int Skv,
    float scale, bool causal, FaStrides sq, FaStrides skv, FaStrides so) {
  static_assert(D == 128, "fa3 fwd: D=128 only");
  constexpr int KBYTES = BN * D * 2;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  auto k_lds = [&](int bb) -> char* { return smem + bb * KBYTES; };
  auto v_lds = [&](int bb) -> char* { return smem + (NBUF + bb) * KBYTES; };

  const int bh = blockIdx.x;
  const int h = bh % H;
  const int b = bh / H;
  const int hkv = h / (H / Hkv);
  const int q0 = blockIdx.y * BM;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int iq = lane & 31;
  const int hi = lane >> 5;

  const bf16* Qb = Q + (int64_t)b * sq.bs + (int64_t)h * sq.hs;
  const bf16* Kb = K + (int64_t)b * skv.bs + (int64_t)hkv * skv.hs;
  const bf16* Vb = V + (int64_t)b * skv.bs + (int64_t)hkv * skv.hs;

  const int my_q = q0 + wid * 32 + iq;
  const int diag = Skv - S;

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

  // ---- A1: per-slot glds layout constants + walking source pointers ----
  const int wlane16 = lane * 16;
  int kpos[NGLDS], krow_c[NGLDS], kd_c[NGLDS], vkey_c[NGLDS], vd_c[NGLDS];
#pragma unroll
  for (int i = 0; i < NGLDS; ++i) {
    int pos = (wid * NGLDS + i) * 1024 + wlane16;
    kpos[i] = pos;
    krow_c[i] = pos >> 8;
    kd_c[i] = ((pos & 255) ^ ((krow_c[i] & 15) << 4)) >> 1;
    int st = pos >> 10;
    vkey_c[i] = (st >> 3) * 32 + ((pos >> 5) & 31);
    vd_c[i] = (st & 7) * 16 + ((pos >> 4) & 1) * 8;
  }
  // walking pointers for the FAST (unclamped) path; advanced per issue
  const bf16* kp[NGLDS];
  const bf16* vp[NGLDS];
#pragma unroll
  for (int i = 0; i < NGLDS; ++i) {
    kp[i] = Kb + (int64_t)krow_c[i] * skv.rs + kd_c[i];
    vp[i] = Vb + (int64_t)vkey_c[i] * skv.rs + vd_c[i];
  }
  const int64_t issue_step = (int64_t)BN * skv.rs;

// fast path: tile fully in range — walking pointers, no clamping
#define ISSUE_FAST(buf)                                                     \
  do {                                                                      \
    _Pragma("unroll")                                                       \
    for (int i = 0; i < NGLDS; ++i) {                                       \
      __builtin_amdgcn_global_load_lds(                                     \
          (const __attribute__((address_space(1))) void*)kp[i],             \
          (__attribute__((address_space(3))) void*)(k_lds(buf) + kpos[i]),  \
          16, 0, 0);                                                        \
      __builtin_amdgcn_global_load_lds(                                     \
          (const __attribute__((address_space(1))) void*)vp[i],             \
          (__attribute__((address_space(3))) void*)(v_lds(buf) + kpos[i]),  \
          16, 0, 0);                                                        \
      kp[i] += issue_step;                                                  \
      vp[i] += issue_step;                                                  \
    }                                                                       \
  } while (0)

// ragged tail: clamped recompute (rare)
#define ISSUE_CLAMPED(k0v, buf)                                             \
  do {                                                                      \
    _Pragma("unroll")                                                       \
    for (int i = 0; i < NGLDS; ++i) {                                       \
      const bf16* ks_ =                                                     \
          Kb + (int64_t)min((k0v) + krow_c[i], Skv - 1) * skv.rs + kd_c[i]; \
      const bf16* vs_ =                                                     \
          Vb + (int64_t)min((k0v) + vkey_c[i], Skv - 1) * skv.rs + vd_c[i]; \
      __builtin_amdgcn_global_load_lds(                                     \
          (const __attribute__((address_space(1))) void*)ks_,               \
          (__attribute__((address_space(3))) void*)(k_lds(buf) + kpos[i]),  \
          16, 0, 0);                                                        \
      __builtin_amdgcn_global_load_lds(                                     \
          (const __attribute__((address_space(1))) void*)vs_,               \
          (__attribute__((address_space(3))) void*)(v_lds(buf) + kpos[i]),  \
          16, 0, 0);                                                        \
      kp[i] += issue_step;                                                  \
      vp[i] += issue_step;                                                  \
    }                                                                       \
  } while (0)

#define ISSUE3(k0v, buf)                                                    \
  do {                                                                      \
    if ((k0v) + BN <= Skv) ISSUE_FAST(buf);                                 \
    else ISSUE_CLAMPED(k0v, buf);                                           \
  } while (0)

  // ---- A2: LDS read-address lane constants -----------------------------
  unsigned kaddr_c[2][8];          // [ct][kk] byte offset within K image
#pragma unroll
  for (int ct = 0; ct < 2; ++ct)
#pragma unroll
    for (int kk = 0; kk < 8; ++kk)
      kaddr_c[ct][kk] = kswz3(32 * ct + iq, kk * 32 + hi * 16);
  unsigned vaddr_c[4];             // [ks] tr-read base within V image
  {
    const int dbase0 = 16 * ((lane >> 4) & 1) + 4 * (lane & 3);
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
      const int keyb = 16 * ks + 8 * hi + ((lane >> 2) & 3);
      vaddr_c[ks] = voff3(keyb, dbase0);
    }
  }

  f32x16 o_acc[4];
#pragma unroll
  for (int dt = 0; dt < 4; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r) o_acc[dt][r] = 0.f;
  float m_i = -INFINITY, l_i = 0.f;

  const int wave_qmax = q0 + wid * 32 + 31;
  int n_tiles = (Skv + BN - 1) / BN;
  if (causal) {
    int lim = (q0 + BM + diag + BN - 1) / BN;
    n_tiles = min(n_tiles, max(lim, 1));
  }

  ISSUE3(0, 0);
  if (n_tiles > 1) ISSUE3(BN, 1);
  else ISSUE_CLAMPED(0, 1);   // duplicate: keeps the counted wait sound

  for (int t = 0; t < n_tiles; ++t) {
    const int k0 = t * BN;
    const int cur = t % NBUF;
    asm volatile("s_waitcnt vmcnt(%0)" :: "i"(2 * NGLDS) : "memory");
    __builtin_amdgcn_s_barrier();
    if (t + 2 < n_tiles) ISSUE3(k0 + 2 * BN, (t + 2) % NBUF);

    const bool active = !causal || (k0 <= wave_qmax + diag);

    float p[2][16];
    if (active) {
      char* kbase = k_lds(cur);
#pragma unroll
      for (int ct = 0; ct < 2; ++ct) {
        f32x16 acc;
#pragma unroll
        for (int r = 0; r < 16; ++r) acc[r] = 0.f;
#pragma unroll
        for (int kk = 0; kk < 8; ++kk) {
          bf16x8 kf = *reinterpret_cast<const bf16x8*>(
              kbase + kaddr_c[ct][kk]);
          acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qreg[kk], acc,
                                                        0, 0, 0);
        }
#pragma unroll
        for (int r = 0; r < 16; ++r) p[ct][r] = acc[r];
      }

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
      // A3: skip the 64-mul rescale when no lane rescaled this tile
      if (__any(alpha != 1.f)) {
#pragma unroll
        for (int dt = 0; dt < 4; ++dt)
#pragma unroll
          for (int r = 0; r < 16; ++r) o_acc[dt][r] *= alpha;
      }

      char* vbase = v_lds(cur);
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        bf16x8 pfrag;
        {
          const int ct = ks >> 1;
          const int rb = (ks & 1) * 8;
          unsigned w0, w1, w2, w3;
          asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2"
                       : "=v"(w0) : "v"(p[ct][rb + 0]), "v"(p[ct][rb + 1]));
          asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2"
                       : "=v"(w2) : "v"(p[ct][rb + 4]), "v"(p[ct][rb + 5]));
          asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2"
                       : "=v"(w1) : "v"(p[ct][rb + 2]), "v"(p[ct][rb + 3]));
          asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2"
                       : "=v"(w3) : "v"(p[ct][rb + 6]), "v"(p[ct][rb + 7]));
          asm volatile("s_nop 1\n\tv_permlane32_swap_b32 %0, %1"
                       : "+v"(w0), "+v"(w2));
          asm volatile("s_nop 1\n\tv_permlane32_swap_b32 %0, %1"
                       : "+v"(w1), "+v"(w3));
          union { unsigned u[4]; bf16x8 v; } pk;
          pk.u[0] = w0; pk.u[1] = w1; pk.u[2] = w2; pk.u[3] = w3;
          pfrag = pk.v;
        }
        unsigned a1 = (unsigned)(uintptr_t)(
            (__attribute__((address_space(3))) char*)(vbase + vaddr_c[ks]));
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
#undef ISSUE3
#undef ISSUE_FAST
#undef ISSUE_CLAMPED

  if (my_q < S) {
    float inv = (l_i > 0.f) ? 1.f / l_i : 0.f;
    bf16* orow = O + (int64_t)b * so.bs + (int64_t)h * so.hs
                 + (int64_t)my_q * so.rs;
#pragma unroll
    for (int dt = 0; dt < 4; ++dt) {
#pragma unroll
      for (int g = 0; g < 4; ++g) {
        int d0 = 32 * dt + 8 * g + 4 * hi;
        unsigned lo, hsp;
        float f0 = o_acc[dt][4 * g + 0] * inv;
        float f1 = o_acc[dt][4 * g + 1] * inv;
        float f2 = o_acc[dt][4 * g + 2] * inv;
        float f3 = o_acc[dt][4 * g + 3] * inv;
        asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2"
                     : "=v"(lo) : "v"(f0), "v"(f1));
        asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2"
                     : "=v"(hsp) : "v"(f2), "v"(f3));
        union { unsigned u[2]; uint2 v; } stv;
        stv.u[0] = lo; stv.u[1] = hsp;
        *reinterpret_cast<uint2*>(orow + d0) = stv.v;
      }
    }
    if (hi == 0) {
      LSE[(int64_t)bh * S + my_q] =
          (l_i > 0.f) ? m_i + __logf(l_i) : -INFINITY;
    }
  }
}

}  // namespace

std::vector<torch::Tensor> flash_attn_fwd_v3(torch::Tensor q,
                                             torch::Tensor k,
                                             torch::Tensor v, bool causal,
                                             double scale) {
  TORCH_CHECK(q.dim() == 4 && q.scalar_type() == at::kBFloat16);
  const int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  const int Hkv = k.size(1), Skv = k.size(2);
  TORCH_CHECK(D == 128, "fa3: D=128 only");
  auto o = torch::empty_like(q);
  auto lse = torch::empty({B, H, S}, q.options().dtype(at::kFloat));
  auto stream = hetu_current_stream();
  FaStrides sq{(long long)H * S * D, (long long)S * D, (long long)D};
  FaStrides skv{(long long)Hkv * Skv * D, (long long)Skv * D,
                (long long)D};
  dim3 grid(B * H, (S + BM - 1) / BM);
  size_t lds = 2 * NBUF * (size_t)BN * 128 * 2;
  hipLaunchKernelGGL(fa3_fwd_kernel<128>, grid, dim3(THREADS), lds, stream,
                     (const bf16*)q.data_ptr(), (const bf16*)k.data_ptr(),
                     (const bf16*)v.data_ptr(), (bf16*)o.data_ptr(),
                     lse.data_ptr<float>(), B, H, Hkv, S, Skv,
                     (float)scale, causal, sq, skv, sq);
  return {o, lse};
}
