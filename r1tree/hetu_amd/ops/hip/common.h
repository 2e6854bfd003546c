// Common helpers for hetu_amd CDNA4 (gfx950) kernels.
//
// MI355X-native from scratch; wave width is 64 (not 32), LDS is 160 KiB/CU,
// HBM3E ~8 TB/s. Memory-bound kernels vectorize bf16 loads as ushort8
// (16 B/lane) per the CDNA HIP guide (Guideline 13).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE 64
#define DEV __device__ __forceinline__

typedef unsigned int u32;
typedef unsigned long long u64;

// per-tensor addressing for attention kernels: element strides for batch,
// head and token.  BHSD contiguous: {H*S*D, S*D, D}; fused-qkv / BS[HD]
// views: {S*row, D, row} with row = (H + 2*Hkv) * D.
struct FaStrides {
  long long bs, hs, rs;
};
typedef __hip_bfloat16 bf16;

// ---- vector types ---------------------------------------------------------
struct alignas(16) ushort8 { unsigned short v[8]; };
struct alignas(16) float4v { float v[4]; };

DEV float bf2f(unsigned short u) {
  u32 x = ((u32)u) << 16;
  return __uint_as_float(x);
}
DEV unsigned short f2bf(float f) {
  u32 x = __float_as_uint(f);
  // round-to-nearest-even
  u32 lsb = (x >> 16) & 1;
  x += 0x7fff + lsb;
  return (unsigned short)(x >> 16);
}

// ---- wave / block reductions ---------------------------------------------
DEV float wave_sum(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    x += __shfl_xor(x, off, WAVE);
  return x;
}

DEV float wave_max(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    x = fmaxf(x, __shfl_xor(x, off, WAVE));
  return x;
}

// Block reduction over up to 1024 threads (<=16 waves). smem must hold 16
// floats. Returns the reduced value on every thread.
template <typename Red>
DEV float block_reduce(float x, float* smem, Red red, float init) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x >> 6;
  const int nwaves = (blockDim.x + WAVE - 1) >> 6;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    x = red(x, __shfl_xor(x, off, WAVE));
  if (lane == 0) smem[wid] = x;
  __syncthreads();
  x = (lane < nwaves) ? smem[lane] : init;
#pragma unroll
  for (int off = 8; off > 0; off >>= 1)
    x = red(x, __shfl_xor(x, off, WAVE));
  x = __shfl(x, 0, WAVE);
  __syncthreads();
  return x;
}

struct SumRed { DEV float operator()(float a, float b) const { return a + b; } };
struct MaxRed { DEV float operator()(float a, float b) const { return fmaxf(a, b); } };

DEV float block_sum(float x, float* smem) { return block_reduce(x, smem, SumRed{}, 0.f); }
DEV float block_max(float x, float* smem) { return block_reduce(x, smem, MaxRed{}, -INFINITY); }

// ---- Philox 4x32-10 (stateless dropout RNG) -------------------------------
struct Philox {
  u32 k0, k1, c0, c1, c2, c3;
  DEV Philox(u64 seed, u64 offset, u32 idx) {
    k0 = (u32)seed; k1 = (u32)(seed >> 32);
    c0 = (u32)offset; c1 = (u32)(offset >> 32); c2 = idx; c3 = 0;
  }
  DEV static u32 mulhilo(u32 a, u32 b, u32* hi) {
    u64 p = (u64)a * b; *hi = (u32)(p >> 32); return (u32)p;
  }
  DEV void round_(u32& x0, u32& x1, u32& x2, u32& x3, u32 kk0, u32 kk1) {
    u32 hi0, hi1;
    u32 lo0 = mulhilo(0xD2511F53u, x0, &hi0);
    u32 lo1 = mulhilo(0xCD9E8D57u, x2, &hi1);
    u32 y0 = hi1 ^ x1 ^ kk0, y1 = lo1, y2 = hi0 ^ x3 ^ kk1, y3 = lo0;
    x0 = y0; x1 = y1; x2 = y2; x3 = y3;
  }
  DEV void next4(u32 out[4]) {
    u32 x0 = c0, x1 = c1, x2 = c2, x3 = c3;
    u32 kk0 = k0, kk1 = k1;
#pragma unroll
    for (int i = 0; i < 10; ++i) {
      round_(x0, x1, x2, x3, kk0, kk1);
      kk0 += 0x9E3779B9u; kk1 += 0xBB67AE85u;
    }
    out[0] = x0; out[1] = x1; out[2] = x2; out[3] = x3;
    ++c3;
  }
};

DEV float u32_to_uniform(u32 x) {
  // (0, 1]
  return (x >> 8) * (1.0f / 16777216.0f);
}

// ---- vectorized load/store: bf16 as ushort8 (16 B/lane), f32 as float4 ---
template <typename T> struct VecIO;

template <> struct VecIO<bf16> {
  static constexpr int VEC = 8;
  DEV static void load(const bf16* p, float* out) {
    ushort8 u = *reinterpret_cast<const ushort8*>(p);
#pragma unroll
    for (int i = 0; i < 8; ++i) out[i] = bf2f(u.v[i]);
  }
  DEV static void store(bf16* p, const float* in) {
    ushort8 u;
#pragma unroll
    for (int i = 0; i < 8; ++i) u.v[i] = f2bf(in[i]);
    *reinterpret_cast<ushort8*>(p) = u;
  }
};

template <> struct VecIO<float> {
  static constexpr int VEC = 4;
  DEV static void load(const float* p, float* out) {
    float4v u = *reinterpret_cast<const float4v*>(p);
#pragma unroll
    for (int i = 0; i < 4; ++i) out[i] = u.v[i];
  }
  DEV static void store(float* p, const float* in) {
    float4v u;
#pragma unroll
    for (int i = 0; i < 4; ++i) u.v[i] = in[i];
    *reinterpret_cast<float4v*>(p) = u;
  }
};

#define DISPATCH_FLOAT(TENSOR, NAME, ...)                                  \
  do {                                                                     \
    if ((TENSOR).scalar_type() == at::kBFloat16) {                         \
      using scalar_t = bf16; __VA_ARGS__();                                \
    } else if ((TENSOR).scalar_type() == at::kFloat) {                     \
      using scalar_t = float; __VA_ARGS__();                               \
    } else {                                                               \
      TORCH_CHECK(false, NAME ": unsupported dtype");                      \
    }                                                                      \
  } while (0)

#define HIP_CHECK(cmd)                                                     \
  do {                                                                     \
    hipError_t e = (cmd);                                                  \
    if (e != hipSuccess) {                                                 \
      TORCH_CHECK(false, "HIP error: ", hipGetErrorString(e));             \
    }                                                                      \
  } while (0)
