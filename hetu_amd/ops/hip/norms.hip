// Fused RMSNorm / LayerNorm forward+backward for gfx950.
//
// Design (MI355X): one WAVE per row — D=4096 rows are only 8 KiB, so a
// 256-thread block per row wastes the chip on two LDS reductions per row
// and a 1024-block grid cap (atomic bound) starved the CUs (575us vs the
// ~100us bandwidth bound).  Wave-per-row uses shuffle reductions (no
// barriers), keeps 4 rows in flight per block, and the backward
// accumulates dw/db in an LDS image per block (ds-atomics) flushed once
// with global atomics — grid*D atomic traffic at a 2048-block grid.
// bf16 IO is vectorized 16 B/lane.  fp32 stats.
// Capability parity: reference RMSNorm.cu / FusedLayerNorm.cu:455-760.
#include <torch/extension.h>
#include "ext_stream.h"
#include "common.h"

namespace {

constexpr int BLOCK = 256;
constexpr int WAVES = BLOCK / 64;
constexpr int MAX_D = 8192;        // LDS dw/db image bound (bwd)

// ---------------- RMSNorm ----------------

template <typename T>
__global__ void rmsnorm_fwd_kernel(const T* __restrict__ x,
                                   const T* __restrict__ w,
                                   T* __restrict__ y,
                                   float* __restrict__ rstd_out,
                                   int64_t rows, int D, float eps) {
  constexpr int V = VecIO<T>::VEC;
  const int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
  for (int64_t row = (int64_t)blockIdx.x * WAVES + wid; row < rows;
       row += (int64_t)gridDim.x * WAVES) {
    const T* xr = x + row * D;
    T* yr = y + row * D;
    float ss = 0.f;
    for (int i = lane * V; i < D; i += 64 * V) {
      float v[VecIO<T>::VEC];
      VecIO<T>::load(xr + i, v);
#pragma unroll
      for (int j = 0; j < V; ++j) ss += v[j] * v[j];
    }
    ss = wave_sum(ss);
    float rstd = rsqrtf(ss / D + eps);
    if (lane == 0) rstd_out[row] = rstd;
    for (int i = lane * V; i < D; i += 64 * V) {
      float v[VecIO<T>::VEC], wv[VecIO<T>::VEC];
      VecIO<T>::load(xr + i, v);
      VecIO<T>::load(w + i, wv);
#pragma unroll
      for (int j = 0; j < V; ++j) v[j] = v[j] * rstd * wv[j];
      VecIO<T>::store(yr + i, v);
    }
  }
}

template <typename T>
__global__ void rmsnorm_bwd_kernel(const T* __restrict__ dy,
                                   const T* __restrict__ x,
                                   const T* __restrict__ w,
                                   const float* __restrict__ rstd,
                                   T* __restrict__ dx,
                                   float* __restrict__ dw_accum,
                                   int64_t rows, int D) {
  constexpr int V = VecIO<T>::VEC;
  const int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
  extern __shared__ float dw_s[];          // [D]
  for (int i = threadIdx.x; i < D; i += BLOCK) dw_s[i] = 0.f;
  __syncthreads();

  for (int64_t row = (int64_t)blockIdx.x * WAVES + wid; row < rows;
       row += (int64_t)gridDim.x * WAVES) {
    const T* dyr = dy + row * D;
    const T* xr = x + row * D;
    T* dxr = dx + row * D;
    const float r = rstd[row];
    float dot = 0.f;
    for (int i = lane * V; i < D; i += 64 * V) {
      float dv[VecIO<T>::VEC], xv[VecIO<T>::VEC], wv[VecIO<T>::VEC];
      VecIO<T>::load(dyr + i, dv);
      VecIO<T>::load(xr + i, xv);
      VecIO<T>::load(w + i, wv);
#pragma unroll
      for (int j = 0; j < V; ++j) dot += dv[j] * wv[j] * xv[j] * r;
    }
    dot = wave_sum(dot) / D;
    for (int i = lane * V; i < D; i += 64 * V) {
      float dv[VecIO<T>::VEC], xv[VecIO<T>::VEC], wv[VecIO<T>::VEC];
      VecIO<T>::load(dyr + i, dv);
      VecIO<T>::load(xr + i, xv);
      VecIO<T>::load(w + i, wv);
      float o[VecIO<T>::VEC];
#pragma unroll
      for (int j = 0; j < V; ++j) {
        float xhat = xv[j] * r;
        o[j] = (dv[j] * wv[j] - xhat * dot) * r;
        atomicAdd(dw_s + i + j, dv[j] * xhat);
      }
      VecIO<T>::store(dxr + i, o);
    }
  }
  __syncthreads();
  for (int i = threadIdx.x; i < D; i += BLOCK)
    if (dw_s[i] != 0.f) atomicAdd(dw_accum + i, dw_s[i]);
}

// ---------------- LayerNorm ----------------

template <typename T>
__global__ void layernorm_fwd_kernel(const T* __restrict__ x,
                                     const T* __restrict__ w,
                                     const T* __restrict__ b,
                                     T* __restrict__ y,
                                     float* __restrict__ mean_out,
                                     float* __restrict__ rstd_out,
                                     int64_t rows, int D, float eps) {
  constexpr int V = VecIO<T>::VEC;
  const int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
  for (int64_t row = (int64_t)blockIdx.x * WAVES + wid; row < rows;
       row += (int64_t)gridDim.x * WAVES) {
    const T* xr = x + row * D;
    T* yr = y + row * D;
    float s = 0.f, ss = 0.f;
    for (int i = lane * V; i < D; i += 64 * V) {
      float v[VecIO<T>::VEC];
      VecIO<T>::load(xr + i, v);
#pragma unroll
      for (int j = 0; j < V; ++j) { s += v[j]; ss += v[j] * v[j]; }
    }
    s = wave_sum(s);
    ss = wave_sum(ss);
    float mu = s / D;
    float var = ss / D - mu * mu;
    float rstd = rsqrtf(var + eps);
    if (lane == 0) { mean_out[row] = mu; rstd_out[row] = rstd; }
    for (int i = lane * V; i < D; i += 64 * V) {
      float v[VecIO<T>::VEC], wv[VecIO<T>::VEC], bv[VecIO<T>::VEC];
      VecIO<T>::load(xr + i, v);
      VecIO<T>::load(w + i, wv);
      VecIO<T>::load(b + i, bv);
#pragma unroll
      for (int j = 0; j < V; ++j)
        v[j] = (v[j] - mu) * rstd * wv[j] + bv[j];
      VecIO<T>::store(yr + i, v);
    }
  }
}

template <typename T>
__global__ void layernorm_bwd_kernel(const T* __restrict__ dy,
                                     const T* __restrict__ x,
                                     const T* __restrict__ w,
                                     const float* __restrict__ mean,
                                     const float* __restrict__ rstd,
                                     T* __restrict__ dx,
                                     float* __restrict__ dw_accum,
                                     float* __restrict__ db_accum,
                                     int64_t rows, int D) {
  constexpr int V = VecIO<T>::VEC;
  const int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
  extern __shared__ float smem_dwdb[];     // dw [D] | db [D]
  float* dw_s = smem_dwdb;
  float* db_s = smem_dwdb + D;
  for (int i = threadIdx.x; i < 2 * D; i += BLOCK) smem_dwdb[i] = 0.f;
  __syncthreads();

  for (int64_t row = (int64_t)blockIdx.x * WAVES + wid; row < rows;
       row += (int64_t)gridDim.x * WAVES) {
    const T* dyr = dy + row * D;
    const T* xr = x + row * D;
    T* dxr = dx + row * D;
    const float mu = mean[row], r = rstd[row];
    float c1 = 0.f, c2 = 0.f;
    for (int i = lane * V; i < D; i += 64 * V) {
      float dv[VecIO<T>::VEC], xv[VecIO<T>::VEC], wv[VecIO<T>::VEC];
      VecIO<T>::load(dyr + i, dv);
      VecIO<T>::load(xr + i, xv);
      VecIO<T>::load(w + i, wv);
#pragma unroll
      for (int j = 0; j < V; ++j) {
        float xhat = (xv[j] - mu) * r;
        float wdy = dv[j] * wv[j];
        c1 += wdy;
        c2 += wdy * xhat;
      }
    }
    c1 = wave_sum(c1) / D;
    c2 = wave_sum(c2) / D;
    for (int i = lane * V; i < D; i += 64 * V) {
      float dv[VecIO<T>::VEC], xv[VecIO<T>::VEC], wv[VecIO<T>::VEC];
      VecIO<T>::load(dyr + i, dv);
      VecIO<T>::load(xr + i, xv);
      VecIO<T>::load(w + i, wv);
      float o[VecIO<T>::VEC];
#pragma unroll
      for (int j = 0; j < V; ++j) {
        float xhat = (xv[j] - mu) * r;
        float wdy = dv[j] * wv[j];
        o[j] = (wdy - c1 - xhat * c2) * r;
        atomicAdd(dw_s + i + j, dv[j] * xhat);
        atomicAdd(db_s + i + j, dv[j]);
      }
      VecIO<T>::store(dxr + i, o);
    }
  }
  __syncthreads();
  for (int i = threadIdx.x; i < D; i += BLOCK) {
    if (dw_s[i] != 0.f) atomicAdd(dw_accum + i, dw_s[i]);
    if (db_s[i] != 0.f) atomicAdd(db_accum + i, db_s[i]);
  }
}

inline int row_grid(int64_t rows) {
  // wave-per-row: blocks cover WAVES rows each; >> 256 workgroups fills
  // the 256 CUs / 8 XCDs
  int64_t g = (rows + WAVES - 1) / WAVES;
  return (int)std::min<int64_t>(std::max<int64_t>(g, 1), 8192);
}

inline int bwd_grid(int64_t rows) {
  // atomic flush costs grid*D; 2048 blocks (8192 waves) still fills the
  // chip while keeping the flush ~2048*D*4B
  int64_t g = (rows + WAVES - 1) / WAVES;
  return (int)std::min<int64_t>(std::max<int64_t>(g, 1), 2048);
}

}  // namespace

std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w,
                                       double eps) {
  const int D = x.size(-1);
  const int64_t rows = x.numel() / D;
  TORCH_CHECK(D % 8 == 0, "rmsnorm: D must be a multiple of 8");
  auto y = torch::empty_like(x);
  auto rstd = torch::empty({rows}, x.options().dtype(at::kFloat));
  auto stream = hetu_current_stream();
  DISPATCH_FLOAT(x, "rmsnorm_fwd", [&] {
    hipLaunchKernelGGL(rmsnorm_fwd_kernel<scalar_t>, dim3(row_grid(rows)),
                       dim3(BLOCK), 0, stream,
                       (const scalar_t*)x.data_ptr(),
                       (const scalar_t*)w.data_ptr(),
                       (scalar_t*)y.data_ptr(), rstd.data_ptr<float>(),
                       rows, D, (float)eps);
  });
  return {y, rstd.view(at::IntArrayRef(x.sizes().begin(), x.sizes().end() - 1))};
}

std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x,
                                       torch::Tensor w, torch::Tensor rstd) {
  const int D = x.size(-1);
  const int64_t rows = x.numel() / D;
  TORCH_CHECK(D <= MAX_D, "rmsnorm_bwd: D too large for the LDS image");
  auto dx = torch::empty_like(x);
  auto dw32 = torch::zeros({D}, x.options().dtype(at::kFloat));
  auto stream = hetu_current_stream();
  DISPATCH_FLOAT(x, "rmsnorm_bwd", [&] {
    hipLaunchKernelGGL(rmsnorm_bwd_kernel<scalar_t>,
                       dim3(bwd_grid(rows)), dim3(BLOCK),
                       (size_t)D * sizeof(float), stream,
                       (const scalar_t*)dy.data_ptr(),
                       (const scalar_t*)x.data_ptr(),
                       (const scalar_t*)w.data_ptr(),
                       rstd.data_ptr<float>(),
                       (scalar_t*)dx.data_ptr(), dw32.data_ptr<float>(),
                       rows, D);
  });
  return {dx, dw32.to(w.scalar_type())};
}

std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor w,
                                         torch::Tensor b, double eps) {
  const int D = x.size(-1);
  const int64_t rows = x.numel() / D;
  TORCH_CHECK(D % 8 == 0, "layernorm: D must be a multiple of 8");
  auto y = torch::empty_like(x);
  auto mean = torch::empty({rows}, x.options().dtype(at::kFloat));
  auto rstd = torch::empty({rows}, x.options().dtype(at::kFloat));
  auto stream = hetu_current_stream();
  DISPATCH_FLOAT(x, "layernorm_fwd", [&] {
    hipLaunchKernelGGL(layernorm_fwd_kernel<scalar_t>,
                       dim3(row_grid(rows)), dim3(BLOCK), 0, stream,
                       (const scalar_t*)x.data_ptr(),
                       (const scalar_t*)w.data_ptr(),
                       (const scalar_t*)b.data_ptr(),
                       (scalar_t*)y.data_ptr(), mean.data_ptr<float>(),
                       rstd.data_ptr<float>(), rows, D, (float)eps);
  });
  auto row_sizes = at::IntArrayRef(x.sizes().begin(), x.sizes().end() - 1);
  return {y, mean.view(row_sizes), rstd.view(row_sizes)};
}

std::vector<torch::Tensor> layernorm_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor w, torch::Tensor mean,
                                         torch::Tensor rstd) {
  const int D = x.size(-1);
  const int64_t rows = x.numel() / D;
  TORCH_CHECK(D <= MAX_D / 2, "layernorm_bwd: D too large for LDS image");
  auto dx = torch::empty_like(x);
  auto dw32 = torch::zeros({D}, x.options().dtype(at::kFloat));
  auto db32 = torch::zeros({D}, x.options().dtype(at::kFloat));
  auto stream = hetu_current_stream();
  DISPATCH_FLOAT(x, "layernorm_bwd", [&] {
    hipLaunchKernelGGL(layernorm_bwd_kernel<scalar_t>,
                       dim3(bwd_grid(rows)), dim3(BLOCK),
                       (size_t)2 * D * sizeof(float), stream,
                       (const scalar_t*)dy.data_ptr(),
                       (const scalar_t*)x.data_ptr(),
                       (const scalar_t*)w.data_ptr(),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       (scalar_t*)dx.data_ptr(), dw32.data_ptr<float>(),
                       db32.data_ptr<float>(), rows, D);
  });
  return {dx, dw32.to(w.scalar_type()), db32.to(w.scalar_type())};
}
