// Fused RMSNorm / LayerNorm forward+backward for gfx950.
//
// Design (MI355X): one 256-thread block per row (grid-stride over rows),
// wave64 shuffle reductions, bf16 loads vectorized as ushort8 (16 B/lane —
// scalar bf16 loads halve HBM throughput on CDNA4). fp32 stats; dw/db
// accumulated per-block in registers then one atomicAdd per element.
// Capability parity: reference RMSNorm.cu / FusedLayerNorm.cu:455-760.
#include <torch/extension.h>
#include "ext_stream.h"
#include "common.h"

namespace {

constexpr int BLOCK = 256;

// ---------------- RMSNorm ----------------

template <typename T, bool RES>
__global__ void rmsnorm_fwd_kernel(const T* __restrict__ x,
                                   const T* __restrict__ resid,
                                   const T* __restrict__ w,
                                   T* __restrict__ y,
                                   T* __restrict__ sum_out,
                                   float* __restrict__ rstd_out,
                                   int64_t rows, int D, float eps) {
  constexpr int V = VecIO<T>::VEC;
  __shared__ float smem[16];
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* xr = x + row * D;
    const T* rr = RES ? resid + row * D : nullptr;
    T* yr = y + row * D;
    T* sr = RES ? sum_out + row * D : nullptr;
    float ss = 0.f;
    for (int i = threadIdx.x * V; i < D; i += BLOCK * V) {
      float v[VecIO<T>::VEC];
      VecIO<T>::load(xr + i, v);
      if (RES) {
        float rv[VecIO<T>::VEC];
        VecIO<T>::load(rr + i, rv);
#pragma unroll
        for (int j = 0; j < V; ++j) v[j] += rv[j];
        VecIO<T>::store(sr + i, v);
      }
#pragma unroll
      for (int j = 0; j < V; ++j) ss += v[j] * v[j];
    }
    ss = block_sum(ss, smem);
    float rstd = rsqrtf(ss / D + eps);
    if (threadIdx.x == 0) rstd_out[row] = rstd;
    const T* sum_r = RES ? sr : xr;
    for (int i = threadIdx.x * V; i < D; i += BLOCK * V) {
      float v[VecIO<T>::VEC], wv[VecIO<T>::VEC];
      VecIO<T>::load(sum_r + i, v);
      VecIO<T>::load(w + i, wv);
#pragma unroll
      for (int j = 0; j < V; ++j) v[j] = v[j] * rstd * wv[j];
      VecIO<T>::store(yr + i, v);
    }
  }
}

// dw accumulated per-thread in registers across this block's rows; ONE
// atomicAdd per element per block at the end (grid is capped so total
// atomic traffic is ~grid*D, not rows*D).
template <typename T, int NCHUNK>
__global__ void rmsnorm_bwd_kernel(const T* __restrict__ dy,
                                   const T* __restrict__ x,
                                   const T* __restrict__ w,
                                   const float* __restrict__ rstd,
                                   T* __restrict__ dx,
                                   float* __restrict__ dw_accum,
                                   int64_t rows, int D) {
  constexpr int V = VecIO<T>::VEC;
  __shared__ float smem[16];
  float dwacc[NCHUNK][V];
#pragma unroll
  for (int c = 0; c < NCHUNK; ++c)
#pragma unroll
    for (int j = 0; j < V; ++j) dwacc[c][j] = 0.f;

  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* dyr = dy + row * D;
    const T* xr = x + row * D;
    T* dxr = dx + row * D;
    const float r = rstd[row];
    float dot = 0.f;
    for (int i = threadIdx.x * V; i < D; i += BLOCK * V) {
      float dv[VecIO<T>::VEC], xv[VecIO<T>::VEC], wv[VecIO<T>::VEC];
      VecIO<T>::load(dyr + i, dv);
      VecIO<T>::load(xr + i, xv);
      VecIO<T>::load(w + i, wv);
#pragma unroll
      for (int j = 0; j < V; ++j) dot += dv[j] * wv[j] * xv[j] * r;
    }
    dot = block_sum(dot, smem) / D;
    int c = 0;
    for (int i = threadIdx.x * V; i < D; i += BLOCK * V, ++c) {
      float dv[VecIO<T>::VEC], xv[VecIO<T>::VEC], wv[VecIO<T>::VEC];
      VecIO<T>::load(dyr + i, dv);
      VecIO<T>::load(xr + i, xv);
      VecIO<T>::load(w + i, wv);
      float o[VecIO<T>::VEC];
#pragma unroll
      for (int j = 0; j < V; ++j) {
        float xhat = xv[j] * r;
        o[j] = (dv[j] * wv[j] - xhat * dot) * r;
        dwacc[c][j] += dv[j] * xhat;
      }
      VecIO<T>::store(dxr + i, o);
    }
  }
  {
    int c = 0;
    for (int i = threadIdx.x * V; i < D; i += BLOCK * V, ++c)
#pragma unroll
      for (int j = 0; j < V; ++j)
        atomicAdd(dw_accum + i + j, dwacc[c][j]);
  }
}

// ---------------- LayerNorm ----------------

// RES: fuse the transformer residual add into the norm — loads x+resid,
// writes the sum (feeds the next residual) alongside y, killing the
// standalone elementwise add kernel (the at::native tail).
template <typename T, bool RES>
__global__ void layernorm_fwd_kernel(const T* __restrict__ x,
                                     const T* __restrict__ resid,
                                     const T* __restrict__ w,
                                     const T* __restrict__ b,
                                     T* __restrict__ y,
                                     T* __restrict__ sum_out,
                                     float* __restrict__ mean_out,
                                     float* __restrict__ rstd_out,
                                     int64_t rows, int D, float eps) {
  constexpr int V = VecIO<T>::VEC;
  __shared__ float smem[16];
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* xr = x + row * D;
    const T* rr = RES ? resid + row * D : nullptr;
    T* yr = y + row * D;
    T* sr = RES ? sum_out + row * D : nullptr;
    float s = 0.f, ss = 0.f;
    for (int i = threadIdx.x * V; i < D; i += BLOCK * V) {
      float v[VecIO<T>::VEC];
      VecIO<T>::load(xr + i, v);
      if (RES) {
        float rv[VecIO<T>::VEC];
        VecIO<T>::load(rr + i, rv);
#pragma unroll
        for (int j = 0; j < V; ++j) v[j] += rv[j];
        VecIO<T>::store(sr + i, v);
      }
#pragma unroll
      for (int j = 0; j < V; ++j) { s += v[j]; ss += v[j] * v[j]; }
    }
    s = block_sum(s, smem);
    ss = block_sum(ss, smem);
    float mu = s / D;
    float var = ss / D - mu * mu;
    float rstd = rsqrtf(var + eps);
    if (threadIdx.x == 0) { mean_out[row] = mu; rstd_out[row] = rstd; }
    const T* sum_r = RES ? sr : xr;
    for (int i = threadIdx.x * V; i < D; i += BLOCK * V) {
      float v[VecIO<T>::VEC], wv[VecIO<T>::VEC], bv[VecIO<T>::VEC];
      VecIO<T>::load(sum_r + i, v);
      VecIO<T>::load(w + i, wv);
      VecIO<T>::load(b + i, bv);
#pragma unroll
      for (int j = 0; j < V; ++j)
        v[j] = (v[j] - mu) * rstd * wv[j] + bv[j];
      VecIO<T>::store(yr + i, v);
    }
  }
}

template <typename T, int NCHUNK>
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
  __shared__ float smem[16];
  float dwacc[NCHUNK][V], dbacc[NCHUNK][V];
#pragma unroll
  for (int c = 0; c < NCHUNK; ++c)
#pragma unroll
    for (int j = 0; j < V; ++j) { dwacc[c][j] = 0.f; dbacc[c][j] = 0.f; }
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* dyr = dy + row * D;
    const T* xr = x + row * D;
    T* dxr = dx + row * D;
    const float mu = mean[row], r = rstd[row];
    float c1 = 0.f, c2 = 0.f;
    for (int i = threadIdx.x * V; i < D; i += BLOCK * V) {
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
    c1 = block_sum(c1, smem) / D;
    c2 = block_sum(c2, smem) / D;
    int c = 0;
    for (int i = threadIdx.x * V; i < D; i += BLOCK * V, ++c) {
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
        dwacc[c][j] += dv[j] * xhat;
        dbacc[c][j] += dv[j];
      }
      VecIO<T>::store(dxr + i, o);
    }
  }
  {
    int c = 0;
    for (int i = threadIdx.x * V; i < D; i += BLOCK * V, ++c)
#pragma unroll
      for (int j = 0; j < V; ++j) {
        atomicAdd(dw_accum + i + j, dwacc[c][j]);
        atomicAdd(db_accum + i + j, dbacc[c][j]);
      }
  }
}

inline int row_grid(int64_t rows) {
  // >> 256 workgroups to fill 256 CUs / 8 XCDs; cap and grid-stride
  int64_t g = rows < 8192 ? rows : 8192;
  return (int)(g > 0 ? g : 1);
}

}  // namespace

std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w,
                                       double eps) {
  const int D = x.size(-1);
  const int64_t rows = x.numel() / D;
  TORCH_CHECK(D % 8 == 0, "rmsnorm: D must be a multiple of 8");
  TORCH_CHECK(w.scalar_type() == x.scalar_type(),
              "rmsnorm: w dtype must match x");
  auto y = torch::empty_like(x);
  auto rstd = torch::empty({rows}, x.options().dtype(at::kFloat));
  auto stream = hetu_current_stream();
  DISPATCH_FLOAT(x, "rmsnorm_fwd", [&] {
    hipLaunchKernelGGL((rmsnorm_fwd_kernel<scalar_t, false>),
                       dim3(row_grid(rows)), dim3(BLOCK), 0, stream,
                       (const scalar_t*)x.data_ptr(),
                       (const scalar_t*)nullptr,
                       (const scalar_t*)w.data_ptr(),
                       (scalar_t*)y.data_ptr(), (scalar_t*)nullptr,
                       rstd.data_ptr<float>(), rows, D, (float)eps);
  });
  return {y, rstd.view(at::IntArrayRef(x.sizes().begin(), x.sizes().end() - 1))};
}

std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x,
                                       torch::Tensor w, torch::Tensor rstd) {
  const int D = x.size(-1);
  TORCH_CHECK(w.scalar_type() == x.scalar_type() &&
              dy.scalar_type() == x.scalar_type(),
              "rmsnorm_bwd: dy/w dtype must match x");
  const int64_t rows = x.numel() / D;
  auto dx = torch::empty_like(x);
  auto dw32 = torch::zeros({D}, x.options().dtype(at::kFloat));
  auto stream = hetu_current_stream();
  int grid = (int)std::min<int64_t>(rows, 1024);
  DISPATCH_FLOAT(x, "rmsnorm_bwd", [&] {
    constexpr int V = VecIO<scalar_t>::VEC;
    const int nchunk = (D + BLOCK * V - 1) / (BLOCK * V);
    auto launch = [&](auto tag) {
      constexpr int NC = decltype(tag)::value;
      hipLaunchKernelGGL((rmsnorm_bwd_kernel<scalar_t, NC>), dim3(grid),
                         dim3(BLOCK), 0, stream,
                         (const scalar_t*)dy.data_ptr(),
                         (const scalar_t*)x.data_ptr(),
                         (const scalar_t*)w.data_ptr(),
                         rstd.data_ptr<float>(),
                         (scalar_t*)dx.data_ptr(), dw32.data_ptr<float>(),
                         rows, D);
    };
    if (nchunk <= 1) launch(std::integral_constant<int, 1>{});
    else if (nchunk <= 2) launch(std::integral_constant<int, 2>{});
    else if (nchunk <= 4) launch(std::integral_constant<int, 4>{});
    else if (nchunk <= 8) launch(std::integral_constant<int, 8>{});
    else TORCH_CHECK(false, "rmsnorm_bwd: D too large");
  });
  return {dx, dw32.to(w.scalar_type())};
}

std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor w,
                                         torch::Tensor b, double eps) {
  const int D = x.size(-1);
  const int64_t rows = x.numel() / D;
  TORCH_CHECK(D % 8 == 0, "layernorm: D must be a multiple of 8");
  TORCH_CHECK(w.scalar_type() == x.scalar_type() &&
              b.scalar_type() == x.scalar_type(),
              "layernorm: w/b dtype must match x");
  auto y = torch::empty_like(x);
  auto mean = torch::empty({rows}, x.options().dtype(at::kFloat));
  auto rstd = torch::empty({rows}, x.options().dtype(at::kFloat));
  auto stream = hetu_current_stream();
  DISPATCH_FLOAT(x, "layernorm_fwd", [&] {
    hipLaunchKernelGGL((layernorm_fwd_kernel<scalar_t, false>),
                       dim3(row_grid(rows)), dim3(BLOCK), 0, stream,
                       (const scalar_t*)x.data_ptr(),
                       (const scalar_t*)nullptr,
                       (const scalar_t*)w.data_ptr(),
                       (const scalar_t*)b.data_ptr(),
                       (scalar_t*)y.data_ptr(), (scalar_t*)nullptr,
                       mean.data_ptr<float>(),
                       rstd.data_ptr<float>(), rows, D, (float)eps);
  });
  auto row_sizes = at::IntArrayRef(x.sizes().begin(), x.sizes().end() - 1);
  return {y, mean.view(row_sizes), rstd.view(row_sizes)};
}

std::vector<torch::Tensor> layernorm_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor w, torch::Tensor mean,
                                         torch::Tensor rstd) {
  const int D = x.size(-1);
  TORCH_CHECK(w.scalar_type() == x.scalar_type() &&
              dy.scalar_type() == x.scalar_type(),
              "layernorm_bwd: dy/w dtype must match x");
  const int64_t rows = x.numel() / D;
  auto dx = torch::empty_like(x);
  auto dw32 = torch::zeros({D}, x.options().dtype(at::kFloat));
  auto db32 = torch::zeros({D}, x.options().dtype(at::kFloat));
  auto stream = hetu_current_stream();
  int grid = (int)std::min<int64_t>(rows, 1024);
  DISPATCH_FLOAT(x, "layernorm_bwd", [&] {
    constexpr int V = VecIO<scalar_t>::VEC;
    const int nchunk = (D + BLOCK * V - 1) / (BLOCK * V);
    auto launch = [&](auto tag) {
      constexpr int NC = decltype(tag)::value;
      hipLaunchKernelGGL((layernorm_bwd_kernel<scalar_t, NC>), dim3(grid),
                         dim3(BLOCK), 0, stream,
                         (const scalar_t*)dy.data_ptr(),
                         (const scalar_t*)x.data_ptr(),
                         (const scalar_t*)w.data_ptr(),
                         mean.data_ptr<float>(), rstd.data_ptr<float>(),
                         (scalar_t*)dx.data_ptr(), dw32.data_ptr<float>(),
                         db32.data_ptr<float>(), rows, D);
    };
    if (nchunk <= 1) launch(std::integral_constant<int, 1>{});
    else if (nchunk <= 2) launch(std::integral_constant<int, 2>{});
    else if (nchunk <= 4) launch(std::integral_constant<int, 4>{});
    else if (nchunk <= 8) launch(std::integral_constant<int, 8>{});
    else TORCH_CHECK(false, "layernorm_bwd: D too large");
  });
  return {dx, dw32.to(w.scalar_type()), db32.to(w.scalar_type())};
}

// ---------------------------------------------------------------------------
// v2 backward: split dx (wave-per-row, shuffle reductions, no atomics)
// from dw/db (column-tile reduction kernel).  The fused block-per-row
// version above is grid-capped at 1024 blocks by its dw atomic flush and
// measured 5.7x off the bandwidth bound at the 7B shape.
// ---------------------------------------------------------------------------
namespace {

template <typename T, bool LN>
__global__ void norm_bwd_dx_kernel(const T* __restrict__ dy,
                                   const T* __restrict__ x,
                                   const T* __restrict__ w,
                                   const float* __restrict__ mean,
                                   const float* __restrict__ rstd,
                                   const T* __restrict__ dresid,
                                   T* __restrict__ dx,
                                   int64_t rows, int D) {
  constexpr int V = VecIO<T>::VEC;
  const int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
  constexpr int WV = 256 / 64;
  for (int64_t row = (int64_t)blockIdx.x * WV + wid; row < rows;
       row += (int64_t)gridDim.x * WV) {
    const T* dyr = dy + row * D;
    const T* xr = x + row * D;
    T* dxr = dx + row * D;
    const float mu = LN ? mean[row] : 0.f;
    const float r = rstd[row];
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
        o[j] = LN ? (wdy - c1 - xhat * c2) * r : (wdy - xhat * c2) * r;
      }
      if (dresid != nullptr) {
        // fused residual-grad accumulation (the adjoint of add+LN):
        // dx = dLN/dsum + dresid, no standalone add kernel in backward
        float rv[VecIO<T>::VEC];
        VecIO<T>::load(dresid + row * D + i, rv);
#pragma unroll
        for (int j = 0; j < V; ++j) o[j] += rv[j];
      }
      VecIO<T>::store(dxr + i, o);
    }
  }
}

// dw/db: column sums of dy*xhat / dy over a [ROWS_CHUNK x 512-col] tile
// per block; each thread owns 2 adjacent columns (one dword load per row
// per tensor), one atomicAdd pair per column at the end.
template <typename T, bool LN>
__global__ void norm_bwd_dwdb_kernel(const T* __restrict__ dy,
                                     const T* __restrict__ x,
                                     const float* __restrict__ mean,
                                     const float* __restrict__ rstd,
                                     float* __restrict__ dw_accum,
                                     float* __restrict__ db_accum,
                                     int64_t rows, int D, int rows_chunk) {
  const int c0 = blockIdx.x * 512 + threadIdx.x * 2;   // my column pair
  if (c0 >= D) return;
  const int64_t r0 = (int64_t)blockIdx.y * rows_chunk;
  const int64_t r1 = min((int64_t)rows, r0 + rows_chunk);
  float dw0 = 0.f, dw1 = 0.f, db0 = 0.f, db1 = 0.f;
  for (int64_t row = r0; row < r1; ++row) {
    const float mu = LN ? mean[row] : 0.f;
    const float r = rstd[row];
    float d0, d1, x0, x1;
    if (sizeof(T) == 2) {
      // bf16: one dword load covers the column pair
      union { unsigned u; T t[2]; } dv, xv;
      dv.u = *reinterpret_cast<const unsigned*>(dy + row * D + c0);
      xv.u = *reinterpret_cast<const unsigned*>(x + row * D + c0);
      d0 = (float)dv.t[0]; d1 = (float)dv.t[1];
      x0 = (float)xv.t[0]; x1 = (float)xv.t[1];
    } else {
      d0 = (float)dy[row * D + c0]; d1 = (float)dy[row * D + c0 + 1];
      x0 = (float)x[row * D + c0]; x1 = (float)x[row * D + c0 + 1];
    }
    dw0 += d0 * ((x0 - mu) * r);
    dw1 += d1 * ((x1 - mu) * r);
    if (LN) { db0 += d0; db1 += d1; }
  }
  atomicAdd(dw_accum + c0, dw0);
  atomicAdd(dw_accum + c0 + 1, dw1);
  if (LN) {
    atomicAdd(db_accum + c0, db0);
    atomicAdd(db_accum + c0 + 1, db1);
  }
}

inline int v2_grid(int64_t rows) {
  int64_t g = (rows + 3) / 4;
  return (int)std::min<int64_t>(std::max<int64_t>(g, 1), 8192);
}

template <typename T, bool LN>
void norm_bwd_v2_launch(const T* dy, const T* x, const T* w,
                        const float* mean, const float* rstd, T* dx,
                        float* dw, float* db, int64_t rows, int D,
                        hipStream_t stream, const T* dresid = nullptr) {
  hipLaunchKernelGGL((norm_bwd_dx_kernel<T, LN>), dim3(v2_grid(rows)),
                     dim3(256), 0, stream, dy, x, w, mean, rstd,
                     dresid, dx, rows, D);
  // pick rows_chunk so the grid lands around ~2048 blocks
  int col_blocks = (D + 511) / 512;
  int target = (2048 + col_blocks - 1) / col_blocks;
  int rows_chunk = (int)std::max<int64_t>((rows + target - 1) / target, 64);
  int row_blocks = (int)((rows + rows_chunk - 1) / rows_chunk);
  hipLaunchKernelGGL((norm_bwd_dwdb_kernel<T, LN>),
                     dim3(col_blocks, row_blocks), dim3(256), 0, stream,
                     dy, x, mean, rstd, dw, db, rows, D, rows_chunk);
}

}  // namespace

std::vector<torch::Tensor> layernorm_bwd2(torch::Tensor dy, torch::Tensor x,
                                          torch::Tensor w,
                                          torch::Tensor mean,
                                          torch::Tensor rstd) {
  const int D = x.size(-1);
  TORCH_CHECK(w.scalar_type() == x.scalar_type() &&
              dy.scalar_type() == x.scalar_type(),
              "layernorm_bwd2: dy/w dtype must match x");
  const int64_t rows = x.numel() / D;
  TORCH_CHECK(D % 2 == 0);
  auto dx = torch::empty_like(x);
  auto dw32 = torch::zeros({D}, x.options().dtype(at::kFloat));
  auto db32 = torch::zeros({D}, x.options().dtype(at::kFloat));
  auto stream = hetu_current_stream();
  DISPATCH_FLOAT(x, "layernorm_bwd2", [&] {
    norm_bwd_v2_launch<scalar_t, true>(
        (const scalar_t*)dy.data_ptr(), (const scalar_t*)x.data_ptr(),
        (const scalar_t*)w.data_ptr(), mean.data_ptr<float>(),
        rstd.data_ptr<float>(), (scalar_t*)dx.data_ptr(),
        dw32.data_ptr<float>(), db32.data_ptr<float>(), rows, D, stream);
  });
  return {dx, dw32.to(w.scalar_type()), db32.to(w.scalar_type())};
}

std::vector<torch::Tensor> rmsnorm_bwd2(torch::Tensor dy, torch::Tensor x,
                                        torch::Tensor w,
                                        torch::Tensor rstd) {
  const int D = x.size(-1);
  TORCH_CHECK(w.scalar_type() == x.scalar_type() &&
              dy.scalar_type() == x.scalar_type(),
              "rmsnorm_bwd2: dy/w dtype must match x");
  const int64_t rows = x.numel() / D;
  TORCH_CHECK(D % 2 == 0);
  auto dx = torch::empty_like(x);
  auto dw32 = torch::zeros({D}, x.options().dtype(at::kFloat));
  auto stream = hetu_current_stream();
  DISPATCH_FLOAT(x, "rmsnorm_bwd2", [&] {
    norm_bwd_v2_launch<scalar_t, false>(
        (const scalar_t*)dy.data_ptr(), (const scalar_t*)x.data_ptr(),
        (const scalar_t*)w.data_ptr(), nullptr, rstd.data_ptr<float>(),
        (scalar_t*)dx.data_ptr(), dw32.data_ptr<float>(), nullptr, rows,
        D, stream);
  });
  return {dx, dw32.to(w.scalar_type())};
}


// ---- fused residual-add + LayerNorm (transformer pre-norm hot path) ----
// fwd: s = x + resid; y = LN(s).  Returns {y, s, mean, rstd}.
std::vector<torch::Tensor> layernorm_fwd_res(torch::Tensor x,
                                             torch::Tensor resid,
                                             torch::Tensor w,
                                             torch::Tensor b, double eps) {
  const int D = x.size(-1);
  const int64_t rows = x.numel() / D;
  TORCH_CHECK(D % 8 == 0, "layernorm: D must be a multiple of 8");
  TORCH_CHECK(w.scalar_type() == x.scalar_type() &&
              b.scalar_type() == x.scalar_type() &&
              resid.scalar_type() == x.scalar_type(),
              "layernorm_fwd_res: dtype mismatch");
  TORCH_CHECK(resid.numel() == x.numel());
  auto y = torch::empty_like(x);
  auto s_out = torch::empty_like(x);
  auto mean = torch::empty({rows}, x.options().dtype(at::kFloat));
  auto rstd = torch::empty({rows}, x.options().dtype(at::kFloat));
  auto stream = hetu_current_stream();
  DISPATCH_FLOAT(x, "layernorm_fwd_res", [&] {
    hipLaunchKernelGGL((layernorm_fwd_kernel<scalar_t, true>),
                       dim3(row_grid(rows)), dim3(BLOCK), 0, stream,
                       (const scalar_t*)x.data_ptr(),
                       (const scalar_t*)resid.data_ptr(),
                       (const scalar_t*)w.data_ptr(),
                       (const scalar_t*)b.data_ptr(),
                       (scalar_t*)y.data_ptr(),
                       (scalar_t*)s_out.data_ptr(),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       rows, D, (float)eps);
  });
  auto row_sizes = at::IntArrayRef(x.sizes().begin(), x.sizes().end() - 1);
  return {y, s_out, mean.view(row_sizes), rstd.view(row_sizes)};
}

// bwd: given dy (grad of y) and optional ds_ext (grad of s from its other
// consumers), returns {dsum = dLN/ds + ds_ext, dw, db} — dsum is the grad
// of BOTH x and resid.
std::vector<torch::Tensor> layernorm_bwd2_res(torch::Tensor dy,
                                              torch::Tensor s,
                                              torch::Tensor w,
                                              torch::Tensor mean,
                                              torch::Tensor rstd,
                                              torch::Tensor ds_ext) {
  const int D = s.size(-1);
  TORCH_CHECK(w.scalar_type() == s.scalar_type() &&
              dy.scalar_type() == s.scalar_type(),
              "layernorm_bwd2_res: dy/w dtype must match s");
  const int64_t rows = s.numel() / D;
  TORCH_CHECK(D % 2 == 0);
  auto dx = torch::empty_like(s);
  auto dw32 = torch::zeros({D}, s.options().dtype(at::kFloat));
  auto db32 = torch::zeros({D}, s.options().dtype(at::kFloat));
  auto stream = hetu_current_stream();
  const bool has_ext = ds_ext.defined() && ds_ext.numel() > 0;
  DISPATCH_FLOAT(s, "layernorm_bwd2_res", [&] {
    norm_bwd_v2_launch<scalar_t, true>(
        (const scalar_t*)dy.data_ptr(), (const scalar_t*)s.data_ptr(),
        (const scalar_t*)w.data_ptr(), mean.data_ptr<float>(),
        rstd.data_ptr<float>(), (scalar_t*)dx.data_ptr(),
        dw32.data_ptr<float>(), db32.data_ptr<float>(), rows, D, stream,
        has_ext ? (const scalar_t*)ds_ext.data_ptr() : nullptr);
  });
  return {dx, dw32.to(w.scalar_type()), db32.to(w.scalar_type())};
}


// ---- fused residual-add + RMSNorm (Llama pre-norm hot path) ------------
std::vector<torch::Tensor> rmsnorm_fwd_res(torch::Tensor x,
                                           torch::Tensor resid,
                                           torch::Tensor w, double eps) {
  const int D = x.size(-1);
  const int64_t rows = x.numel() / D;
  TORCH_CHECK(D % 8 == 0, "rmsnorm: D must be a multiple of 8");
  TORCH_CHECK(w.scalar_type() == x.scalar_type() &&
              resid.scalar_type() == x.scalar_type(),
              "rmsnorm_fwd_res: dtype mismatch");
  auto y = torch::empty_like(x);
  auto s_out = torch::empty_like(x);
  auto rstd = torch::empty({rows}, x.options().dtype(at::kFloat));
  auto stream = hetu_current_stream();
  DISPATCH_FLOAT(x, "rmsnorm_fwd_res", [&] {
    hipLaunchKernelGGL((rmsnorm_fwd_kernel<scalar_t, true>),
                       dim3(row_grid(rows)), dim3(BLOCK), 0, stream,
                       (const scalar_t*)x.data_ptr(),
                       (const scalar_t*)resid.data_ptr(),
                       (const scalar_t*)w.data_ptr(),
                       (scalar_t*)y.data_ptr(),
                       (scalar_t*)s_out.data_ptr(),
                       rstd.data_ptr<float>(), rows, D, (float)eps);
  });
  auto row_sizes = at::IntArrayRef(x.sizes().begin(), x.sizes().end() - 1);
  return {y, s_out, rstd.view(row_sizes)};
}

std::vector<torch::Tensor> rmsnorm_bwd2_res(torch::Tensor dy,
                                            torch::Tensor s,
                                            torch::Tensor w,
                                            torch::Tensor rstd,
                                            torch::Tensor ds_ext) {
  const int D = s.size(-1);
  TORCH_CHECK(w.scalar_type() == s.scalar_type() &&
              dy.scalar_type() == s.scalar_type(),
              "rmsnorm_bwd2_res: dy/w dtype must match s");
  const int64_t rows = s.numel() / D;
  TORCH_CHECK(D % 2 == 0);
  auto dx = torch::empty_like(s);
  auto dw32 = torch::zeros({D}, s.options().dtype(at::kFloat));
  auto stream = hetu_current_stream();
  const bool has_ext = ds_ext.defined() && ds_ext.numel() > 0;
  DISPATCH_FLOAT(s, "rmsnorm_bwd2_res", [&] {
    norm_bwd_v2_launch<scalar_t, false>(
        (const scalar_t*)dy.data_ptr(), (const scalar_t*)s.data_ptr(),
        (const scalar_t*)w.data_ptr(), nullptr, rstd.data_ptr<float>(),
        (scalar_t*)dx.data_ptr(), dw32.data_ptr<float>(), nullptr, rows,
        D, stream,
        has_ext ? (const scalar_t*)ds_ext.data_ptr() : nullptr);
  });
  return {dx, dw32.to(w.scalar_type())};
}
