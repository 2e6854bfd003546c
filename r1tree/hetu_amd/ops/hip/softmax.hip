// Row softmax fwd/bwd (last dim). Reference: hetu/impl/kernel/Softmax.cu.
// Block per row, online max+sum in fp32, bf16x8 vector IO.
#include <torch/extension.h>
#include "ext_stream.h"
#include "common.h"

namespace {
constexpr int BLOCK = 256;

template <typename T>
__global__ void softmax_fwd_kernel(const T* __restrict__ x,
                                   T* __restrict__ y, int64_t rows, int D) {
  constexpr int V = VecIO<T>::VEC;
  __shared__ float smem[16];
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* xr = x + row * D;
    T* yr = y + row * D;
    float mx = -INFINITY, sum = 0.f;
    for (int i = threadIdx.x * V; i < D; i += BLOCK * V) {
      float v[VecIO<T>::VEC];
      VecIO<T>::load(xr + i, v);
#pragma unroll
      for (int j = 0; j < V; ++j) mx = fmaxf(mx, v[j]);
    }
    mx = block_max(mx, smem);
    for (int i = threadIdx.x * V; i < D; i += BLOCK * V) {
      float v[VecIO<T>::VEC];
      VecIO<T>::load(xr + i, v);
#pragma unroll
      for (int j = 0; j < V; ++j) sum += __expf(v[j] - mx);
    }
    sum = block_sum(sum, smem);
    float inv = 1.f / sum;
    for (int i = threadIdx.x * V; i < D; i += BLOCK * V) {
      float v[VecIO<T>::VEC];
      VecIO<T>::load(xr + i, v);
#pragma unroll
      for (int j = 0; j < V; ++j) v[j] = __expf(v[j] - mx) * inv;
      VecIO<T>::store(yr + i, v);
    }
  }
}

template <typename T>
__global__ void softmax_bwd_kernel(const T* __restrict__ dy,
                                   const T* __restrict__ y,
                                   T* __restrict__ dx, int64_t rows, int D) {
  constexpr int V = VecIO<T>::VEC;
  __shared__ float smem[16];
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* dyr = dy + row * D;
    const T* yr = y + row * D;
    T* dxr = dx + row * D;
    float dot = 0.f;
    for (int i = threadIdx.x * V; i < D; i += BLOCK * V) {
      float g[VecIO<T>::VEC], v[VecIO<T>::VEC];
      VecIO<T>::load(dyr + i, g);
      VecIO<T>::load(yr + i, v);
#pragma unroll
      for (int j = 0; j < V; ++j) dot += g[j] * v[j];
    }
    dot = block_sum(dot, smem);
    for (int i = threadIdx.x * V; i < D; i += BLOCK * V) {
      float g[VecIO<T>::VEC], v[VecIO<T>::VEC];
      VecIO<T>::load(dyr + i, g);
      VecIO<T>::load(yr + i, v);
#pragma unroll
      for (int j = 0; j < V; ++j) g[j] = (g[j] - dot) * v[j];
      VecIO<T>::store(dxr + i, g);
    }
  }
}
}  // namespace

torch::Tensor softmax_fwd(torch::Tensor x) {
  const int D = x.size(-1);
  const int64_t rows = x.numel() / D;
  TORCH_CHECK(D % 8 == 0, "softmax: D must be a multiple of 8");
  auto y = torch::empty_like(x);
  auto stream = hetu_current_stream();
  int grid = (int)std::min<int64_t>(rows, 8192);
  DISPATCH_FLOAT(x, "softmax_fwd", [&] {
    hipLaunchKernelGGL(softmax_fwd_kernel<scalar_t>, dim3(grid), dim3(BLOCK),
                       0, stream, (const scalar_t*)x.data_ptr(),
                       (scalar_t*)y.data_ptr(), rows, D);
  });
  return y;
}

torch::Tensor softmax_bwd(torch::Tensor dy, torch::Tensor y) {
  const int D = y.size(-1);
  const int64_t rows = y.numel() / D;
  auto dx = torch::empty_like(dy);
  auto stream = hetu_current_stream();
  int grid = (int)std::min<int64_t>(rows, 8192);
  DISPATCH_FLOAT(y, "softmax_bwd", [&] {
    hipLaunchKernelGGL(softmax_bwd_kernel<scalar_t>, dim3(grid), dim3(BLOCK),
                       0, stream, (const scalar_t*)dy.data_ptr(),
                       (const scalar_t*)y.data_ptr(),
                       (scalar_t*)dx.data_ptr(), rows, D);
  });
  return dx;
}
